"""Remote-exec fabric: how the launcher reaches worker pods.

The reference replaces ssh with `kubectl exec/cp` through a ConfigMap-mounted
kubexec.sh (/root/reference/python/dglrun/tools/launch.py:14-50). Here the
fabric is pluggable:
  * KubexecFabric — same contract as the reference (kubexec.sh + kubectl cp),
    for real cluster deployments.
  * LocalFabric  — pods are directories on one host (the single-node
    8xMI355X deployment and the test environment): exec = subprocess with
    cwd, cp = file copy. This is also what makes the workflow testable
    without a cluster (SURVEY.md §4 gap).
"""
from __future__ import annotations

import os
import shlex
import shutil
import subprocess
from dataclasses import dataclass, field
from typing import Dict, Optional


class Fabric:
    def exec(self, pod: str, command: str, env: Optional[Dict[str, str]] = None,
             block: bool = True):
        raise NotImplementedError

    def copy(self, local_path: str, pod: str, remote_path: str,
             container: Optional[str] = None):
        raise NotImplementedError


@dataclass
class KubexecFabric(Fabric):
    """Shell out through the operator's kubexec.sh / kubectl, path contract
    from the env vars DGL_OPERATOR_KUBEXEC_PATH / DGL_OPERATOR_KUBECTL_PATH
    (dgljob_controller.go:58-63)."""

    kubexec_path: str = field(
        default_factory=lambda: os.environ.get(
            "DGL_OPERATOR_KUBEXEC_PATH", "/etc/dgl/kubexec.sh"
        )
    )
    kubectl_path: str = field(
        default_factory=lambda: os.environ.get(
            "DGL_OPERATOR_KUBECTL_PATH", "/opt/kube/kubectl"
        )
    )

    def exec(self, pod, command, env=None, block=True):
        envs = " ".join(f"{k}={shlex.quote(v)}" for k, v in (env or {}).items())
        cmd = f"{self.kubexec_path} {pod} {envs} {command}"
        proc = subprocess.Popen(cmd, shell=True)
        if block:
            rc = proc.wait()
            if rc != 0:
                raise RuntimeError(f"exec on {pod} failed rc={rc}: {command}")
            return rc
        return proc

    def copy(self, local_path, pod, remote_path, container=None):
        c = f" -c {container}" if container else ""
        cmd = f"{self.kubectl_path} cp {local_path} {pod}:{remote_path}{c}"
        rc = subprocess.call(cmd, shell=True)
        if rc != 0:
            raise RuntimeError(f"kubectl cp to {pod} failed rc={rc}")


@dataclass
class LocalFabric(Fabric):
    """Pods are directories under ``root``; exec runs with that cwd."""

    root: str

    def pod_dir(self, pod: str) -> str:
        d = os.path.join(self.root, pod)
        os.makedirs(d, exist_ok=True)
        return d

    def exec(self, pod, command, env=None, block=True):
        full_env = dict(os.environ)
        full_env.update(env or {})
        proc = subprocess.Popen(
            command, shell=True, cwd=self.pod_dir(pod), env=full_env
        )
        if block:
            rc = proc.wait()
            if rc != 0:
                raise RuntimeError(f"exec on {pod} failed rc={rc}: {command}")
            return rc
        return proc

    def copy(self, local_path, pod, remote_path, container=None):
        dst = os.path.join(self.pod_dir(pod), remote_path.lstrip("/"))
        os.makedirs(os.path.dirname(dst), exist_ok=True)
        if os.path.isdir(local_path):
            shutil.copytree(local_path, dst, dirs_exist_ok=True)
        else:
            shutil.copy2(local_path, dst)


def get_fabric() -> Fabric:
    """Pick the fabric from the operator env contract."""
    if os.environ.get("DGL_OPERATOR_ENV") == "1" and os.path.exists(
        os.environ.get("DGL_OPERATOR_KUBEXEC_PATH", "/etc/dgl/kubexec.sh")
    ):
        return KubexecFabric()
    return LocalFabric(os.environ.get("DGL_LOCAL_FABRIC_ROOT", "/tmp/dgl_pods"))
