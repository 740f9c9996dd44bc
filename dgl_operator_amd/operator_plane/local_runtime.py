"""Local pod runtime: executes DGLJob pods as host processes.

The single-host stand-in for a kubelet + container runtime, completing the
operator loop without Kubernetes: the reconciler creates Pod objects in a
FakeCluster; this runtime materializes the job ConfigMap into each pod's
directory (the /etc/dgl mount), launches pod commands as subprocesses in
those directories (the LocalFabric layout, so kubectl-style exec/cp from
dglrun lands in the right place), honors the launcher's watcher-loop
init-container gating, and reports phase transitions back to the cluster.

This is both the integration-test harness the reference never had
(envtest has no kubelet, the distributed path was only
testable by really deploying) and a way to run DGLJob manifests on a single
8xMI355X box with no cluster at all:

    python -m dgl_operator_amd.operator_plane.local_runtime \
        --job examples/v1alpha1/GraphSAGE_dist.yaml --root /tmp/dgljob
"""
from __future__ import annotations

import os
import subprocess
import time
from typing import Dict, Optional

from .api import CONFIG_MOUNT, DGLJob, JobPhase, PodPhase, ReplicaType
from .cluster import FakeCluster, Pod
from .manager import Manager
from .watcher import parse_watchfile


class LocalPodRuntime:
    def __init__(self, cluster: FakeCluster, root: str,
                 extra_env: Optional[Dict[str, str]] = None):
        self.cluster = cluster
        # pods run with their pod dir as cwd and resolve the fabric root
        # from env — a relative root would resolve against EACH pod's cwd
        self.root = os.path.abspath(root)
        self.procs: Dict[str, subprocess.Popen] = {}
        self.started: set = set()
        self.extra_env = extra_env or {}
        os.makedirs(root, exist_ok=True)

    # -- helpers -----------------------------------------------------------
    def pod_dir(self, pod: Pod) -> str:
        d = os.path.join(self.root, pod.name)
        os.makedirs(d, exist_ok=True)
        return d

    def _materialize_configmaps(self):
        """Write every job ConfigMap into each of its pods' etc_dgl dir —
        the kubelet's live-updated ConfigMap mount."""
        for cm in self.cluster.configmaps.values():
            for pod in self.cluster.list_pods(cm.namespace, cm.owner):
                mount = os.path.join(self.pod_dir(pod), "etc_dgl")
                os.makedirs(mount, exist_ok=True)
                for fname, text in cm.data.items():
                    with open(os.path.join(mount, fname), "w") as f:
                        f.write(text)

    def _launcher_gates_open(self, pod: Pod) -> bool:
        """Replicate the watcher-loop init containers: partfile pods must be
        Succeeded (mode=finished), hostfile pods Running (mode=ready)."""
        mount = os.path.join(self.pod_dir(pod), "etc_dgl")
        part = os.path.join(mount, "partfile")
        if os.path.exists(part):
            with open(part) as f:
                names = parse_watchfile(f.read())
            for n in names:
                p = self.cluster.get_pod(pod.namespace, n)
                if p is None or p.phase != PodPhase.SUCCEEDED:
                    return False
        host = os.path.join(mount, "hostfile")
        names = []
        if os.path.exists(host):
            with open(host) as f:
                names = parse_watchfile(f.read())
        if not names:
            return False  # hostfile not populated yet
        for n in names:
            p = self.cluster.get_pod(pod.namespace, n)
            if p is None or not p.is_real_running():
                return False
        return True

    def _spawn(self, pod: Pod):
        spec = pod.spec
        cmd = list(spec.get("command") or [])
        args = spec.get("args")
        if args is None and spec.get("containers"):
            cont = spec["containers"][0]
            cmd = list(cont.get("command") or cmd)
            args = cont.get("args")
        cmd = cmd + list(args or [])
        env = dict(os.environ)
        env.update(self.extra_env)
        env.update(spec.get("env") or {})
        env["DGL_LOCAL_FABRIC_ROOT"] = self.root
        env.setdefault("WORKSPACE", "workspace")
        proc = subprocess.Popen(cmd, cwd=self.pod_dir(pod), env=env)
        self.procs[pod.name] = proc

    # -- the kubelet tick --------------------------------------------------
    def tick(self):
        self._materialize_configmaps()
        for pod in list(self.cluster.pods.values()):
            is_launcher = pod.labels.get("dgl-replica-type") == "launcher"
            cmd = pod.spec.get("command") or (
                (pod.spec.get("containers") or [{}])[0].get("command")
            )
            sleeper = cmd == ["sleep", "365d"]
            if pod.name not in self.started:
                if sleeper or cmd is None:
                    # worker placeholder: Running immediately
                    self.started.add(pod.name)
                    self.cluster.set_pod_phase(pod.namespace, pod.name,
                                               PodPhase.RUNNING)
                    pod.ip = "127.0.0.1"
                elif is_launcher:
                    # Running (init containers executing), main gated
                    if pod.phase == PodPhase.PENDING:
                        self.cluster.set_pod_phase(
                            pod.namespace, pod.name, PodPhase.RUNNING,
                            containers_ready=False,
                        )
                        pod.ip = "127.0.0.1"
                    if self._launcher_gates_open(pod):
                        self.started.add(pod.name)
                        pod.containers_ready = True
                        self._spawn(pod)
                else:  # partitioner or other one-shot pod
                    self.started.add(pod.name)
                    self.cluster.set_pod_phase(pod.namespace, pod.name,
                                               PodPhase.RUNNING)
                    pod.ip = "127.0.0.1"
                    self._spawn(pod)
            proc = self.procs.get(pod.name)
            if proc is not None and proc.poll() is not None:
                phase = (PodPhase.SUCCEEDED if proc.returncode == 0
                         else PodPhase.FAILED)
                if pod.phase not in (PodPhase.SUCCEEDED, PodPhase.FAILED):
                    self.cluster.set_pod_phase(pod.namespace, pod.name, phase)
                del self.procs[pod.name]

    def shutdown(self):
        for proc in self.procs.values():
            proc.terminate()
        for proc in self.procs.values():
            try:
                proc.wait(10)
            except subprocess.TimeoutExpired:
                proc.kill()


def run_job(manifest: str, root: str, timeout: float = 600.0,
            extra_env: Optional[Dict[str, str]] = None,
            poll: float = 0.3) -> DGLJob:
    """Drive one DGLJob to completion on the local host. Returns the job
    (status.phase is Completed or Failed)."""
    mgr = Manager(reconcile_interval=poll)
    runtime = LocalPodRuntime(mgr.cluster, root, extra_env=extra_env)
    job = mgr.submit(manifest)
    deadline = time.time() + timeout
    last_phase = None
    try:
        while time.time() < deadline:
            mgr.reconcile_once()
            runtime.tick()
            if job.status.phase != last_phase:
                last_phase = job.status.phase
                ready = {rt.value: rs.ready for rt, rs
                         in job.status.replica_statuses.items()}
                print(f"[operator] {job.name} phase -> "
                      f"{last_phase.value if last_phase else None} {ready}",
                      flush=True)
            if job.status.phase in (JobPhase.COMPLETED, JobPhase.FAILED):
                return job
            time.sleep(poll)
        raise TimeoutError(
            f"job {job.name} still {job.status.phase} after {timeout}s"
        )
    finally:
        runtime.shutdown()


def main(argv=None):
    import argparse

    p = argparse.ArgumentParser(prog="dgl-operator-local")
    p.add_argument("--job", required=True, help="DGLJob manifest YAML")
    p.add_argument("--root", default="/tmp/dgl_local_job")
    p.add_argument("--timeout", type=float, default=3600.0)
    args = p.parse_args(argv)
    with open(args.job) as f:
        job = run_job(f.read(), args.root, timeout=args.timeout)
    print(f"job {job.name}: {job.status.phase.value}")
    raise SystemExit(0 if job.status.phase == JobPhase.COMPLETED else 1)


if __name__ == "__main__":
    main()
