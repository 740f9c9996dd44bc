"""Render the reconciler's internal objects into real Kubernetes v1
manifests.

The reconciler state machine works against the abstract Pod/ConfigMap/
Service/RBACObject model (cluster.py) so it can be driven by the fake
kubelet in tests; this module is the bridge to an actual apiserver — it
produces the same concrete objects the reference controller builds in Go:

  * pods: restartPolicy Never, config/kube/dataset/shm volumes with the
    reference mount paths (/root/reference/controllers/dgljob_controller.go:
    66-72), kubexec.sh mode 0555 / hostfiles 0444 (:961-999), /dev/shm
    emptyDir (Memory) sized to half the container memory limit (:961-974),
    20 container ports 30050-30069 (:950-958), init containers with the
    NAMESPACE/WATCHERFILE/WATCHERMODE env contract (:1100-1194)
  * per-worker headless Services (:496-519)
  * name-scoped RBAC (:1333-1413)

Owner references are attached when the DGLJob uid is known so the apiserver
garbage-collects job objects exactly like the reference's
ctrl.SetControllerReference (:872-893).
"""
from __future__ import annotations

import re
from typing import Any, Dict, List, Optional

from .api import (
    CONFIG_MOUNT,
    GROUP,
    KIND,
    KUBECTL_MOUNT,
    VERSION,
)
from .cluster import ConfigMap, Pod, RBACObject, Service

DATASET_MOUNT = "/dgl_workspace/dataset"  # datasetMountPath (go:72)
SHM_MOUNT = "/dev/shm"

_UNITS = {
    "k": 10**3, "M": 10**6, "G": 10**9, "T": 10**12,
    "Ki": 2**10, "Mi": 2**20, "Gi": 2**30, "Ti": 2**40,
}


def parse_quantity(q: Any) -> int:
    """Kubernetes resource quantity -> bytes/units (int)."""
    if isinstance(q, (int, float)):
        return int(q)
    m = re.fullmatch(r"([0-9.]+)\s*([A-Za-z]*)", str(q).strip())
    if not m:
        return 0
    val = float(m.group(1))
    suffix = m.group(2)
    if suffix in _UNITS:
        return int(val * _UNITS[suffix])
    if suffix == "m":  # millis (cpu) — callers that care handle cpu
        return int(val / 1000)
    return int(val)


def _owner_ref(owner: Optional[str], owner_uid: Optional[str]) -> List[dict]:
    if not owner or not owner_uid:
        return []
    return [{
        "apiVersion": f"{GROUP}/{VERSION}",
        "kind": KIND,
        "name": owner,
        "uid": owner_uid,
        "controller": True,
        "blockOwnerDeletion": True,
    }]


def _meta(name, namespace, labels=None, annotations=None, owner=None,
          owner_uid=None) -> dict:
    meta: Dict[str, Any] = {"name": name, "namespace": namespace}
    if labels:
        meta["labels"] = dict(labels)
    if annotations:
        meta["annotations"] = dict(annotations)
    refs = _owner_ref(owner, owner_uid)
    if refs:
        meta["ownerReferences"] = refs
    return meta


def _env_list(env: Dict[str, str]) -> List[dict]:
    out = []
    for k, v in env.items():
        # the internal model stores WATCHERFILE by key; the container reads
        # the mounted path (dgljob_controller.go:1122,1164)
        if k == "WATCHERFILE" and "/" not in v:
            v = f"{CONFIG_MOUNT}/{v}"
        out.append({"name": k, "value": str(v)})
    return out


_VOLUME_MOUNTS = {
    "config": {"name": "config-volume", "mountPath": CONFIG_MOUNT},
    "kube": {"name": "kubectl-volume", "mountPath": KUBECTL_MOUNT},
    "dataset": {"name": "dataset-volume", "mountPath": DATASET_MOUNT},
    "shm": {"name": "dshm", "mountPath": SHM_MOUNT},
}


def _volumes(symbolic: List[str], cm_name: str, shm_bytes: int) -> List[dict]:
    scripts_mode = 0o555
    hostfile_mode = 0o444
    vols = []
    for s in symbolic:
        if s == "config":
            vols.append({
                "name": "config-volume",
                "configMap": {
                    "name": cm_name,
                    "items": [
                        {"key": "kubexec.sh", "path": "kubexec.sh",
                         "mode": scripts_mode},
                        {"key": "hostfile", "path": "hostfile",
                         "mode": hostfile_mode},
                        {"key": "partfile", "path": "partfile",
                         "mode": hostfile_mode},
                        {"key": "leadfile", "path": "leadfile",
                         "mode": hostfile_mode},
                    ],
                },
            })
        elif s == "kube":
            vols.append({"name": "kubectl-volume", "emptyDir": {}})
        elif s == "dataset":
            vols.append({"name": "dataset-volume", "emptyDir": {}})
        elif s == "shm":
            shm_gb = max(shm_bytes // 10**9, 1)
            vols.append({
                "name": "dshm",
                "emptyDir": {"medium": "Memory",
                             "sizeLimit": f"{shm_gb}G"},
            })
    return vols


def _init_container(ic: Dict[str, Any], namespace: str) -> dict:
    c: Dict[str, Any] = {
        "name": ic["name"],
        "image": ic["image"],
        "imagePullPolicy": "Always",
    }
    env = {"NAMESPACE": namespace, **ic.get("env", {})}
    if ic["name"] == "kubectl-download":
        # only the shared kubectl emptyDir; no watcher env
        c["volumeMounts"] = [_VOLUME_MOUNTS["kube"]]
    else:
        c["env"] = _env_list(env)
        mounts = [_VOLUME_MOUNTS["config"]]
        for extra in ic.get("mounts", []):
            mounts.append(_VOLUME_MOUNTS[extra])
        c["volumeMounts"] = mounts
    if "resources" in ic:
        r = {k: str(v) for k, v in ic["resources"].items()}
        c["resources"] = {"limits": dict(r), "requests": dict(r)}
    return c


def pod_manifest(pod: Pod) -> dict:
    """Internal Pod -> v1 Pod manifest (mirrors buildLauncherPod /
    buildWorkerOrPartitionerPod, dgljob_controller.go:897-1317)."""
    spec = pod.spec or {}
    cm_name = f"{pod.owner}-config" if pod.owner else "dgl-config"
    symbolic = list(spec.get("volumes", []))
    containers = [dict(c) for c in spec.get("containers", [])]
    if not containers:
        containers = [{"name": "main", "image": "busybox"}]
    # pod-level command/args override (partitioner takes the launcher's)
    for k in ("command", "args"):
        if k in spec and spec[k] is not None:
            containers[0][k] = spec[k]

    # shm sizing from the first container's memory limit (go:961)
    shm_bytes = 0
    if "shm" in symbolic:
        mem = (containers[0].get("resources", {}).get("limits", {})
               .get("memory", "2Gi"))
        frac = spec.get("shmSizeFraction", 0.5)
        shm_bytes = int(parse_quantity(mem) * frac)

    env = _env_list(spec.get("env", {}))
    ports = [{"name": f"dgl-port-{i}", "containerPort": p, "protocol": "TCP"}
             for i, p in enumerate(spec.get("ports", []))]
    mounts = [_VOLUME_MOUNTS[s] for s in symbolic if s in _VOLUME_MOUNTS]
    for c in containers:
        c["env"] = list(c.get("env", [])) + env
        c["volumeMounts"] = list(c.get("volumeMounts", [])) + mounts
        if ports:
            c.setdefault("ports", ports)

    pod_spec: Dict[str, Any] = {
        "restartPolicy": "Never",  # go:923,1096
        "containers": containers,
        "volumes": _volumes(symbolic, cm_name, shm_bytes),
    }
    inits = spec.get("initContainers")
    if inits:
        pod_spec["initContainers"] = [
            _init_container(ic, pod.namespace) for ic in inits
        ]
    if spec.get("serviceAccount"):
        pod_spec["serviceAccountName"] = spec["serviceAccount"]
    for passthrough in ("nodeSelector", "tolerations", "affinity",
                        "hostNetwork", "imagePullSecrets"):
        if passthrough in spec:
            pod_spec[passthrough] = spec[passthrough]
    return {
        "apiVersion": "v1",
        "kind": "Pod",
        "metadata": _meta(pod.name, pod.namespace, pod.labels,
                          pod.annotations, pod.owner,
                          getattr(pod, "owner_uid", None)),
        "spec": pod_spec,
    }


def configmap_manifest(cm: ConfigMap) -> dict:
    return {
        "apiVersion": "v1",
        "kind": "ConfigMap",
        "metadata": _meta(cm.name, cm.namespace, owner=cm.owner,
                          owner_uid=getattr(cm, "owner_uid", None)),
        "data": dict(cm.data),
    }


def service_manifest(svc: Service) -> dict:
    return {
        "apiVersion": "v1",
        "kind": "Service",
        "metadata": _meta(svc.name, svc.namespace, owner=svc.owner,
                          owner_uid=getattr(svc, "owner_uid", None)),
        "spec": {
            "clusterIP": "None" if svc.cluster_ip is None else svc.cluster_ip,
            "selector": dict(svc.selector),
            "ports": [{"name": f"s-port-{i}", "port": p}
                      for i, p in enumerate(svc.ports)],
        },
    }


def rbac_manifest(obj: RBACObject) -> dict:
    meta = _meta(obj.name, obj.namespace,
                 labels={"app": obj.owner} if obj.owner else None,
                 owner=obj.owner, owner_uid=getattr(obj, "owner_uid", None))
    if obj.kind == "ServiceAccount":
        return {"apiVersion": "v1", "kind": "ServiceAccount",
                "metadata": meta}
    if obj.kind == "Role":
        rules = []
        for r in obj.rules:
            rule = {
                "apiGroups": [""],
                "resources": list(r.get("resources", [])),
                "verbs": list(r.get("verbs", [])),
            }
            if r.get("resourceNames"):
                rule["resourceNames"] = list(r["resourceNames"])
            rules.append(rule)
        return {"apiVersion": "rbac.authorization.k8s.io/v1", "kind": "Role",
                "metadata": meta, "rules": rules}
    if obj.kind == "RoleBinding":
        return {
            "apiVersion": "rbac.authorization.k8s.io/v1",
            "kind": "RoleBinding",
            "metadata": meta,
            "subjects": [{"kind": "ServiceAccount", "name": obj.name,
                          "namespace": obj.namespace}],
            "roleRef": {"apiGroup": "rbac.authorization.k8s.io",
                        "kind": "Role", "name": obj.name},
        }
    raise ValueError(f"unknown RBAC kind {obj.kind}")
