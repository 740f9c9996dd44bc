"""Cluster client abstraction + in-memory fake cluster.

The reconciler talks to this interface instead of client-go. Two
implementations:
  * FakeCluster — in-memory objects with a controllable fake kubelet
    (tests drive pod phase transitions), covering what the reference could
    not test (envtest has no kubelet, so the reference's
    phase-ladder assertions never actually run in CI).
  * KubectlCluster — thin shell-out to kubectl for real clusters (optional;
    works wherever a kubeconfig is present).
"""
from __future__ import annotations

import json
import subprocess
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from .api import PodPhase


@dataclass
class Pod:
    name: str
    namespace: str
    labels: Dict[str, str] = field(default_factory=dict)
    annotations: Dict[str, str] = field(default_factory=dict)
    spec: Dict[str, Any] = field(default_factory=dict)
    phase: PodPhase = PodPhase.PENDING
    reason: Optional[str] = None  # e.g. "Evicted"
    ip: Optional[str] = None
    containers_ready: bool = False
    owner: Optional[str] = None  # owning DGLJob name
    owner_uid: Optional[str] = None  # DGLJob uid (k8s ownerReference GC)

    def is_real_running(self) -> bool:
        """Running AND every container ready (dgljob_controller.go:1511-1528)."""
        return self.phase == PodPhase.RUNNING and self.containers_ready


@dataclass
class ConfigMap:
    name: str
    namespace: str
    data: Dict[str, str] = field(default_factory=dict)
    owner: Optional[str] = None
    owner_uid: Optional[str] = None


@dataclass
class Service:
    name: str
    namespace: str
    selector: Dict[str, str] = field(default_factory=dict)
    cluster_ip: Optional[str] = None  # None => headless
    ports: List[int] = field(default_factory=list)
    owner: Optional[str] = None
    owner_uid: Optional[str] = None


@dataclass
class RBACObject:
    kind: str  # ServiceAccount | Role | RoleBinding
    name: str
    namespace: str
    rules: List[Dict[str, Any]] = field(default_factory=list)
    owner: Optional[str] = None
    owner_uid: Optional[str] = None


class Cluster:
    """Interface the reconciler depends on."""

    def get_pod(self, namespace: str, name: str) -> Optional[Pod]:
        raise NotImplementedError

    def list_pods(self, namespace: str, owner: str) -> List[Pod]:
        raise NotImplementedError

    def create_pod(self, pod: Pod) -> Pod:
        raise NotImplementedError

    def delete_pod(self, namespace: str, name: str) -> None:
        raise NotImplementedError

    def get_configmap(self, namespace: str, name: str) -> Optional[ConfigMap]:
        raise NotImplementedError

    def create_configmap(self, cm: ConfigMap) -> ConfigMap:
        raise NotImplementedError

    def update_configmap(self, cm: ConfigMap) -> ConfigMap:
        raise NotImplementedError

    def get_service(self, namespace: str, name: str) -> Optional[Service]:
        raise NotImplementedError

    def create_service(self, svc: Service) -> Service:
        raise NotImplementedError

    def get_rbac(self, namespace: str, kind: str, name: str) -> Optional[RBACObject]:
        raise NotImplementedError

    def create_rbac(self, obj: RBACObject) -> RBACObject:
        raise NotImplementedError


class FakeCluster(Cluster):
    """In-memory cluster with a driveable kubelet."""

    def __init__(self):
        self.pods: Dict[str, Pod] = {}
        self.configmaps: Dict[str, ConfigMap] = {}
        self.services: Dict[str, Service] = {}
        self.rbac: Dict[str, RBACObject] = {}
        self._next_ip = 2
        self._watchers: List[Any] = []  # event queues (watch_pods)

    @staticmethod
    def _key(namespace: str, name: str) -> str:
        return f"{namespace}/{name}"

    # -- pods --------------------------------------------------------------
    def get_pod(self, namespace, name):
        return self.pods.get(self._key(namespace, name))

    def list_pods(self, namespace, owner):
        return [
            p for p in self.pods.values()
            if p.namespace == namespace and p.owner == owner
        ]

    def create_pod(self, pod):
        k = self._key(pod.namespace, pod.name)
        if k in self.pods:
            raise RuntimeError(f"pod {k} already exists")
        self.pods[k] = pod
        return pod

    def delete_pod(self, namespace, name):
        self.pods.pop(self._key(namespace, name), None)

    # -- fake kubelet ------------------------------------------------------
    def set_pod_phase(self, namespace, name, phase: PodPhase,
                      containers_ready: Optional[bool] = None,
                      reason: Optional[str] = None):
        p = self.pods[self._key(namespace, name)]
        p.phase = phase
        p.reason = reason
        if phase == PodPhase.RUNNING and p.ip is None:
            p.ip = f"10.244.0.{self._next_ip}"
            self._next_ip += 1
        if containers_ready is None:
            containers_ready = phase == PodPhase.RUNNING
        p.containers_ready = containers_ready
        for q in self._watchers:  # pod UPDATE event to every subscriber
            q.put(p)

    def watch_pods(self, namespace: str):
        """Pod event stream (informer UpdateFunc analog): yields a Pod on
        every phase transition, None as a 0.2 s keep-alive heartbeat."""
        import queue as _queue

        q: "_queue.Queue" = _queue.Queue()
        self._watchers.append(q)
        try:
            while True:
                try:
                    ev = q.get(timeout=0.2)
                except _queue.Empty:
                    yield None
                    continue
                if ev.namespace == namespace:
                    yield ev
        finally:
            self._watchers.remove(q)

    def run_all_pending(self, namespace: Optional[str] = None):
        for p in list(self.pods.values()):
            if namespace and p.namespace != namespace:
                continue
            if p.phase == PodPhase.PENDING:
                self.set_pod_phase(p.namespace, p.name, PodPhase.RUNNING)

    # -- configmaps --------------------------------------------------------
    def get_configmap(self, namespace, name):
        return self.configmaps.get(self._key(namespace, name))

    def create_configmap(self, cm):
        self.configmaps[self._key(cm.namespace, cm.name)] = cm
        return cm

    def update_configmap(self, cm):
        self.configmaps[self._key(cm.namespace, cm.name)] = cm
        return cm

    # -- services ----------------------------------------------------------
    def get_service(self, namespace, name):
        return self.services.get(self._key(namespace, name))

    def create_service(self, svc):
        self.services[self._key(svc.namespace, svc.name)] = svc
        return svc

    # -- rbac ---------------------------------------------------------------
    def get_rbac(self, namespace, kind, name):
        return self.rbac.get(f"{namespace}/{kind}/{name}")

    def create_rbac(self, obj):
        self.rbac[f"{obj.namespace}/{obj.kind}/{obj.name}"] = obj
        return obj


class KubectlCluster(Cluster):
    """kubectl-backed Cluster: the real-apiserver implementation of every
    reconciler dependency. Object writes render the internal model into v1
    manifests (k8s.py) and pipe them to ``kubectl create/apply -f -``;
    reads parse ``-o json``. Also carries the DGLJob CR surface the Manager
    watch loop uses (list/patch-status), replacing the reference's
    controller-runtime informer plumbing
    (/root/reference/main.go:73-105, dgljob_controller.go:436-458) with a
    polling loop over the same objects. Requires a kubeconfig (in-cluster
    service account or --kubeconfig)."""

    def __init__(self, kubectl: str = "kubectl"):
        self.kubectl = kubectl

    # -- plumbing ----------------------------------------------------------
    def _run(self, args, stdin: "str | None" = None):
        return subprocess.run(
            [self.kubectl] + args, input=stdin, capture_output=True,
            text=True,
        )

    def _get_json(self, args):
        r = self._run(args + ["-o", "json"])
        if r.returncode != 0:
            return None
        return json.loads(r.stdout)

    def _create(self, manifest: dict):
        r = self._run(["create", "-f", "-"], stdin=json.dumps(manifest))
        if r.returncode != 0 and "AlreadyExists" not in r.stderr:
            raise RuntimeError(
                f"kubectl create {manifest.get('kind')} "
                f"{manifest.get('metadata', {}).get('name')}: {r.stderr.strip()}"
            )

    def _apply(self, manifest: dict):
        r = self._run(["apply", "-f", "-"], stdin=json.dumps(manifest))
        if r.returncode != 0:
            raise RuntimeError(
                f"kubectl apply {manifest.get('kind')} "
                f"{manifest.get('metadata', {}).get('name')}: {r.stderr.strip()}"
            )

    def get_pod(self, namespace, name):
        d = self._get_json(["get", "pod", name, "-n", namespace])
        if d is None:
            return None
        status = d.get("status", {})
        ready = all(
            cs.get("ready", False)
            for cs in status.get("containerStatuses", [{"ready": False}])
        )
        return Pod(
            name=name,
            namespace=namespace,
            labels=d.get("metadata", {}).get("labels", {}),
            phase=PodPhase(status.get("phase", "Unknown")),
            reason=status.get("reason"),
            ip=status.get("podIP"),
            containers_ready=ready,
        )

    def list_pods(self, namespace, owner):
        d = self._get_json(
            ["get", "pods", "-n", namespace, "-l", f"dgl-job-name={owner}"]
        )
        if d is None:
            return []
        return [
            self.get_pod(namespace, item["metadata"]["name"])
            for item in d.get("items", [])
        ]

    def create_pod(self, pod):
        from .k8s import pod_manifest

        self._create(pod_manifest(pod))
        return pod

    def delete_pod(self, namespace, name):
        subprocess.run(
            [self.kubectl, "delete", "pod", name, "-n", namespace,
             "--ignore-not-found", "--wait=false"],
            capture_output=True,
        )

    # -- configmaps --------------------------------------------------------
    def get_configmap(self, namespace, name):
        d = self._get_json(["get", "configmap", name, "-n", namespace])
        if d is None:
            return None
        return ConfigMap(name=name, namespace=namespace,
                         data=d.get("data") or {})

    def create_configmap(self, cm):
        from .k8s import configmap_manifest

        self._create(configmap_manifest(cm))
        return cm

    def update_configmap(self, cm):
        from .k8s import configmap_manifest

        self._apply(configmap_manifest(cm))
        return cm

    # -- services ----------------------------------------------------------
    def get_service(self, namespace, name):
        d = self._get_json(["get", "service", name, "-n", namespace])
        if d is None:
            return None
        spec = d.get("spec", {})
        cip = spec.get("clusterIP")
        return Service(
            name=name, namespace=namespace,
            selector=spec.get("selector") or {},
            cluster_ip=None if cip == "None" else cip,
            ports=[p.get("port") for p in spec.get("ports", [])],
        )

    def create_service(self, svc):
        from .k8s import service_manifest

        self._create(service_manifest(svc))
        return svc

    # -- rbac ----------------------------------------------------------------
    _RBAC_RESOURCE = {"ServiceAccount": "serviceaccount", "Role": "role",
                      "RoleBinding": "rolebinding"}

    def get_rbac(self, namespace, kind, name):
        d = self._get_json(
            ["get", self._RBAC_RESOURCE[kind], name, "-n", namespace])
        if d is None:
            return None
        return RBACObject(kind=kind, name=name, namespace=namespace,
                          rules=d.get("rules") or [])

    def create_rbac(self, obj):
        from .k8s import rbac_manifest

        self._create(rbac_manifest(obj))
        return obj

    def watch_pods(self, namespace: str):
        """Pod event stream over ``kubectl get pods --watch -o name``
        (the informer analog for real clusters): each change event names a
        pod, whose state is then fetched — robust against arbitrary JSON
        in annotations, at the cost of one extra get per event. Yields a
        Pod per event, None heartbeats while the stream is quiet."""
        import select

        proc = subprocess.Popen(
            [self.kubectl, "get", "pods", "-n", namespace, "--watch",
             "-o", "name"],
            stdout=subprocess.PIPE, stderr=subprocess.DEVNULL, text=True,
        )
        try:
            while True:
                r, _, _ = select.select([proc.stdout], [], [], 0.5)
                if not r:
                    if proc.poll() is not None:
                        return
                    yield None
                    continue
                line = proc.stdout.readline()
                if not line:
                    return
                name = line.strip().rsplit("/", 1)[-1]
                if not name:
                    continue
                pod = self.get_pod(namespace, name)
                if pod is not None:
                    yield pod
        finally:
            proc.terminate()

    # -- leader election (coordination.k8s.io Lease; main.go:73-80 parity) --
    def try_acquire_lease(self, namespace: str, name: str, identity: str,
                          ttl_seconds: int = 15) -> bool:
        """Acquire or renew the leader Lease. Returns True while this
        identity holds it. Same semantics as controller-runtime's
        leasing resource lock: a lease held by another identity is only
        taken over once renewTime is older than the ttl."""
        import datetime

        now = datetime.datetime.now(datetime.timezone.utc)
        now_s = now.strftime("%Y-%m-%dT%H:%M:%S.%f")[:-3] + "Z"
        d = self._get_json(["get", "lease", name, "-n", namespace])
        if d is None:
            try:
                self._create({
                    "apiVersion": "coordination.k8s.io/v1",
                    "kind": "Lease",
                    "metadata": {"name": name, "namespace": namespace},
                    "spec": {
                        "holderIdentity": identity,
                        "leaseDurationSeconds": ttl_seconds,
                        "acquireTime": now_s,
                        "renewTime": now_s,
                    },
                })
                return True
            except RuntimeError:
                return False  # lost the creation race
        spec = d.get("spec", {})
        holder = spec.get("holderIdentity")
        if holder and holder != identity:
            renew = spec.get("renewTime") or spec.get("acquireTime")
            if renew:
                try:
                    t = datetime.datetime.strptime(
                        renew.replace("Z", "+0000"),
                        "%Y-%m-%dT%H:%M:%S.%f%z")
                except ValueError:
                    t = datetime.datetime.strptime(
                        renew.replace("Z", "+0000"), "%Y-%m-%dT%H:%M:%S%z")
                if (now - t).total_seconds() < spec.get(
                        "leaseDurationSeconds", ttl_seconds):
                    return False  # current leader is live
        # take over / renew
        patch = json.dumps({"spec": {
            "holderIdentity": identity,
            "leaseDurationSeconds": ttl_seconds,
            "renewTime": now_s,
        }})
        r = self._run(["patch", "lease", name, "-n", namespace,
                       "--type=merge", "-p", patch])
        return r.returncode == 0

    # -- DGLJob CRs (the Manager watch surface) -----------------------------
    def list_dgljob_manifests(self, namespace: "str | None" = None):
        """Raw DGLJob CR dicts (all namespaces unless one is given)."""
        args = ["get", "dgljobs"]
        args += ["-n", namespace] if namespace else ["-A"]
        d = self._get_json(args)
        if d is None:
            return []
        return d.get("items", [])

    def update_dgljob_status(self, namespace: str, name: str,
                             status: dict) -> bool:
        """Write .status back (Status().Update parity,
        dgljob_controller.go:304-315). Tries the status subresource first
        (the CRD declares one); falls back to a plain merge patch."""
        patch = json.dumps({"status": status})
        r = self._run(["patch", "dgljob", name, "-n", namespace,
                       "--subresource=status", "--type=merge", "-p", patch])
        if r.returncode != 0:
            r = self._run(["patch", "dgljob", name, "-n", namespace,
                           "--type=merge", "-p", patch])
        return r.returncode == 0
