"""Operator manager process — the main.go equivalent
(/root/reference/main.go:52-105): flag surface (--metrics-bind-address,
--health-probe-bind-address, --reconcile-interval), Prometheus metrics,
healthz/readyz endpoints, and the reconcile loop over submitted DGLJobs.

controller-runtime's informer machinery is k8s-specific; the observable
contract — every job reconciled on change/interval, metrics exported,
health probes — is implemented against the Cluster abstraction so the same
manager drives the FakeCluster in tests and a kubectl-backed cluster in a
real deployment.
"""
from __future__ import annotations

import argparse
import os
import threading
import time
from http.server import BaseHTTPRequestHandler, HTTPServer
from typing import Dict, Optional

from .api import DGLJob, JobPhase, job_from_manifest
from .cluster import Cluster, FakeCluster
from .reconciler import DGLJobReconciler

try:
    from prometheus_client import Counter, Gauge, Histogram, start_http_server

    _PROM = True
    RECONCILE_TOTAL = Counter(
        "dgljob_reconcile_total", "Total DGLJob reconciles", ["job"]
    )
    RECONCILE_ERRORS = Counter(
        "dgljob_reconcile_errors_total", "Failed reconciles", ["job"]
    )
    RECONCILE_SECONDS = Histogram(
        "dgljob_reconcile_duration_seconds", "Reconcile wall time"
    )
    JOB_PHASE = Gauge("dgljob_phase", "Job phase (1=active)", ["job", "phase"])
    IS_LEADER = Gauge("dgljob_manager_is_leader",
                      "1 while this replica holds the leader lease")
except ImportError:  # pragma: no cover
    _PROM = False


class _HealthHandler(BaseHTTPRequestHandler):
    manager: "Manager" = None

    def do_GET(self):  # noqa: N802
        ok = self.path in ("/healthz", "/readyz") and (
            self.path == "/healthz" or self.manager.ready
        )
        self.send_response(200 if ok else 503)
        self.end_headers()
        self.wfile.write(b"ok" if ok else b"not ready")

    def log_message(self, *a):  # silence
        pass


LEADER_ELECTION_ID = "9007c5fc.qihoo.net"  # main.go:76


class Manager:
    def __init__(self, cluster: Optional[Cluster] = None,
                 reconcile_interval: float = 0.5,
                 watcher_loop_image: str = "watcher-loop",
                 kubectl_download_image: str = "kubectl-download",
                 leader_elect: bool = False,
                 election_namespace: str = "dgl-operator",
                 lease_ttl: float = 15.0):
        self.cluster = cluster or FakeCluster()
        self.reconciler = DGLJobReconciler(
            self.cluster, watcher_loop_image, kubectl_download_image
        )
        self.jobs: Dict[str, DGLJob] = {}
        self.interval = reconcile_interval
        self.ready = False
        self._stop = threading.Event()
        # real-cluster mode: poll DGLJob CRs + write status back
        self.watch_crs = hasattr(self.cluster, "list_dgljob_manifests")
        self._written_statuses: Dict[str, dict] = {}
        # leader election (reference main.go:73-80): only the Lease holder
        # reconciles; replicas keep contending every loop and take over
        # when the holder stops renewing for a ttl
        self.leader_elect = (leader_elect
                             and hasattr(self.cluster, "try_acquire_lease"))
        self.election_namespace = election_namespace
        self.lease_ttl = lease_ttl
        import socket

        self.identity = f"{socket.gethostname()}_{os.getpid()}"
        self.is_leader = not self.leader_elect

    def _ensure_leadership(self) -> bool:
        if not self.leader_elect:
            return True
        was = self.is_leader
        self.is_leader = self.cluster.try_acquire_lease(
            self.election_namespace, LEADER_ELECTION_ID, self.identity,
            int(self.lease_ttl))
        if self.is_leader != was:
            print(f"[manager] leadership {'acquired' if self.is_leader else 'lost'}"
                  f" ({self.identity})", flush=True)
        if _PROM:
            IS_LEADER.set(1.0 if self.is_leader else 0.0)
        return self.is_leader

    # -- job API (the CRD surface) -----------------------------------------
    def submit(self, manifest) -> DGLJob:
        job = manifest if isinstance(manifest, DGLJob) else job_from_manifest(manifest)
        self.jobs[f"{job.namespace}/{job.name}"] = job
        return job

    def delete(self, namespace: str, name: str):
        job = self.jobs.get(f"{namespace}/{name}")
        if job:
            job.deletion_timestamp = time.time()

    def get(self, namespace: str, name: str) -> Optional[DGLJob]:
        return self.jobs.get(f"{namespace}/{name}")

    # -- DGLJob CR watch (real-cluster mode) -------------------------------
    # The reference registers an informer on DGLJobs + owned Pods
    # (main.go:73-105, dgljob_controller.go:436-458); here the same contract
    # — every CR change observed, status written back — is met by polling
    # the apiserver through the Cluster each loop iteration.
    def sync_from_cluster(self):
        lister = getattr(self.cluster, "list_dgljob_manifests", None)
        if lister is None:
            return
        seen = set()
        for item in lister():
            meta = item.get("metadata", {})
            key = f"{meta.get('namespace', 'default')}/{meta.get('name')}"
            seen.add(key)
            cur = self.jobs.get(key)
            if cur is None:
                self.jobs[key] = job_from_manifest(item)
            else:
                fresh = job_from_manifest(item)
                # spec is mutable server-side; status/start-time are ours
                cur.spec = fresh.spec
                cur.uid = fresh.uid or cur.uid
                if fresh.deletion_timestamp and not cur.deletion_timestamp:
                    cur.deletion_timestamp = fresh.deletion_timestamp
        # CRs deleted server-side: clean their pods then drop them
        for key, job in list(self.jobs.items()):
            if key not in seen and job.deletion_timestamp is None:
                job.deletion_timestamp = time.time()

    def _write_status(self, job: DGLJob):
        writer = getattr(self.cluster, "update_dgljob_status", None)
        if writer is None:
            return
        from .api import status_to_manifest

        st = status_to_manifest(job)
        prev = self._written_statuses.get(f"{job.namespace}/{job.name}")
        if st != prev:  # only on change, like Status().Update on diff
            if writer(job.namespace, job.name, st):
                self._written_statuses[f"{job.namespace}/{job.name}"] = st

    # -- reconcile loop ----------------------------------------------------
    def reconcile_once(self):
        if not self._ensure_leadership():
            return
        if self.watch_crs:
            self.sync_from_cluster()
        for key, job in list(self.jobs.items()):
            t0 = time.time()
            try:
                self.reconciler.reconcile(job)
                if _PROM:
                    RECONCILE_TOTAL.labels(job=job.name).inc()
                    RECONCILE_SECONDS.observe(time.time() - t0)
                    for ph in JobPhase:
                        JOB_PHASE.labels(job=job.name, phase=ph.value).set(
                            1.0 if job.status.phase == ph else 0.0
                        )
            except Exception as e:  # noqa: BLE001
                # a failing job must not stall the loop for other jobs —
                # the reference requeues on error (Reconcile returns err)
                if _PROM:
                    RECONCILE_ERRORS.labels(job=job.name).inc()
                import traceback

                print(f"[manager] reconcile error for {key}: {e}")
                traceback.print_exc()
                continue
            if job.deletion_timestamp is not None:
                del self.jobs[key]
                self._written_statuses.pop(key, None)
            elif self.watch_crs:
                self._write_status(job)

    def run(self, metrics_port: Optional[int] = None,
            health_port: Optional[int] = None, block: bool = True):
        if metrics_port and _PROM:
            start_http_server(metrics_port)
        health_server = None
        if health_port:
            _HealthHandler.manager = self
            health_server = HTTPServer(("127.0.0.1", health_port), _HealthHandler)
            threading.Thread(target=health_server.serve_forever,
                             daemon=True).start()
        self.ready = True
        if not block:
            threading.Thread(target=self._loop, daemon=True).start()
            return
        self._loop()

    def _loop(self):
        while not self._stop.is_set():
            self.reconcile_once()
            self._stop.wait(self.interval)

    def stop(self):
        self._stop.set()


def main(argv=None):
    p = argparse.ArgumentParser(prog="dgl-operator-manager")
    p.add_argument("--metrics-bind-address", default=":8080")
    p.add_argument("--health-probe-bind-address", default=":8081")
    p.add_argument("--reconcile-interval", type=float, default=0.5)
    p.add_argument("--leader-elect", action="store_true",
                   help="contend for the leader Lease "
                        f"(id {LEADER_ELECTION_ID}); only the holder "
                        "reconciles — multi-replica deployments")
    p.add_argument("--election-namespace", default="dgl-operator")
    p.add_argument("--watcher-loop-image", default="watcher-loop")
    p.add_argument("--kubectl-download-image", default="kubectl-download")
    p.add_argument("--job", action="append", default=[],
                   help="DGLJob manifest YAML file(s) to manage "
                        "(local mode, no apiserver)")
    p.add_argument("--kubectl", default=None, metavar="PATH",
                   help="run against a real cluster: watch DGLJob CRs and "
                        "reconcile through this kubectl binary")
    args = p.parse_args(argv)
    cluster = None
    if args.kubectl or (not args.job):
        # deployed mode (deploy/v1alpha1/dgl-operator.yaml runs with no
        # --job files): drive the real apiserver via kubectl; fall back to
        # the in-memory cluster only when kubectl cannot reach a cluster
        from shutil import which

        from .cluster import KubectlCluster

        kubectl = args.kubectl or which("kubectl")
        if kubectl:
            probe = KubectlCluster(kubectl)
            if probe._run(["version", "--request-timeout=5s"]).returncode == 0:
                cluster = probe
            elif args.kubectl:
                raise SystemExit(
                    f"--kubectl {args.kubectl}: cannot reach a cluster")
    mgr = Manager(cluster=cluster,
                  reconcile_interval=args.reconcile_interval,
                  watcher_loop_image=args.watcher_loop_image,
                  kubectl_download_image=args.kubectl_download_image,
                  leader_elect=args.leader_elect,
                  election_namespace=args.election_namespace)
    mode = "cluster (DGLJob CR watch)" if mgr.watch_crs else "local --job files"
    print(f"[manager] mode: {mode}", flush=True)
    for path in args.job:
        with open(path) as f:
            mgr.submit(f.read())
    mgr.run(
        metrics_port=int(args.metrics_bind_address.rsplit(":", 1)[1]),
        health_port=int(args.health_probe_bind_address.rsplit(":", 1)[1]),
    )


if __name__ == "__main__":
    main()
