"""DGLJob reconciler — the control plane state machine.

Faithful re-implementation of the reference's Reconcile sequence
(/root/reference/controllers/dgljob_controller.go:105-318) against the
Cluster abstraction:

  fetch job -> terminating/terminated + cleanPodPolicy handling -> set
  StartTime -> default partitioner (DGL-API/ParMETIS) -> get/create ConfigMap
  (kubexec.sh + hostfile/partfile/leadfile rewritten as pod IPs appear) ->
  per-job RBAC -> launcher pod (3 init containers) -> partitioner pod ->
  [phase Partitioned/Training] workers + per-worker headless Services ->
  recompute status via gen_job_phase.

Deliberate fixes over the reference (quirks documented in SURVEY.md §2.3):
  * Skip mode reaches Training/Completed (upstream leaves status Pending
    forever because no Partitioner spec is defaulted).
  * ParMETIS mode is actually wired (runs our native partitioner pod) instead
    of being declared-but-dead.
  * Partitioner replica status is initialized like the other two.
"""
from __future__ import annotations

import copy
import time
from typing import Dict, List

from .api import (
    CONFIG_MOUNT,
    DGL_PORT,
    ENV_HOSTFILE_PATH,
    ENV_KUBECTL_PATH,
    ENV_KUBEXEC_PATH,
    ENV_OPERATOR,
    ENV_PHASE,
    HOST_PORT_NUM,
    JOB_NAME_LABEL,
    KUBECTL_MOUNT,
    REPLICA_INDEX_ANNOTATION,
    REPLICA_NAME_LABEL,
    REPLICA_TYPE_LABEL,
    CleanPodPolicy,
    DGLJob,
    JobPhase,
    PartitionMode,
    PodPhase,
    ReplicaSpec,
    ReplicaStatus,
    ReplicaType,
)
from .cluster import Cluster, ConfigMap, Pod, RBACObject, Service

KUBEXEC_SH = """#!/bin/sh
set -x
POD_NAME=$1
shift
%s exec ${POD_NAME} -- /bin/sh -c "$*"
""" % (KUBECTL_MOUNT + "/kubectl")


class DGLJobReconciler:
    def __init__(self, cluster: Cluster,
                 watcher_loop_image: str = "watcher-loop",
                 kubectl_download_image: str = "kubectl-download"):
        self.cluster = cluster
        # operator flags --watcher-loop-image / --kubectl-download-image
        # (main.go:52-69)
        self.watcher_loop_image = watcher_loop_image
        self.kubectl_download_image = kubectl_download_image

    # ------------------------------------------------------------------
    def reconcile(self, job: DGLJob) -> DGLJob:
        c = self.cluster
        # -- terminating / terminated --------------------------------------
        if job.deletion_timestamp is not None:
            self._cleanup(job, CleanPodPolicy.ALL)
            return job
        if job.status.phase in (JobPhase.COMPLETED, JobPhase.FAILED):
            self._cleanup(job, job.spec.clean_pod_policy)
            return job

        # -- eviction retry (dgljob_controller.go:146-172): an EVICTED
        # launcher of an incomplete job is deleted and recreated below,
        # instead of failing the job ------------------------------------
        launcher = c.get_pod(job.namespace, job.launcher_name())
        if (
            launcher is not None
            and launcher.phase == PodPhase.FAILED
            and launcher.reason == "Evicted"
        ):
            c.delete_pod(job.namespace, launcher.name)

        if job.status.start_time is None:
            job.status.start_time = time.time()

        # -- default partitioner spec (DGL-API and ParMETIS modes) ---------
        needs_partitioner = job.spec.partition_mode in (
            PartitionMode.DGL_API,
            PartitionMode.PARMETIS,
        )
        if needs_partitioner and ReplicaType.PARTITIONER not in job.spec.replica_specs:
            worker = job.spec.replica_specs.get(ReplicaType.WORKER)
            job.spec.replica_specs[ReplicaType.PARTITIONER] = ReplicaSpec(
                replicas=1,
                template=copy.deepcopy(worker.template) if worker else {},
            )

        # -- configmap ------------------------------------------------------
        self._ensure_configmap(job)
        # -- RBAC -----------------------------------------------------------
        self._ensure_rbac(job, needs_partitioner)
        # -- launcher -------------------------------------------------------
        self._ensure_launcher(job)
        # -- partitioner ----------------------------------------------------
        if needs_partitioner:
            self._ensure_partitioner(job)
        # -- workers gated on phase ----------------------------------------
        phase = self._gen_phase(job)
        workers_due = (
            phase in (JobPhase.PARTITIONED, JobPhase.TRAINING)
            or job.spec.partition_mode == PartitionMode.SKIP
        )
        if workers_due:
            self._ensure_workers(job)
        # refresh hostfiles with current IPs
        self._update_files_in_configmap(job)
        # -- status ---------------------------------------------------------
        self._update_status(job)
        return job

    # ------------------------------------------------------------------
    def _pods(self, job: DGLJob) -> Dict[str, Pod]:
        return {p.name: p for p in self.cluster.list_pods(job.namespace, job.name)}

    def _cleanup(self, job: DGLJob, policy: CleanPodPolicy):
        if policy == CleanPodPolicy.NONE:
            return
        for p in self.cluster.list_pods(job.namespace, job.name):
            if policy == CleanPodPolicy.RUNNING and p.phase in (
                PodPhase.SUCCEEDED,
                PodPhase.FAILED,
            ):
                continue  # Running policy keeps already-finished pods
            self.cluster.delete_pod(p.namespace, p.name)

    # -- configmap ------------------------------------------------------
    def cm_name(self, job: DGLJob) -> str:
        return f"{job.name}-config"

    def _ensure_configmap(self, job: DGLJob):
        cm = self.cluster.get_configmap(job.namespace, self.cm_name(job))
        if cm is None:
            # all four keys exist from creation: the config volume's items
            # reference them, and a kubelet fails the mount for a missing
            # key (they fill in as pod IPs appear)
            cm = ConfigMap(
                name=self.cm_name(job),
                namespace=job.namespace,
                data={"kubexec.sh": KUBEXEC_SH, "hostfile": "",
                      "partfile": "", "leadfile": ""},
                owner=job.name,
                owner_uid=job.uid or None,
            )
            self.cluster.create_configmap(cm)

    def _update_files_in_configmap(self, job: DGLJob):
        """hostfile/partfile/leadfile: `ip port podname slots=N` per running
        pod, sorted by name (dgljob_controller.go:1416-1469)."""
        cm = self.cluster.get_configmap(job.namespace, self.cm_name(job))
        if cm is None:
            return
        pods = self._pods(job)
        host_lines = []
        for i in range(job.num_workers()):
            p = pods.get(job.worker_name(i))
            if p is not None and p.ip:
                host_lines.append(
                    f"{p.ip} {DGL_PORT} {p.name} slots={job.spec.slots_per_worker}"
                )
        new = dict(cm.data)
        new["hostfile"] = "\n".join(host_lines) + ("\n" if host_lines else "")
        # partfile/leadfile are the reference's 3-column format (no slots=,
        # dgljob_controller.go:1440-1469)
        part = pods.get(job.partitioner_name())
        new["partfile"] = (
            f"{part.ip} {DGL_PORT} {part.name}\n" if part and part.ip else ""
        )
        lead = pods.get(job.launcher_name())
        new["leadfile"] = (
            f"{lead.ip} {DGL_PORT} {lead.name}\n" if lead and lead.ip else ""
        )
        if new != cm.data:  # update in place only on change (go:1431-1436)
            cm.data = new
            self.cluster.update_configmap(cm)

    # -- RBAC -----------------------------------------------------------
    def _ensure_rbac(self, job: DGLJob, needs_partitioner: bool):
        """Launcher Role: get/list/watch pods + create pods/exec restricted to
        the worker pod names; partitioner Role: exec only into the launcher
        (dgljob_controller.go:1333-1413)."""
        c = self.cluster
        worker_names = [job.worker_name(i) for i in range(job.num_workers())]
        objs = [
            RBACObject("ServiceAccount", f"{job.name}-launcher", job.namespace,
                       owner=job.name),
            RBACObject(
                "Role",
                f"{job.name}-launcher",
                job.namespace,
                rules=[
                    {"resources": ["pods"], "verbs": ["get", "list", "watch"]},
                    {
                        "resources": ["pods/exec"],
                        "verbs": ["create"],
                        "resourceNames": worker_names,
                    },
                ],
                owner=job.name,
            ),
            RBACObject("RoleBinding", f"{job.name}-launcher", job.namespace,
                       owner=job.name),
        ]
        if needs_partitioner:
            objs += [
                RBACObject("ServiceAccount", f"{job.name}-partitioner",
                           job.namespace, owner=job.name),
                RBACObject(
                    "Role",
                    f"{job.name}-partitioner",
                    job.namespace,
                    rules=[
                        {"resources": ["pods"], "verbs": ["get", "list", "watch"]},
                        {
                            "resources": ["pods/exec"],
                            "verbs": ["create"],
                            "resourceNames": [job.launcher_name()],
                        },
                    ],
                    owner=job.name,
                ),
                RBACObject("RoleBinding", f"{job.name}-partitioner",
                           job.namespace, owner=job.name),
            ]
        for o in objs:
            o.owner_uid = job.uid or None
            if c.get_rbac(o.namespace, o.kind, o.name) is None:
                c.create_rbac(o)

    # -- pods -----------------------------------------------------------
    def _base_labels(self, job: DGLJob, rtype: ReplicaType, name: str):
        return {
            JOB_NAME_LABEL: job.name,
            REPLICA_TYPE_LABEL: rtype.value.lower(),
            REPLICA_NAME_LABEL: name,
        }

    def _ensure_launcher(self, job: DGLJob):
        name = job.launcher_name()
        if self.cluster.get_pod(job.namespace, name) is not None:
            return
        spec_t = job.spec.replica_specs.get(ReplicaType.LAUNCHER)
        template = copy.deepcopy(spec_t.template) if spec_t else {}
        pod_spec = template.get("spec", {})
        init_resources = {  # fixed init-container resources
            "cpu": "100m", "memory": "512Mi", "ephemeral-storage": "5Gi",
        }  # dgljob_controller.go:74-76,1139-1150
        init_containers = [{"name": "kubectl-download",
                            "image": self.kubectl_download_image,
                            "resources": init_resources}]
        if job.spec.partition_mode in (PartitionMode.DGL_API, PartitionMode.PARMETIS):
            # watcher-loop-partitioner also mounts the dataset volume so the
            # partitioner can copy partitions into this still-running init
            # container (dgljob_controller.go:1129-1138)
            init_containers.append({
                "name": "watcher-loop-partitioner",
                "image": self.watcher_loop_image,
                "env": {"WATCHERFILE": "partfile", "WATCHERMODE": "finished"},
                "mounts": ["dataset"],
                "resources": init_resources,
            })
        init_containers.append({
            "name": "watcher-loop-worker",
            "image": self.watcher_loop_image,
            "env": {"WATCHERFILE": "hostfile", "WATCHERMODE": "ready"},
            "resources": init_resources,
        })
        # launcher main container defaults to 1 CPU / 2Gi when unset
        # (dgljob_controller.go:77-78,1229-1240)
        for cont in pod_spec.get("containers", []):
            cont.setdefault("resources",
                            {"limits": {"cpu": "1", "memory": "2Gi"}})
        env = dict(pod_spec.get("env", {}))
        env[ENV_KUBEXEC_PATH] = f"{CONFIG_MOUNT}/kubexec.sh"
        env[ENV_HOSTFILE_PATH] = f"{CONFIG_MOUNT}/hostfile"
        env[ENV_KUBECTL_PATH] = f"{KUBECTL_MOUNT}/kubectl"
        env[ENV_OPERATOR] = "1"
        if job.spec.partition_mode == PartitionMode.SKIP:
            env[ENV_PHASE] = "Launcher_Workload"
        pod = Pod(
            name=name,
            namespace=job.namespace,
            labels=self._base_labels(job, ReplicaType.LAUNCHER, name),
            spec={
                **pod_spec,
                "initContainers": init_containers,
                "env": env,
                "serviceAccount": f"{job.name}-launcher",
                "volumes": ["config", "kube", "dataset"],
            },
            owner=job.name,
            owner_uid=job.uid or None,
        )
        self.cluster.create_pod(pod)

    def _worker_like_pod(self, job: DGLJob, rtype: ReplicaType, name: str,
                         index: int) -> Pod:
        spec_t = job.spec.replica_specs.get(rtype) or job.spec.replica_specs.get(
            ReplicaType.WORKER
        )
        template = copy.deepcopy(spec_t.template) if spec_t else {}
        pod_spec = template.get("spec", {})
        env = dict(pod_spec.get("env", {}))
        if rtype == ReplicaType.PARTITIONER:
            # partitioner reuses the worker template but takes the launcher's
            # command + Partitioner phase env (dgljob_controller.go:1025-1034)
            launcher = job.spec.replica_specs.get(ReplicaType.LAUNCHER)
            if launcher:
                lspec = launcher.template.get("spec", {})
                lcont = (lspec.get("containers") or [{}])[0]
                for k in ("command", "args"):
                    src = lspec.get(k, lcont.get(k))
                    if src is not None:
                        pod_spec[k] = copy.deepcopy(src)
                        if pod_spec.get("containers"):
                            pod_spec["containers"][0][k] = copy.deepcopy(src)
            env[ENV_PHASE] = "Partitioner"
            env[ENV_KUBEXEC_PATH] = f"{CONFIG_MOUNT}/kubexec.sh"
            env[ENV_KUBECTL_PATH] = f"{KUBECTL_MOUNT}/kubectl"
        else:
            if "command" not in pod_spec:
                pod_spec["command"] = ["sleep", "365d"]
        # every worker-like pod carries DGL_OPERATOR_ENV=1
        # (dgljob_controller.go:935-939)
        env[ENV_OPERATOR] = "1"
        is_part = rtype == ReplicaType.PARTITIONER
        spec = {
            **pod_spec,
            "env": env,
            # the partitioner has NO ports (go:1026 clears them); it DOES
            # get the kubectl-download init container + kube volume so its
            # phase-2 `kubectl cp` delivery works (go:1009-1051)
            "ports": ([] if is_part
                      else list(range(DGL_PORT, DGL_PORT + HOST_PORT_NUM))),
            "volumes": (["config", "shm", "kube"] if is_part
                        else ["config", "shm"]),
            # /dev/shm emptyDir sized to half the memory limit
            # (dgljob_controller.go:961-974)
            "shmSizeFraction": 0.5,
        }
        if is_part:
            spec["initContainers"] = [{
                "name": "kubectl-download",
                "image": self.kubectl_download_image,
            }]
            spec["serviceAccount"] = f"{job.name}-partitioner"
        return Pod(
            name=name,
            namespace=job.namespace,
            labels=self._base_labels(job, rtype, name),
            annotations={REPLICA_INDEX_ANNOTATION: str(index)},
            spec=spec,
            owner=job.name,
            owner_uid=job.uid or None,
        )

    def _ensure_partitioner(self, job: DGLJob):
        name = job.partitioner_name()
        if self.cluster.get_pod(job.namespace, name) is None:
            self.cluster.create_pod(
                self._worker_like_pod(job, ReplicaType.PARTITIONER, name, 0)
            )

    def _ensure_workers(self, job: DGLJob):
        for i in range(job.num_workers()):
            name = job.worker_name(i)
            if self.cluster.get_pod(job.namespace, name) is None:
                self.cluster.create_pod(
                    self._worker_like_pod(job, ReplicaType.WORKER, name, i)
                )
            if self.cluster.get_service(job.namespace, name) is None:
                self.cluster.create_service(Service(
                    name=name,
                    namespace=job.namespace,
                    selector={REPLICA_NAME_LABEL: name},
                    cluster_ip=None,  # headless
                    ports=list(range(DGL_PORT, DGL_PORT + HOST_PORT_NUM)),
                    owner=job.name,
                    owner_uid=job.uid or None,
                ))

    # -- status -----------------------------------------------------------
    def _gen_phase(self, job: DGLJob) -> JobPhase:
        """State machine of genJobPhase (dgljob_controller.go:1471-1509),
        with the Skip-mode fix."""
        pods = self._pods(job)
        launcher = pods.get(job.launcher_name())
        partitioner = pods.get(job.partitioner_name())
        workers = [pods.get(job.worker_name(i)) for i in range(job.num_workers())]

        for p in [launcher, partitioner, *workers]:
            if p is not None and p.phase == PodPhase.FAILED:
                return JobPhase.FAILED
        if launcher is not None and launcher.phase == PodPhase.SUCCEEDED:
            return JobPhase.COMPLETED

        skip = job.spec.partition_mode == PartitionMode.SKIP
        if not skip:
            if partitioner is None:
                return JobPhase.PENDING
            if partitioner.is_real_running():
                return JobPhase.PARTITIONING
            part_done = partitioner.phase == PodPhase.SUCCEEDED
        else:
            part_done = True

        all_workers = workers and all(w is not None for w in workers)
        workers_running = all_workers and all(w.is_real_running() for w in workers)
        any_worker_running = any(
            w is not None and w.is_real_running() for w in workers
        )
        if part_done and not any_worker_running and not workers_running:
            if launcher is not None and launcher.is_real_running() and skip:
                return JobPhase.TRAINING
            return JobPhase.PARTITIONED if not skip else JobPhase.STARTING
        if (
            part_done
            and workers_running
            and launcher is not None
            and launcher.is_real_running()
        ):
            return JobPhase.TRAINING
        return JobPhase.STARTING

    def _update_status(self, job: DGLJob):
        pods = self._pods(job)
        phase = self._gen_phase(job)
        job.status.phase = phase
        if phase in (JobPhase.COMPLETED, JobPhase.FAILED):
            if job.status.completion_time is None:
                job.status.completion_time = time.time()

        def stat(names: List[str]) -> ReplicaStatus:
            st = ReplicaStatus()
            ready = 0
            for n in names:
                p = pods.get(n)
                if p is None:
                    continue
                if p.phase in (PodPhase.PENDING, PodPhase.RUNNING):
                    st.active += 1
                elif p.phase == PodPhase.SUCCEEDED:
                    st.succeeded += 1
                elif p.phase == PodPhase.FAILED:
                    st.failed += 1
                if p.is_real_running() or p.phase == PodPhase.SUCCEEDED:
                    ready += 1
            st.ready = f"{ready}/{len(names)}"
            return st

        job.status.replica_statuses = {
            ReplicaType.LAUNCHER: stat([job.launcher_name()]),
            ReplicaType.WORKER: stat(
                [job.worker_name(i) for i in range(job.num_workers())]
            ),
        }
        if job.spec.partition_mode != PartitionMode.SKIP:
            # the reference forgets Partitioner here (controller quirk); fixed
            job.status.replica_statuses[ReplicaType.PARTITIONER] = stat(
                [job.partitioner_name()]
            )
