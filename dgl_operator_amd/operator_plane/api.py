"""DGLJob API types — group qihoo.net/v1alpha1, faithful to the reference CRD
(/root/reference/api/v1alpha1/dgljob_types.go) so examples/v1alpha1/*.yaml
manifests apply unchanged. Implemented in Python (this image has no Go
toolchain); semantics mirror the Go types field for field.
"""
from __future__ import annotations

import copy
from dataclasses import dataclass, field
from enum import Enum
from typing import Any, Dict, Optional

import yaml

GROUP = "qihoo.net"
VERSION = "v1alpha1"
KIND = "DGLJob"

# constants from dgljob_types.go:27-35
DGL_PORT = 30050
HOST_PORT_NUM = 20  # ports DGL_PORT..DGL_PORT+19 (dgljob_controller.go:80)
REPLICA_TYPE_LABEL = "dgl-replica-type"
REPLICA_NAME_LABEL = "dgl-replica-name"
REPLICA_INDEX_ANNOTATION = "dgl-replica"
JOB_NAME_LABEL = "dgl-job-name"

# env contract (dgljob_controller.go:58-63)
ENV_PHASE = "DGL_OPERATOR_PHASE_ENV"
ENV_KUBEXEC_PATH = "DGL_OPERATOR_KUBEXEC_PATH"
ENV_KUBECTL_PATH = "DGL_OPERATOR_KUBECTL_PATH"
ENV_HOSTFILE_PATH = "DGL_OPERATOR_HOSTFILE_PATH"
ENV_PARTFILE_PATH = "DGL_OPERATOR_PARTFILE_PATH"
ENV_OPERATOR = "DGL_OPERATOR_ENV"
CONFIG_MOUNT = "/etc/dgl"
KUBECTL_MOUNT = "/opt/kube"


class PartitionMode(str, Enum):
    DGL_API = "DGL-API"
    PARMETIS = "ParMETIS"
    SKIP = "Skip"


class CleanPodPolicy(str, Enum):
    ALL = "All"
    RUNNING = "Running"
    NONE = "None"


class ReplicaType(str, Enum):
    LAUNCHER = "Launcher"
    WORKER = "Worker"
    PARTITIONER = "Partitioner"


class JobPhase(str, Enum):
    PENDING = "Pending"
    STARTING = "Starting"
    PARTITIONING = "Partitioning"
    PARTITIONED = "Partitioned"
    TRAINING = "Training"
    COMPLETED = "Completed"
    FAILED = "Failed"
    # declared by the reference but never produced there
    # (dgljob_types.go:48-49); kept for API compatibility
    SUCCEED = "Succeed"
    EVICTED = "Evicted"


class PodPhase(str, Enum):
    PENDING = "Pending"
    RUNNING = "Running"
    SUCCEEDED = "Succeeded"
    FAILED = "Failed"
    UNKNOWN = "Unknown"


@dataclass
class ReplicaSpec:
    replicas: int = 1
    template: Dict[str, Any] = field(default_factory=dict)  # PodTemplateSpec


@dataclass
class ReplicaStatus:
    active: int = 0
    succeeded: int = 0
    failed: int = 0
    ready: str = "0/0"


@dataclass
class DGLJobSpec:
    partition_mode: PartitionMode = PartitionMode.DGL_API
    clean_pod_policy: CleanPodPolicy = CleanPodPolicy.RUNNING
    slots_per_worker: int = 1
    replica_specs: Dict[ReplicaType, ReplicaSpec] = field(default_factory=dict)


@dataclass
class DGLJobStatus:
    phase: Optional[JobPhase] = None
    replica_statuses: Dict[ReplicaType, ReplicaStatus] = field(default_factory=dict)
    start_time: Optional[float] = None
    completion_time: Optional[float] = None


@dataclass
class DGLJob:
    name: str
    namespace: str = "default"
    spec: DGLJobSpec = field(default_factory=DGLJobSpec)
    status: DGLJobStatus = field(default_factory=DGLJobStatus)
    deletion_timestamp: Optional[float] = None
    uid: str = ""

    # -- helpers -----------------------------------------------------------
    def launcher_name(self) -> str:
        return f"{self.name}-launcher"

    def worker_name(self, i: int) -> str:
        return f"{self.name}-worker-{i}"

    def partitioner_name(self) -> str:
        return f"{self.name}-partitioner"

    def num_workers(self) -> int:
        spec = self.spec.replica_specs.get(ReplicaType.WORKER)
        return spec.replicas if spec else 0


def job_from_manifest(manifest: Dict[str, Any] | str) -> DGLJob:
    """Parse a DGLJob from a dict or YAML string compatible with the
    reference's examples/v1alpha1/*.yaml."""
    if isinstance(manifest, str):
        manifest = yaml.safe_load(manifest)
    assert manifest.get("kind") == KIND, f"not a DGLJob: {manifest.get('kind')}"
    api = manifest.get("apiVersion", "")
    assert api == f"{GROUP}/{VERSION}", f"unsupported apiVersion {api}"
    meta = manifest.get("metadata", {})
    spec_d = manifest.get("spec", {})
    replica_specs: Dict[ReplicaType, ReplicaSpec] = {}
    for key, val in (spec_d.get("dglReplicaSpecs") or {}).items():
        rt = ReplicaType(key)
        replica_specs[rt] = ReplicaSpec(
            replicas=int(val.get("replicas", 1)),
            template=copy.deepcopy(val.get("template", {})),
        )
    spec = DGLJobSpec(
        partition_mode=PartitionMode(spec_d.get("partitionMode", "DGL-API")),
        clean_pod_policy=CleanPodPolicy(spec_d.get("cleanPodPolicy", "Running")),
        slots_per_worker=int(spec_d.get("slotsPerWorker", 1)),
        replica_specs=replica_specs,
    )
    job = DGLJob(
        name=meta.get("name", "dgljob"),
        namespace=meta.get("namespace", "default"),
        spec=spec,
        uid=meta.get("uid", ""),
    )
    if meta.get("deletionTimestamp"):
        import time as _time

        job.deletion_timestamp = _time.time()
    return job


def _rfc3339(ts: Optional[float]) -> Optional[str]:
    if ts is None:
        return None
    import datetime

    return datetime.datetime.fromtimestamp(
        ts, tz=datetime.timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")


def status_to_manifest(job: DGLJob) -> Dict[str, Any]:
    """job.status -> the CR .status shape (dgljob_types.go:94-108)."""
    st: Dict[str, Any] = {}
    if job.status.phase is not None:
        st["phase"] = job.status.phase.value
    if job.status.start_time is not None:
        st["startTime"] = _rfc3339(job.status.start_time)
    if job.status.completion_time is not None:
        st["completionTime"] = _rfc3339(job.status.completion_time)
    if job.status.replica_statuses:
        st["replicaStatuses"] = {
            rt.value: {
                "active": rs.active,
                "succeeded": rs.succeeded,
                "failed": rs.failed,
                "ready": rs.ready,
            }
            for rt, rs in job.status.replica_statuses.items()
        }
    return st


def job_to_manifest(job: DGLJob) -> Dict[str, Any]:
    return {
        "apiVersion": f"{GROUP}/{VERSION}",
        "kind": KIND,
        "metadata": {"name": job.name, "namespace": job.namespace},
        "spec": {
            "partitionMode": job.spec.partition_mode.value,
            "cleanPodPolicy": job.spec.clean_pod_policy.value,
            "slotsPerWorker": job.spec.slots_per_worker,
            "dglReplicaSpecs": {
                rt.value: {"replicas": rs.replicas, "template": rs.template}
                for rt, rs in job.spec.replica_specs.items()
            },
        },
    }
