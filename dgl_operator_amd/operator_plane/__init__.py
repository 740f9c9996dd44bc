from .api import (
    DGLJob,
    DGLJobSpec,
    DGLJobStatus,
    JobPhase,
    PartitionMode,
    CleanPodPolicy,
    PodPhase,
    ReplicaType,
    ReplicaSpec,
    job_from_manifest,
    job_to_manifest,
)
from .cluster import Cluster, FakeCluster, Pod, ConfigMap, Service
from .reconciler import DGLJobReconciler
from . import watcher

__all__ = [
    "DGLJob", "DGLJobSpec", "DGLJobStatus", "JobPhase", "PartitionMode",
    "CleanPodPolicy", "PodPhase", "ReplicaType", "ReplicaSpec",
    "job_from_manifest", "job_to_manifest",
    "Cluster", "FakeCluster", "Pod", "ConfigMap", "Service",
    "DGLJobReconciler", "watcher",
]
