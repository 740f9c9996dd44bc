"""Full-system integration: DGLJob manifest -> reconciler -> local pod
runtime -> dglrun phases -> 2-node torchrun GraphSAGE training -> Completed.

This is the vertical slice the reference can only exercise on a live
cluster; the LocalPodRuntime plays kubelet."""
import os
import socket

import pytest

from dgl_operator_amd.operator_plane import JobPhase
from dgl_operator_amd.operator_plane.local_runtime import run_job

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


MANIFEST = """
apiVersion: qihoo.net/v1alpha1
kind: DGLJob
metadata:
  name: e2e-sage
  namespace: default
spec:
  partitionMode: DGL-API
  cleanPodPolicy: Running
  slotsPerWorker: 1
  dglReplicaSpecs:
    Launcher:
      replicas: 1
      template:
        spec:
          containers:
          - name: launcher
            image: local
            command: ["python", "-m", "dgl_operator_amd.tools.dglrun"]
            args:
            - --graph-name=toy
            - --partition-entry-point={repo}/examples/graphsage_dist/load_and_partition_graph.py
            - --partition-entry-args=--nodes 300 --edges 2500 --feat 8 --classes 3 --algorithm range
            - --num-partitions=2
            - --train-entry-point={repo}/examples/graphsage_dist/train_dist.py
            - --train-entry-args=--num-epochs 1 --batch-size 32 --fan-out 3,3 --log-every 100
            - --workspace=workspace
            - --hostfile=etc_dgl/hostfile
            - --leadfile=etc_dgl/leadfile
            - --master-port={port}
    Worker:
      replicas: 2
      template:
        spec:
          containers:
          - name: worker
            image: local
"""


@pytest.mark.timeout(800)
def test_dgljob_end_to_end_local_runtime(tmp_path):
    job = None
    for attempt in range(2):  # retry absorbs master-port races
        root = tmp_path / f"try{attempt}"
        manifest = MANIFEST.format(repo=REPO, port=_free_port())
        job = run_job(
            manifest, str(root), timeout=360,
            extra_env={"PYTHONPATH": REPO},
        )
        if job.status.phase == JobPhase.COMPLETED:
            tmp_path = root
            break
    assert job.status.phase == JobPhase.COMPLETED, job.status
    # partitions were dispatched into both worker pod dirs and training ran
    for i in range(2):
        wd = tmp_path / f"e2e-sage-worker-{i}" / "workspace"
        assert (wd / "workload" / f"part{i}" / "graph.pt").exists()
        assert (wd / "hostfile_revised").exists()


KE_MANIFEST = """
apiVersion: qihoo.net/v1alpha1
kind: DGLJob
metadata:
  name: e2e-ke
  namespace: default
spec:
  partitionMode: Skip
  cleanPodPolicy: Running
  dglReplicaSpecs:
    Launcher:
      replicas: 1
      template:
        spec:
          containers:
          - name: launcher
            image: local
            command: ["python", "{repo}/examples/dgl_ke/train_ke.py"]
            args:
            - --model-name=ComplEx
            - --hidden-dim=16
            - --batch-size=64
            - --neg-sample-size=8
            - --max-step=20
            - --log-interval=10
            - --num-entities=3000
            - --num-relations=10
            - --num-triples=10000
            - --save-path=ckpts
    Worker:
      replicas: 1
      template:
        spec:
          containers:
          - name: worker
            image: local
"""


@pytest.mark.timeout(300)
def test_dgljob_ke_skip_mode_local_runtime(tmp_path):
    """Skip-mode DGLJob: launcher runs the KE workload directly (the
    reference's Launcher_Workload path) and checkpoints shards."""
    manifest = KE_MANIFEST.format(repo=REPO)
    job = run_job(manifest, str(tmp_path), timeout=240,
                  extra_env={"PYTHONPATH": REPO})
    assert job.status.phase == JobPhase.COMPLETED, job.status
    ck = tmp_path / "e2e-ke-launcher" / "ckpts"
    assert (ck / "entity_shard0.pt").exists()
    assert (ck / "relation_shard0.pt").exists()
