"""Control-plane tests: the phase ladder the reference asserts in its envtest
suite (/root/reference/controllers/dgljob_controller_test.go:131-214) —
which never actually runs there because envtest has no kubelet. Here the
FakeCluster's driveable kubelet makes the ladder testable:
Pending -> Partitioning -> Partitioned -> Training -> Completed.
"""
import time

import pytest

from dgl_operator_amd.operator_plane import (
    CleanPodPolicy,
    DGLJobReconciler,
    FakeCluster,
    JobPhase,
    PartitionMode,
    PodPhase,
    ReplicaType,
    job_from_manifest,
    watcher,
)

GRAPHSAGE_YAML = """
apiVersion: qihoo.net/v1alpha1
kind: DGLJob
metadata:
  name: graphsage-dist
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
          - name: dgl-launcher
            image: dgl-operator-amd:worker
            command: ["dglrun"]
            args: ["--graph-name", "ogbn-products", "--partition-entry-point",
                   "code/load_and_partition_graph.py", "--num-partitions", "2",
                   "--train-entry-point", "code/train_dist.py"]
    Worker:
      replicas: 2
      template:
        spec:
          containers:
          - name: dgl-worker
            image: dgl-operator-amd:worker
"""


def make_job():
    return job_from_manifest(GRAPHSAGE_YAML)


def test_manifest_parsing():
    job = make_job()
    assert job.name == "graphsage-dist"
    assert job.spec.partition_mode == PartitionMode.DGL_API
    assert job.spec.clean_pod_policy == CleanPodPolicy.RUNNING
    assert job.num_workers() == 2
    assert job.spec.replica_specs[ReplicaType.LAUNCHER].template["spec"][
        "containers"
    ][0]["command"] == ["dglrun"]


def test_phase_ladder_full():
    c = FakeCluster()
    r = DGLJobReconciler(c)
    job = make_job()

    # 1. first reconcile: launcher + partitioner created, workers gated
    r.reconcile(job)
    assert c.get_pod("default", "graphsage-dist-launcher") is not None
    assert c.get_pod("default", "graphsage-dist-partitioner") is not None
    assert c.get_pod("default", "graphsage-dist-worker-0") is None
    assert job.status.phase in (JobPhase.PENDING, JobPhase.STARTING)
    assert job.status.start_time is not None

    # 2. partitioner runs -> Partitioning
    c.set_pod_phase("default", "graphsage-dist-partitioner", PodPhase.RUNNING)
    r.reconcile(job)
    assert job.status.phase == JobPhase.PARTITIONING
    assert job.status.replica_statuses[ReplicaType.PARTITIONER].active == 1

    # 3. partitioner succeeds -> Partitioned; workers get created
    c.set_pod_phase("default", "graphsage-dist-partitioner", PodPhase.SUCCEEDED)
    r.reconcile(job)
    assert job.status.phase == JobPhase.PARTITIONED
    assert c.get_pod("default", "graphsage-dist-worker-0") is not None
    assert c.get_pod("default", "graphsage-dist-worker-1") is not None
    # headless service per worker
    svc = c.get_service("default", "graphsage-dist-worker-0")
    assert svc is not None and svc.cluster_ip is None
    assert svc.selector == {"dgl-replica-name": "graphsage-dist-worker-0"}

    # 4. workers + launcher run -> Training; hostfile has worker IPs
    c.run_all_pending()
    r.reconcile(job)
    assert job.status.phase == JobPhase.TRAINING
    ws = job.status.replica_statuses[ReplicaType.WORKER]
    assert ws.active == 2 and ws.ready == "2/2"
    cm = c.get_configmap("default", "graphsage-dist-config")
    lines = cm.data["hostfile"].strip().splitlines()
    assert len(lines) == 2
    assert lines[0].split()[2] == "graphsage-dist-worker-0"
    assert lines[0].split()[1] == "30050"
    assert lines[0].split()[3] == "slots=1"
    assert "kubexec.sh" in cm.data

    # 5. launcher succeeds -> Completed
    c.set_pod_phase("default", "graphsage-dist-launcher", PodPhase.SUCCEEDED)
    r.reconcile(job)
    assert job.status.phase == JobPhase.COMPLETED
    assert job.status.completion_time is not None
    ls = job.status.replica_statuses[ReplicaType.LAUNCHER]
    assert ls.succeeded == 1


def test_failed_pod_fails_job():
    c = FakeCluster()
    r = DGLJobReconciler(c)
    job = make_job()
    r.reconcile(job)
    c.set_pod_phase("default", "graphsage-dist-partitioner", PodPhase.FAILED)
    r.reconcile(job)
    assert job.status.phase == JobPhase.FAILED


def test_clean_pod_policy_running_keeps_finished():
    c = FakeCluster()
    r = DGLJobReconciler(c)
    job = make_job()
    r.reconcile(job)
    c.set_pod_phase("default", "graphsage-dist-partitioner", PodPhase.SUCCEEDED)
    r.reconcile(job)
    c.run_all_pending()
    r.reconcile(job)
    c.set_pod_phase("default", "graphsage-dist-worker-0", PodPhase.SUCCEEDED)
    c.set_pod_phase("default", "graphsage-dist-launcher", PodPhase.SUCCEEDED)
    r.reconcile(job)
    assert job.status.phase == JobPhase.COMPLETED
    # next reconcile applies cleanPodPolicy Running: running pods deleted,
    # finished ones kept (dgljob_controller.go:776-783)
    r.reconcile(job)
    assert c.get_pod("default", "graphsage-dist-worker-0") is not None  # Succeeded
    assert c.get_pod("default", "graphsage-dist-worker-1") is None  # was Running
    assert c.get_pod("default", "graphsage-dist-launcher") is not None


def test_skip_mode_reaches_completed():
    yaml_skip = GRAPHSAGE_YAML.replace("DGL-API", "Skip")
    job = job_from_manifest(yaml_skip)
    c = FakeCluster()
    r = DGLJobReconciler(c)
    r.reconcile(job)
    # no partitioner pod in Skip mode; launcher env has Launcher_Workload
    assert c.get_pod("default", "graphsage-dist-partitioner") is None
    lp = c.get_pod("default", "graphsage-dist-launcher")
    assert lp.spec["env"]["DGL_OPERATOR_PHASE_ENV"] == "Launcher_Workload"
    c.run_all_pending()
    r.reconcile(job)
    assert job.status.phase == JobPhase.TRAINING
    c.set_pod_phase("default", "graphsage-dist-launcher", PodPhase.SUCCEEDED)
    r.reconcile(job)
    assert job.status.phase == JobPhase.COMPLETED


def test_parmetis_mode_runs_partitioner():
    yaml_pm = GRAPHSAGE_YAML.replace("DGL-API", "ParMETIS")
    job = job_from_manifest(yaml_pm)
    c = FakeCluster()
    r = DGLJobReconciler(c)
    r.reconcile(job)
    assert c.get_pod("default", "graphsage-dist-partitioner") is not None


def test_rbac_scoping():
    c = FakeCluster()
    r = DGLJobReconciler(c)
    job = make_job()
    r.reconcile(job)
    role = c.get_rbac("default", "Role", "graphsage-dist-launcher")
    exec_rule = [ru for ru in role.rules if "pods/exec" in ru["resources"]][0]
    assert set(exec_rule["resourceNames"]) == {
        "graphsage-dist-worker-0",
        "graphsage-dist-worker-1",
    }
    prole = c.get_rbac("default", "Role", "graphsage-dist-partitioner")
    exec_rule = [ru for ru in prole.rules if "pods/exec" in ru["resources"]][0]
    assert exec_rule["resourceNames"] == ["graphsage-dist-launcher"]


def test_partitioner_inherits_launcher_command():
    c = FakeCluster()
    r = DGLJobReconciler(c)
    job = make_job()
    r.reconcile(job)
    p = c.get_pod("default", "graphsage-dist-partitioner")
    assert p.spec["command"] == ["dglrun"]
    assert p.spec["env"]["DGL_OPERATOR_PHASE_ENV"] == "Partitioner"
    # worker ports 30050-30069 (HOST_PORT_NUM=20)
    w = None
    c.set_pod_phase("default", "graphsage-dist-partitioner", PodPhase.SUCCEEDED)
    r.reconcile(job)
    w = c.get_pod("default", "graphsage-dist-worker-0")
    assert w.spec["ports"][0] == 30050 and len(w.spec["ports"]) == 20
    assert w.spec["command"] == ["sleep", "365d"]


def test_deletion_cleans_all():
    c = FakeCluster()
    r = DGLJobReconciler(c)
    job = make_job()
    r.reconcile(job)
    c.run_all_pending()
    job.deletion_timestamp = 1.0
    r.reconcile(job)
    assert c.list_pods("default", "graphsage-dist") == []


def test_watcher_ready_and_finished():
    c = FakeCluster()
    r = DGLJobReconciler(c)
    job = make_job()
    r.reconcile(job)
    c.set_pod_phase("default", "graphsage-dist-partitioner", PodPhase.SUCCEEDED)
    r.reconcile(job)
    c.run_all_pending()
    r.reconcile(job)
    cm = c.get_configmap("default", "graphsage-dist-config")
    names = watcher.parse_watchfile(cm.data["hostfile"])
    assert names == ["graphsage-dist-worker-0", "graphsage-dist-worker-1"]
    assert watcher.watch(c, "default", names, "ready", timeout=2)
    # finished-mode times out while workers still run
    assert not watcher.watch(
        c, "default", names, "finished", poll_interval=0.05, timeout=0.2
    )
    c.set_pod_phase("default", "graphsage-dist-worker-0", PodPhase.SUCCEEDED)
    c.set_pod_phase("default", "graphsage-dist-worker-1", PodPhase.SUCCEEDED)
    assert watcher.watch(c, "default", names, "finished", timeout=2)
    # launcher entries are skipped
    assert watcher.parse_watchfile(cm.data["leadfile"]) == []


def test_watch_events_informer_mode():
    from dgl_operator_amd.operator_plane.cluster import Pod
    """Event-driven watch (the reference's informer + workqueue shape,
    watcher-loop/controllers/controller.go:84-152): an already-satisfied
    pod drains in the initial sweep; the rest drain on UPDATE events with
    no polling; timeout works via heartbeats."""
    import threading

    c = FakeCluster()
    for i in range(3):
        c.create_pod(Pod(f"w-{i}", "default", owner="j"))
    c.set_pod_phase("default", "w-0", PodPhase.RUNNING)  # pre-satisfied

    def kubelet():
        time.sleep(0.2)
        c.set_pod_phase("default", "w-1", PodPhase.RUNNING)
        time.sleep(0.1)
        c.set_pod_phase("default", "w-2", PodPhase.RUNNING)

    t = threading.Thread(target=kubelet, daemon=True)
    t.start()
    t0 = time.time()
    assert watcher.watch_events(c, "default", ["w-0", "w-1", "w-2"],
                                "ready", timeout=5)
    assert time.time() - t0 < 3
    t.join()
    # finished-mode timeout (no events arrive)
    assert not watcher.watch_events(c, "default", ["w-1"], "finished",
                                    timeout=0.5)
    # all subscriber queues unregistered
    assert c._watchers == []
    # clusters without a stream fall back to polling
    class NoStream(FakeCluster):
        watch_pods = None
    ns = NoStream()
    ns.create_pod(Pod("x", "default"))
    ns.set_pod_phase("default", "x", PodPhase.RUNNING)
    assert watcher.watch_events(ns, "default", ["x"], "ready", timeout=1)


def test_evicted_launcher_is_retried():
    c = FakeCluster()
    r = DGLJobReconciler(c)
    job = make_job()
    r.reconcile(job)
    c.set_pod_phase("default", "graphsage-dist-launcher", PodPhase.FAILED,
                    reason="Evicted")
    r.reconcile(job)
    # launcher recreated instead of failing the job
    lp = c.get_pod("default", "graphsage-dist-launcher")
    assert lp is not None and lp.phase == PodPhase.PENDING
    assert job.status.phase != JobPhase.FAILED


def test_manager_loop_and_health(tmp_path):
    import urllib.request

    from dgl_operator_amd.operator_plane.manager import Manager

    c = FakeCluster()
    mgr = Manager(cluster=c, reconcile_interval=0.05)
    job = mgr.submit(GRAPHSAGE_YAML)
    port = 18125
    mgr.run(health_port=port, block=False)
    try:
        import time

        time.sleep(0.3)
        assert c.get_pod("default", "graphsage-dist-launcher") is not None
        # drive to completion through the fake kubelet
        c.set_pod_phase("default", "graphsage-dist-partitioner",
                        PodPhase.SUCCEEDED)
        time.sleep(0.2)
        c.run_all_pending()
        time.sleep(0.2)
        c.set_pod_phase("default", "graphsage-dist-launcher",
                        PodPhase.SUCCEEDED)
        time.sleep(0.2)
        assert mgr.get("default", "graphsage-dist").status.phase == JobPhase.COMPLETED
        with urllib.request.urlopen(f"http://127.0.0.1:{port}/healthz") as resp:
            assert resp.status == 200
        with urllib.request.urlopen(f"http://127.0.0.1:{port}/readyz") as resp:
            assert resp.status == 200
        # deletion: pods cleaned, job dropped from the store
        mgr.delete("default", "graphsage-dist")
        time.sleep(0.3)
        assert mgr.get("default", "graphsage-dist") is None
    finally:
        mgr.stop()


def test_shipped_manifests_parse():
    import glob
    import os

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    manifests = glob.glob(os.path.join(repo, "examples/v1alpha1/*.yaml"))
    assert len(manifests) >= 3
    for path in manifests:
        with open(path) as f:
            job = job_from_manifest(f.read())
        assert job.name
        assert job.num_workers() >= 1


def test_manager_survives_bad_job():
    from dgl_operator_amd.operator_plane.manager import Manager

    mgr = Manager()
    good = mgr.submit(GRAPHSAGE_YAML)
    bad = mgr.submit(GRAPHSAGE_YAML.replace("graphsage-dist", "bad-job"))
    # corrupt the bad job so reconcile raises
    bad.spec.replica_specs = None
    mgr.reconcile_once()  # must not raise; good job still progresses
    assert mgr.cluster.get_pod("default", "graphsage-dist-launcher") is not None


def test_kubectl_cluster_and_watcher_cli(tmp_path, monkeypatch):
    """KubectlCluster parses pod JSON from a stubbed kubectl; the watcher-loop
    CLI (the init-container binary equivalent) exits 0 once pods are ready."""
    import json as _json
    import subprocess
    import sys

    fake = tmp_path / "kubectl"
    pod_json = {
        "metadata": {"labels": {"dgl-job-name": "j"}},
        "status": {
            "phase": "Running",
            "podIP": "10.0.0.5",
            "containerStatuses": [{"ready": True}],
        },
    }
    fake.write_text(
        "#!/bin/sh\n"
        f"echo '{_json.dumps(pod_json)}'\n"
    )
    fake.chmod(0o755)
    monkeypatch.setenv("PATH", f"{tmp_path}:{__import__('os').environ['PATH']}")

    from dgl_operator_amd.operator_plane.cluster import KubectlCluster

    c = KubectlCluster()
    p = c.get_pod("default", "j-worker-0")
    assert p.phase.value == "Running" and p.ip == "10.0.0.5"
    assert p.is_real_running()

    watchfile = tmp_path / "hostfile"
    watchfile.write_text("10.0.0.5 30050 j-worker-0 slots=1\n")
    r = subprocess.run(
        [sys.executable, "-m", "dgl_operator_amd.operator_plane.watcher",
         "--watcherfile", str(watchfile), "--mode", "ready",
         "--timeout", "10"],
        capture_output=True, text=True, cwd="/root/repo",
        env={**__import__('os').environ, "PATH": f"{tmp_path}:" + __import__('os').environ["PATH"]},
    )
    assert r.returncode == 0, r.stderr
    assert "waiting for 1 pods" in r.stdout


def test_manifest_roundtrip_serialization():
    import glob
    import os

    import yaml as _yaml

    from dgl_operator_amd.operator_plane import job_to_manifest

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for path in glob.glob(os.path.join(repo, "examples/v1alpha1/*.yaml")):
        with open(path) as f:
            raw = _yaml.safe_load(f)
        job = job_from_manifest(raw)
        out = job_to_manifest(job)
        job2 = job_from_manifest(out)
        assert job2.name == job.name
        assert job2.spec.partition_mode == job.spec.partition_mode
        assert job2.spec.clean_pod_policy == job.spec.clean_pod_policy
        assert job2.spec.slots_per_worker == job.spec.slots_per_worker
        assert set(job2.spec.replica_specs) == set(job.spec.replica_specs)
        for rt, rs in job.spec.replica_specs.items():
            assert job2.spec.replica_specs[rt].replicas == rs.replicas
            assert job2.spec.replica_specs[rt].template == rs.template


def test_reconciler_random_soak():
    """Randomized kubelet-event soak: arbitrary phase flips, deletions and
    evictions interleaved with reconciles must never crash the reconciler,
    always produce well-formed statuses, and terminal cleanup must honor
    cleanPodPolicy=All."""
    import random

    rng = random.Random(7)
    for trial in range(40):
        mode = rng.choice(["DGL-API", "Skip", "ParMETIS"])
        workers = rng.randint(1, 4)
        yaml_doc = GRAPHSAGE_YAML.replace("DGL-API", mode).replace(
            "replicas: 2", f"replicas: {workers}")
        job = job_from_manifest(yaml_doc)
        job.spec.clean_pod_policy = CleanPodPolicy.ALL
        c = FakeCluster()
        r = DGLJobReconciler(c)
        phases = ["Pending", "Running", "Succeeded", "Failed"]
        for step in range(rng.randint(3, 15)):
            r.reconcile(job)
            pods = list(c.pods.values())
            if pods and rng.random() < 0.8:
                p = rng.choice(pods)
                ph = PodPhase(rng.choice(phases))
                reason = "Evicted" if (ph == PodPhase.FAILED
                                       and rng.random() < 0.3) else None
                c.set_pod_phase(p.namespace, p.name, ph, reason=reason)
            if pods and rng.random() < 0.1:
                p = rng.choice(pods)
                c.delete_pod(p.namespace, p.name)
            # status always well-formed
            for st in job.status.replica_statuses.values():
                got, total = st.ready.split("/")
                assert 0 <= int(got) <= int(total)
                assert st.active >= 0 and st.failed >= 0
            assert job.status.phase is None or isinstance(
                job.status.phase, JobPhase)
        # CR deletion: cleanup must remove every pod under policy All
        job.deletion_timestamp = 1.0
        r.reconcile(job)
        assert c.list_pods(job.namespace, job.name) == [], (
            trial, mode, workers)
