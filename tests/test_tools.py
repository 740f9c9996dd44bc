"""Workflow-plane tests: hostfile revision, dispatch over the local fabric,
partition save/load roundtrip, dglrun phase driver (Partitioner + Skip)."""
import json
import os
import subprocess
import sys

import torch

from dgl_operator_amd.graph import partition_graph, load_partition, rmat_graph
from dgl_operator_amd.tools.dispatch import dispatch_partitions
from dgl_operator_amd.tools.fabric import LocalFabric
from dgl_operator_amd.tools.hostfile import (
    HostEntry,
    parse_hostfile,
    revise_for_dgl,
    revise_for_dglke,
)

HOSTFILE = """10.244.0.5 30050 job-worker-0 slots=1
10.244.0.6 30050 job-worker-1 slots=1
"""


def test_hostfile_parse_and_revise():
    entries = parse_hostfile(HOSTFILE)
    assert len(entries) == 2
    assert entries[0].ip == "10.244.0.5"
    assert entries[0].slots == 1
    assert revise_for_dgl(entries) == "10.244.0.5 30050\n10.244.0.6 30050\n"
    assert revise_for_dglke(entries, 2) == (
        "10.244.0.5 30050 2\n10.244.0.6 30050 2\n"
    )


def test_partition_roundtrip(tmp_path):
    g = rmat_graph(100, 800, num_feats=4, num_classes=3, seed=1)
    spec = partition_graph(g, "toy", 2, str(tmp_path), algorithm="range")
    assert spec.num_parts == 2
    assert spec.boundaries[0] == 0 and spec.boundaries[-1] == 100
    gpart, feats, spec2 = load_partition(str(tmp_path / "toy.json"), 0)
    assert spec2.num_parts == 2
    lo, hi = gpart["owned_range"]
    assert (gpart["dst_global"] >= lo).all() and (gpart["dst_global"] < hi).all()
    assert feats["feat"].shape == (hi - lo, 4)
    # edge conservation: in-edges of both parts sum to total
    g1, _, _ = load_partition(str(tmp_path / "toy.json"), 1)
    assert gpart["src_global"].numel() + g1["src_global"].numel() == g.num_edges


def test_partition_ldg_balanced(tmp_path):
    g = rmat_graph(300, 3000, seed=2)
    spec = partition_graph(g, "toy", 4, str(tmp_path), algorithm="ldg")
    sizes = [spec.boundaries[i + 1] - spec.boundaries[i] for i in range(4)]
    assert min(sizes) > 0.5 * max(sizes)  # balanced within the LDG slack


def test_dispatch_over_local_fabric(tmp_path):
    g = rmat_graph(60, 400, num_feats=4, seed=3)
    ds = tmp_path / "dataset"
    partition_graph(g, "toy", 2, str(ds), algorithm="range")
    fab = LocalFabric(str(tmp_path / "pods"))
    hosts = parse_hostfile(HOSTFILE)
    ws = "/w"
    dispatch_partitions(str(ds), "toy", hosts, fabric=fab, workspace=ws)
    for i, h in enumerate(hosts):
        pd = fab.pod_dir(h.pod)
        assert os.path.exists(f"{pd}/w/workload/part{i}/graph.pt")
        assert os.path.exists(f"{pd}/w/workload/part{i}/node_feat.pt")
        meta = json.load(open(f"{pd}/w/workload/toy.json"))
        assert meta[f"part-{i}"]["part_graph"] == f"part{i}/graph.pt"


def test_dispatch_slots_gt_one_partitions_per_rank(tmp_path):
    """slotsPerWorker=2: pod i receives the contiguous partition block for
    its ranks (one partition per rank, 2 pods x 2 slots = 4 parts)."""
    g = rmat_graph(80, 600, num_feats=4, seed=4)
    ds = tmp_path / "dataset"
    partition_graph(g, "toy", 4, str(ds), algorithm="range")
    fab = LocalFabric(str(tmp_path / "pods"))
    hosts = parse_hostfile(
        "10.244.0.5 30050 job-worker-0 slots=2\n"
        "10.244.0.6 30050 job-worker-1 slots=2\n"
    )
    dispatch_partitions(str(ds), "toy", hosts, fabric=fab, workspace="/w")
    for pod, parts in (("job-worker-0", (0, 1)), ("job-worker-1", (2, 3))):
        pd = fab.pod_dir(pod)
        for p in parts:
            assert os.path.exists(f"{pd}/w/workload/part{p}/graph.pt"), (pod, p)
        assert os.path.exists(f"{pd}/w/workload/toy.json")


def test_dispatch_rejects_mismatched_counts(tmp_path):
    g = rmat_graph(60, 400, seed=3)
    ds = tmp_path / "dataset"
    partition_graph(g, "toy", 3, str(ds), algorithm="range")
    fab = LocalFabric(str(tmp_path / "pods"))
    hosts = parse_hostfile(HOSTFILE)  # 2 workers vs 3 parts
    try:
        dispatch_partitions(str(ds), "toy", hosts, fabric=fab)
        assert False, "should have raised"
    except AssertionError as e:
        assert "one partition per rank" in str(e)


def test_dist_graph_from_partition(tmp_path):
    g = rmat_graph(80, 600, num_feats=4, seed=5)
    partition_graph(g, "toy", 2, str(tmp_path), algorithm="range")
    from dgl_operator_amd.distributed import DistGraph

    dg = DistGraph.from_partition(str(tmp_path / "toy.json"), 0)
    assert dg.num_owned == 40
    # sampling works locally on owned seeds with global neighbor ids
    seeds = torch.arange(dg.lo, dg.lo + 10)
    nbrs, counts = dg.sample_neighbors_dist(seeds, fanout=3, seed=1)
    assert counts.numel() == 10


def test_dglrun_skip_mode(tmp_path):
    env = dict(os.environ)
    env["DGL_OPERATOR_PHASE_ENV"] = "Launcher_Workload"
    script = tmp_path / "train.py"
    script.write_text("print('train-ran')\n")
    r = subprocess.run(
        [sys.executable, "-m", "dgl_operator_amd.tools.dglrun",
         "--train-entry-point", str(script)],
        env=env, capture_output=True, text=True, cwd="/root/repo",
    )
    assert r.returncode == 0, r.stderr
    assert "train-ran" in r.stdout
    assert "workload (Skip mode)" in r.stdout


def test_partition_balance_objectives(tmp_path):
    """balance_train / balance_edges keep the per-part train-node and edge
    loads balanced (the reference's METIS balance objectives)."""
    g = rmat_graph(400, 6000, seed=7)
    gen = torch.Generator().manual_seed(1)
    g.ndata["train_mask"] = torch.rand(400, generator=gen) < 0.2
    spec = partition_graph(g, "bal", 4, str(tmp_path), algorithm="ldg",
                           balance_train=True, balance_edges=True)
    # per-part train counts within 2x of each other
    tloads, eloads = [], []
    for p in range(4):
        gp, feats, _ = load_partition(str(tmp_path / "bal.json"), p)
        tloads.append(int(feats["train_mask"].sum()))
        eloads.append(int(gp["src_global"].numel()))
    assert min(tloads) > 0
    assert max(tloads) <= 2.5 * max(1, min(tloads)), tloads
    assert max(eloads) <= 2.5 * max(1, min(eloads)), eloads


def test_bench_deterministic_on_cpu(tmp_path):
    """Same seeds -> identical edge totals (samplers are counter-based)."""
    import json
    import subprocess
    import sys

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    vals = []
    for _ in range(2):
        r = subprocess.run(
            [sys.executable, "bench.py", "--steps", "2", "--warmup", "0",
             "--nodes", "5000", "--edges", "30000"],
            capture_output=True, text=True, cwd=REPO, timeout=240,
        )
        assert r.returncode == 0, r.stderr
        d = json.loads(
            [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
        )
        vals.append(d["value"] * d["ms_per_step"])  # edges per step
    assert abs(vals[0] - vals[1]) / vals[0] < 1e-6


def test_kubexec_fabric_command_contract(monkeypatch):
    """KubexecFabric shells through the operator's kubexec.sh / kubectl paths
    (the ConfigMap contract, dgljob_controller.go:874-893)."""
    import subprocess as sp

    from dgl_operator_amd.tools.fabric import KubexecFabric

    calls = []

    class FakeProc:
        def wait(self):
            return 0

    def fake_popen(cmd, shell=None):
        calls.append(cmd)
        return FakeProc()

    monkeypatch.setattr(sp, "Popen", fake_popen)
    monkeypatch.setattr(sp, "call", lambda cmd, shell=None: calls.append(cmd) or 0)
    fab = KubexecFabric(kubexec_path="/etc/dgl/kubexec.sh",
                        kubectl_path="/opt/kube/kubectl")
    fab.exec("job-worker-0", "echo hi", env={"A": "1"})
    fab.copy("/tmp/x", "job-launcher", "/w/x", container="watcher-loop-partitioner")
    assert calls[0].startswith("/etc/dgl/kubexec.sh job-worker-0 A=1 echo hi")
    assert calls[1] == ("/opt/kube/kubectl cp /tmp/x job-launcher:/w/x"
                       " -c watcher-loop-partitioner")


def test_partition_edge_features(tmp_path):
    """g.edata is partitioned alongside the structure (reference ships
    edge_feat.dgl per part, dispatch.py:80-91): per-part rows align with
    the (src_global, dst_global) edge order."""
    from dgl_operator_amd.graph.partition import load_partition

    g = rmat_graph(50, 300, num_feats=4, seed=6)
    g.edata["w"] = torch.arange(g.num_edges, dtype=torch.float32)
    partition_graph(g, "ew", 2, str(tmp_path), algorithm="range")
    total = 0
    src, dst = g.edges()
    for p in range(2):
        gpart, feats, spec = load_partition(str(tmp_path / "ew.json"), p)
        assert "edge_feats" in gpart
        w = gpart["edge_feats"]["w"]
        assert w.numel() == gpart["src_global"].numel()
        total += w.numel()
        # edge weights identify original edges: endpoints must match
        # (range partition => new ids == old ids)
        for k in range(min(5, w.numel())):
            e = int(w[k])
            assert int(src[e]) == int(gpart["src_global"][k])
            assert int(dst[e]) == int(gpart["dst_global"][k])
    assert total == g.num_edges


def test_ldg_recovers_planted_communities():
    """On a graph WITH structure (planted partition, communities interleaved
    mod P so a range split is as bad as random), LDG + refinement must find
    the communities: cut well below random, near the planted optimum."""
    from dgl_operator_amd.ops import backend

    ext = backend.load_extension()
    if ext is None or not hasattr(ext, "ldg_partition"):
        import pytest

        pytest.skip("native extension not built")
    n, P, deg, pout = 60_000, 4, 12, 0.05
    g = torch.Generator().manual_seed(3)
    E = n * deg
    comm_of = torch.arange(n) % P
    src = torch.randint(0, n, (E,), generator=g)
    intra = torch.rand(E, generator=g) > pout
    offs = torch.randint(0, n // P, (E,), generator=g)
    dst = torch.where(intra, comm_of[src] + offs * P,
                      torch.randint(0, n, (E,), generator=g))
    keep = src != dst
    src, dst = src[keep], dst[keep]

    def build_csr(row, col):
        d = torch.bincount(row, minlength=n)
        indptr = torch.zeros(n + 1, dtype=torch.int64)
        torch.cumsum(d, 0, out=indptr[1:])
        return indptr, col[torch.argsort(row)].contiguous()

    ip, ix = build_csr(src, dst)
    cip, cix = build_csr(dst, src)
    a = ext.ldg_partition(ip, ix, cip, cix, P, None, True)
    cut = float((a[src] != a[dst]).float().mean())
    planted = float((comm_of[src] != comm_of[dst]).float().mean())
    # random cut would be ~1 - 1/P = 0.75; planted ~0.09
    assert cut < 0.25, f"cut {cut} (planted {planted})"
    # balance cap held
    sizes = torch.bincount(a, minlength=P)
    assert int(sizes.max()) <= int(n / P * 1.05) + 2


def test_dispatch_cli_reference_contract(tmp_path):
    """dispatch accepts the reference dglrun's invocation spellings
    (--workspace --rel_data_path --rel_workload_path --part_config
    --ip_config, reference dglrun:182-188)."""
    from dgl_operator_amd.tools import dispatch as dispatch_mod

    g = rmat_graph(40, 200, num_feats=2, seed=8)
    ds = tmp_path / "ws" / "dataset"
    partition_graph(g, "ref", 2, str(ds), algorithm="range")
    hostfile = tmp_path / "hostfile"
    hostfile.write_text(HOSTFILE)
    import os as _os
    env_root = str(tmp_path / "pods")
    old = _os.environ.get("DGL_LOCAL_FABRIC_ROOT")
    _os.environ["DGL_LOCAL_FABRIC_ROOT"] = env_root
    try:
        dispatch_mod.main([
            "--workspace", str(tmp_path / "ws"),
            "--rel_data_path", "dataset",
            "--rel_workload_path", "workload",
            "--part_config", str(ds / "ref.json"),
            "--ip_config", str(hostfile),
        ])
    finally:
        if old is None:
            _os.environ.pop("DGL_LOCAL_FABRIC_ROOT", None)
        else:
            _os.environ["DGL_LOCAL_FABRIC_ROOT"] = old
    for i in range(2):
        pod = f"job-worker-{i}"
        assert os.path.exists(
            os.path.join(env_root, pod,
                         str(tmp_path / "ws").lstrip("/"),
                         "workload", f"part{i}", "graph.pt"))


def test_hostfile_cli_reference_contract(tmp_path):
    """revise_hostfile accepts the reference spellings (--workspace
    --ip_config --framework DGLKE --num_servers)."""
    from dgl_operator_amd.tools import hostfile as hf

    src = tmp_path / "hostfile"
    src.write_text(HOSTFILE)
    ws = tmp_path / "ws"
    ws.mkdir()
    hf.main(["--workspace", str(ws), "--ip_config", str(src),
             "--framework", "DGLKE", "--num_servers", "3"])
    out = (ws / "hostfile_revised").read_text()
    assert out == "10.244.0.5 30050 3\n10.244.0.6 30050 3\n"
    hf.main(["--workspace", str(ws), "--ip_config", str(src),
             "--framework", "DGL"])
    assert (ws / "hostfile_revised").read_text() == \
        "10.244.0.5 30050\n10.244.0.6 30050\n"


def test_launch_cli_reference_contract(tmp_path, monkeypatch):
    """launch.py accepts the reference invocation style: positional
    command, --ip_config, --source_file_paths/--target_dir
    (exec/dglrun:202-234, exec/dglkerun:190-233)."""
    from dgl_operator_amd.tools import launch as launch_mod

    hostfile = tmp_path / "hostfile"
    hostfile.write_text(HOSTFILE)
    monkeypatch.setenv("DGL_LOCAL_FABRIC_ROOT", str(tmp_path / "pods"))
    # exec_batch with positional command (touch a file in each pod dir)
    launch_mod.main([
        "--ip_config", str(hostfile), "--cmd_type", "exec_batch",
        "touch proof.txt",
    ])
    for pod in ("job-worker-0", "job-worker-1"):
        assert (tmp_path / "pods" / pod / "proof.txt").exists()
    # copy_batch with --source_file_paths/--target_dir
    src = tmp_path / "payload.txt"
    src.write_text("data")
    launch_mod.main([
        "--ip_config", str(hostfile), "--cmd_type", "copy_batch",
        "--source_file_paths", str(src), "--target_dir", "/w/payload.txt",
    ])
    for pod in ("job-worker-0", "job-worker-1"):
        assert (tmp_path / "pods" / pod / "w" /
                "payload.txt").read_text() == "data"


def test_dglrun_ignore_partition_reuse(tmp_path, monkeypatch):
    """--ignore-partition reuses an existing partition (the reference
    dglkerun PVC-reuse path applied to dglrun): the second partitioner
    run must SKIP phase 1 and still deliver."""
    import subprocess

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    leadfile = tmp_path / "leadfile"
    leadfile.write_text("127.0.0.1 30050 job-launcher slots=1\n")
    env = dict(os.environ)
    env["DGL_LOCAL_FABRIC_ROOT"] = str(tmp_path / "pods")
    env["DGL_OPERATOR_PHASE_ENV"] = "Partitioner"
    env["PYTHONPATH"] = REPO
    argv = [sys.executable, "-m", "dgl_operator_amd.tools.dglrun",
            "--graph-name", "toy", "--workspace", "ws",
            "--leadfile", str(leadfile),
            "--partition-entry-point",
            os.path.join(REPO, "examples/graphsage_dist/load_and_partition_graph.py"),
            "--partition-entry-args",
            "--nodes 100 --edges 500 --feat 4 --classes 2 --algorithm range",
            "--num-partitions", "2", "--ignore-partition"]
    r1 = subprocess.run(argv, capture_output=True, text=True,
                        cwd=str(tmp_path), env=env, timeout=120)
    assert r1.returncode == 0, r1.stderr
    assert "Phase 1/5 partition" in r1.stdout  # first run partitions
    meta = tmp_path / "ws" / "dataset" / "toy.json"
    mtime = meta.stat().st_mtime
    r2 = subprocess.run(argv, capture_output=True, text=True,
                        cwd=str(tmp_path), env=env, timeout=120)
    assert r2.returncode == 0, r2.stderr
    assert "Phase 1/5 skipped (--ignore-partition)" in r2.stdout
    assert meta.stat().st_mtime == mtime  # untouched partition
