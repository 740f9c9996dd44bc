"""GPU end-to-end runs of every example workload family (tiny configs):
exercises GCN/edge-softmax/GAT/KGE paths through the real entry points on an
MI355X, beyond the per-kernel numerics tests.

(Named test_z* so pytest collects it AFTER the kernel numerics suite —
the driver runs -x, and a per-kernel failure is the diagnostic one.)"""
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(script, args, timeout=420):
    assert torch.cuda.is_available()
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, script)] + args,
        capture_output=True, text=True, timeout=timeout, cwd=REPO,
    )
    assert r.returncode == 0, f"{script}: {r.stderr[-3000:]}"
    return r.stdout


@pytest.mark.timeout(600)
def test_gcn_node_classification_gpu():
    out = _run("examples/node_classification/train.py",
               ["--epochs", "30", "--feat", "128", "--nodes", "5000",
                "--edges", "40000"])
    assert "acc" in out


@pytest.mark.timeout(600)
def test_gat_link_predict_minibatch_gpu():
    out = _run("examples/link_predict/train.py",
               ["--minibatch", "--epochs", "2", "--nodes", "20000",
                "--edges", "200000", "--feat", "64", "--batch-edges", "512",
                "--steps", "10"])
    assert "AUC" in out


@pytest.mark.timeout(600)
def test_kge_train_eval_gpu():
    out = _run("examples/dgl_ke/train_ke.py",
               ["--model-name", "ComplEx", "--hidden-dim", "400",
                "--gamma", "143.0", "--batch-size", "1024",
                "--neg-sample-size", "256", "--max-step", "60",
                "--log-interval", "30", "--num-entities", "200000",
                "--num-relations", "500", "--num-triples", "1000000",
                "--json", "--eval", "--num-eval", "100"])
    assert "triples/s" in out and "eval:" in out


@pytest.mark.timeout(600)
def test_kge_transe_fused_path_gpu():
    out = _run("examples/dgl_ke/train_ke.py",
               ["--model-name", "TransE_l2", "--hidden-dim", "400",
                "--batch-size", "1024", "--neg-sample-size", "256",
                "--max-step", "40", "--log-interval", "20",
                "--num-entities", "200000", "--num-relations", "500",
                "--num-triples", "1000000", "--json"])
    assert "triples/s" in out
    # capture is best-effort: some driver/runtime stacks refuse stream
    # capture (observed round 1); a clean fallback line is acceptable
    assert ("# capture: enabled" in out or "# capture: disabled" in out), \
        out[:2000]


@pytest.mark.timeout(600)
def test_graph_classification_gpu():
    out = _run("examples/graph_classification/train.py",
               ["--epochs", "3", "--num-graphs", "60"])
    assert "acc" in out
