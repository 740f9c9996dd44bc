"""CPU emulation of the hipGraph-capture pipeline's padded sampling +
valid-edge accounting (bench._build_captured_step): the device-side formula
    edges = outer_indptr[batch] + inner_indptr[batch + n_new_outer]
must equal the TRUE number of message-passing edges for real seeds, despite
the worst-case padding (garbage rows sample node 0's neighborhood, claims
from garbage rows land beyond the valid prefix only for hop >= 2)."""
import torch

from dgl_operator_amd.graph import rmat_graph
from dgl_operator_amd.ops.sampling import _sample_ref


def emulate_padded_hop(indptr, indices, cur, fanout, claimed, n_valid_seeds):
    """Mirror sample_block_capture's device semantics on CPU.

    cur: padded seed vector (garbage tail = node 0). Returns (counts per
    padded seed, srcdata layout [cur..., claims..., zeros], n_new counting
    ONLY claims made by the first n_valid_seeds rows)."""
    counts = torch.zeros(cur.numel(), dtype=torch.int64)
    draws = []
    for i, v in enumerate(cur.tolist()):
        nb, c = _sample_ref(indptr, indices, torch.tensor([v]), fanout,
                            False, seed=9000 + i)
        counts[i] = c[0]
        draws.append(nb)
    # claim pass in row order (kernel order is atomic/arbitrary; the VALID
    # claim SET for the prefix rows is order-independent)
    new_nodes = []
    n_new_valid = 0
    for i, nb in enumerate(draws):
        for u in nb.tolist():
            if u not in claimed:
                claimed.add(u)
                new_nodes.append(u)
                if i < n_valid_seeds:
                    n_new_valid += 1
    return counts, draws, new_nodes, n_new_valid


def test_capture_edge_formula_matches_truth():
    torch.manual_seed(0)
    g = rmat_graph(500, 6000, seed=3)
    indptr, indices, _ = g.csc()
    batch, f1, f2 = 40, 5, 3  # fanouts [f2, f1] -> outer f1, inner f2
    seeds = torch.randperm(500)[:batch]

    # hop 1 (outer): exactly the real seeds, no padding
    claimed = set(seeds.tolist())
    c1, draws1, new1, n_new_valid1 = emulate_padded_hop(
        indptr, indices, seeds, f1, claimed, batch
    )
    outer_edges_formula = int(c1.sum())  # indptr[batch]
    assert n_new_valid1 == len(new1)  # all outer claims are valid

    # hop 2 (inner): padded seed vector [seeds, claims, zeros...]
    pad_len = batch + batch * f1
    cur2 = torch.zeros(pad_len, dtype=torch.int64)
    cur2[:batch] = seeds
    cur2[batch : batch + len(new1)] = torch.tensor(new1)
    n_valid2 = batch + len(new1)
    claimed2 = set(cur2[:n_valid2].tolist()) | {0}
    c2, draws2, _, _ = emulate_padded_hop(
        indptr, indices, cur2, f2, claimed2, n_valid2
    )
    # the formula counts the inner block's first (batch + n_new_outer) rows
    inner_edges_formula = int(c2[:n_valid2].sum())

    # ground truth: edges whose destination traces back to a real seed
    truth_outer = sum(int(cc) for cc in c1)
    truth_inner = sum(int(c2[i]) for i in range(n_valid2))
    assert outer_edges_formula == truth_outer
    assert inner_edges_formula == truth_inner
    # and the garbage tail contributes nothing to the formula even though it
    # sampled real work (node 0's neighborhood)
    tail_edges = int(c2[n_valid2:].sum())
    deg0 = int(indptr[1] - indptr[0])
    if deg0 > 0:
        assert tail_edges > 0  # the padding DID sample (work happens)
    total_counted = outer_edges_formula + inner_edges_formula
    assert total_counted == truth_outer + truth_inner
    assert total_counted < truth_outer + truth_inner + max(tail_edges, 1)
