"""Graph partitioning + on-disk partition layout.

Replaces the reference's partitioner pod step
(dgl.distributed.partition_graph METIS call at
/root/reference/examples/GraphSAGE_dist/code/load_and_partition_graph.py:124-127)
and the partition JSON consumed by the launcher's dispatch phase
(/root/reference/python/dglrun/tools/dispatch.py:52-91).

No METIS library ships in this image, so two native strategies are provided:
  * ``range``  — contiguous node-id ranges (fast; good when ids are already
                 locality-friendly).
  * ``ldg``    — Linear Deterministic Greedy streaming partitioning
                 (Stanton & Kliot) in C++ via the native extension when built,
                 with a pure-Python fallback for small graphs. Minimizes edge
                 cut like METIS does, so halo feature pulls over xGMI shrink.

After partitioning, nodes are RELABELED so each part owns a contiguous global
id range — the partition book is then just a boundary array (this mirrors
DGL's homogeneous relabeling and makes ownership lookup a bucketize).

On-disk layout (one dir per job, torch.save payloads):
    <out>/<name>.json              partition metadata (num_parts, boundaries,
                                   per-part relative paths)
    <out>/part<i>/graph.pt         local subgraph COO (src, dst in LOCAL ids,
                                   halo srcs in GLOBAL ids) + num local nodes
    <out>/part<i>/node_feat.pt     features/labels/masks of owned nodes
"""
from __future__ import annotations

import json
import os
from dataclasses import dataclass
from typing import Dict, List, Tuple

import torch

from .graph import Graph


@dataclass
class PartitionSpec:
    name: str
    num_parts: int
    num_nodes: int
    num_edges: int
    boundaries: List[int]  # len num_parts+1; part i owns [b[i], b[i+1])
    parts: Dict[str, Dict[str, str]]  # "part-0" -> {"part_graph": ..., "node_feats": ...}

    def to_json(self) -> dict:
        return {
            "graph_name": self.name,
            "num_parts": self.num_parts,
            "num_nodes": self.num_nodes,
            "num_edges": self.num_edges,
            "node_map": self.boundaries,
            **self.parts,
        }

    @staticmethod
    def from_json(d: dict) -> "PartitionSpec":
        parts = {k: v for k, v in d.items() if k.startswith("part-")}
        return PartitionSpec(
            name=d["graph_name"],
            num_parts=d["num_parts"],
            num_nodes=d["num_nodes"],
            num_edges=d["num_edges"],
            boundaries=d["node_map"],
            parts=parts,
        )


def _assign_range(num_nodes: int, num_parts: int) -> torch.Tensor:
    part_size = (num_nodes + num_parts - 1) // num_parts
    return torch.arange(num_nodes, dtype=torch.int64) // part_size


def _assign_ldg_python(g: Graph, num_parts: int) -> torch.Tensor:
    """Pure-Python LDG streaming partitioning (small graphs / fallback)."""
    n = g.num_nodes
    indptr, indices, _ = g.csr()
    indptr_l = indptr.tolist()
    indices_l = indices.tolist()
    # also use in-edges for scoring
    cindptr, cindices, _ = g.csc()
    cindptr_l = cindptr.tolist()
    cindices_l = cindices.tolist()
    cap = (n + num_parts - 1) // num_parts * 1.05 + 1
    assign = [-1] * n
    sizes = [0] * num_parts
    order = torch.randperm(n, generator=torch.Generator().manual_seed(0)).tolist()
    for v in order:
        counts = [0] * num_parts
        for u in indices_l[indptr_l[v] : indptr_l[v + 1]]:
            if assign[u] >= 0:
                counts[assign[u]] += 1
        for u in cindices_l[cindptr_l[v] : cindptr_l[v + 1]]:
            if assign[u] >= 0:
                counts[assign[u]] += 1
        best, best_score = 0, -1.0
        for p in range(num_parts):
            score = counts[p] * (1.0 - sizes[p] / cap)
            if score > best_score:
                best, best_score = p, score
        assign[v] = best
        sizes[best] += 1
    return torch.tensor(assign, dtype=torch.int64)


def _assign_ldg(g: Graph, num_parts: int, balance_train: bool = False,
                balance_edges: bool = False) -> torch.Tensor:
    try:
        from ..ops import backend

        ext = backend.load_extension(required=False)
        if ext is not None and hasattr(ext, "ldg_partition"):
            indptr, indices, _ = g.csr()
            cindptr, cindices, _ = g.csc()
            mask = g.ndata.get("train_mask") if balance_train else None
            return ext.ldg_partition(
                indptr.cpu(), indices.cpu(), cindptr.cpu(), cindices.cpu(),
                num_parts,
                mask.cpu() if mask is not None else None,
                balance_edges,
            )
    except Exception:
        pass
    return _assign_ldg_python(g, num_parts)


def relabel_by_assignment(g: Graph, assign: torch.Tensor,
                          num_parts: "int | None" = None):
    """Relabel nodes so each part owns one contiguous id range. Returns
    (relabeled_graph, boundaries, new_of_old). ndata rows are permuted;
    the same relabel partition_graph applies before writing parts."""
    if num_parts is None:
        num_parts = int(assign.max()) + 1 if assign.numel() else 1
    perm = torch.argsort(assign, stable=True)  # new id -> old id
    new_of_old = torch.empty_like(perm)
    new_of_old[perm] = torch.arange(g.num_nodes, dtype=torch.int64)
    counts = torch.bincount(assign, minlength=num_parts)
    boundaries = torch.zeros(num_parts + 1, dtype=torch.int64)
    boundaries[1:] = torch.cumsum(counts, 0)
    src, dst = g.edges()
    dev = src.device
    new_dev = new_of_old.to(dev)
    g2 = Graph(new_dev[src], new_dev[dst], g.num_nodes)
    perm_dev = perm.to(dev)
    g2.ndata = {k: v[perm_dev] for k, v in g.ndata.items()}
    g2.edata = dict(g.edata)
    return g2, boundaries.tolist(), new_of_old


def ldg_assignment(g: Graph, num_parts: int, balance_train: bool = False,
                   balance_edges: bool = True) -> torch.Tensor:
    """Public LDG node->part assignment (CPU; the bench's --partition ldg
    path and anything else that wants in-memory sharding)."""
    return _assign_ldg(g, num_parts, balance_train, balance_edges)


def partition_graph(
    g: Graph,
    name: str,
    num_parts: int,
    out_path: str,
    algorithm: str = "ldg",
    balance_train: bool = False,
    balance_edges: bool = False,
) -> PartitionSpec:
    """Partition ``g`` and write the on-disk layout. Returns the metadata spec.

    ``g.ndata`` entries are split by ownership and stored per part.
    ``balance_train``/``balance_edges`` mirror the reference's METIS balance
    objectives (load_and_partition_graph.py:124-127) in the LDG scorer.
    """
    if algorithm == "range":
        assign = _assign_range(g.num_nodes, num_parts)
    elif algorithm == "ldg":
        assign = _assign_ldg(g, num_parts, balance_train, balance_edges)
    elif algorithm == "random":
        assign = torch.randint(
            0, num_parts, (g.num_nodes,), generator=torch.Generator().manual_seed(0)
        )
    else:
        raise ValueError(f"unknown partition algorithm {algorithm!r}")

    # Relabel: nodes sorted by (part, old id) -> new contiguous ids per part.
    perm = torch.argsort(assign, stable=True)  # new id -> old id
    new_of_old = torch.empty_like(perm)
    new_of_old[perm] = torch.arange(g.num_nodes, dtype=torch.int64)
    counts = torch.bincount(assign, minlength=num_parts)
    boundaries = torch.zeros(num_parts + 1, dtype=torch.int64)
    boundaries[1:] = torch.cumsum(counts, 0)

    src, dst = g.edges()
    src = new_of_old[src]
    dst = new_of_old[dst]

    os.makedirs(out_path, exist_ok=True)
    parts: Dict[str, Dict[str, str]] = {}
    for p in range(num_parts):
        lo, hi = int(boundaries[p]), int(boundaries[p + 1])
        pdir = os.path.join(out_path, f"part{p}")
        os.makedirs(pdir, exist_ok=True)
        # edges whose dst is owned by part p (in-edges of owned nodes)
        emask = (dst >= lo) & (dst < hi)
        psrc = src[emask]  # may include halo (non-owned) sources, global ids
        pdst = dst[emask]
        torch.save(
            {
                "num_owned": hi - lo,
                "owned_range": (lo, hi),
                "src_global": psrc,
                "dst_global": pdst,
            },
            os.path.join(pdir, "graph.pt"),
        )
        feats = {}
        for k, v in g.ndata.items():
            feats[k] = v[perm[lo:hi]].clone()
        torch.save(feats, os.path.join(pdir, "node_feat.pt"))
        parts[f"part-{p}"] = {
            "part_graph": f"part{p}/graph.pt",
            "node_feats": f"part{p}/node_feat.pt",
        }
        if g.edata:
            # edge features of the part's in-edges, aligned with the
            # (src_global, dst_global) order in graph.pt — the reference
            # dispatches edge_feat.dgl per part the same way
            # (dispatch.py:80-91)
            efeats = {k: v[emask].clone() for k, v in g.edata.items()}
            torch.save(efeats, os.path.join(pdir, "edge_feat.pt"))
            parts[f"part-{p}"]["edge_feats"] = f"part{p}/edge_feat.pt"

    spec = PartitionSpec(
        name=name,
        num_parts=num_parts,
        num_nodes=g.num_nodes,
        num_edges=g.num_edges,
        boundaries=boundaries.tolist(),
        parts=parts,
    )
    with open(os.path.join(out_path, f"{name}.json"), "w") as f:
        json.dump(spec.to_json(), f, indent=2)
    return spec


def load_partition(
    json_path: str, part_id: int
) -> Tuple[dict, Dict[str, torch.Tensor], PartitionSpec]:
    """Load one partition. Returns (graph payload, node feature dict, spec).
    Edge features, when partitioned, ride in the graph payload under
    ``edge_feats`` (rows aligned with src_global/dst_global)."""
    with open(json_path) as f:
        spec = PartitionSpec.from_json(json.load(f))
    base = os.path.dirname(json_path)
    entry = spec.parts[f"part-{part_id}"]
    gpart = torch.load(os.path.join(base, entry["part_graph"]), weights_only=True)
    feats = torch.load(os.path.join(base, entry["node_feats"]), weights_only=True)
    if "edge_feats" in entry:
        gpart["edge_feats"] = torch.load(
            os.path.join(base, entry["edge_feats"]), weights_only=True)
    return gpart, feats, spec
