"""KG triple partitioner — the ``dglke_partition`` equivalent invoked by
dglkerun Phase 1 (reference: /root/reference/python/dglrun/exec/dglkerun:
145-172 runs dglke_partition over a built-in or custom dataset).

Input: a triple file per split (built-in synthetic generator when no files
are given, matching this repo's no-network environment), raw string or
integer ids, column order configurable like DGL-KE's ``--format
[raw_]udd_{hrt}`` permutations.

Output layout under ``<out>/<name>/`` (consumed by
examples/dgl_ke/train_ke.py --data-path and sharded by
distributed.kge.DistKGEModel):

  <name>.json        num_parts, num_entities, num_relations,
                     entity_boundaries, relation_boundaries, file names
  part<k>/train.pt   int64 [n_k, 3] triples whose HEAD entity part k owns
  entity_map.pt      new-of-old entity relabel (original id -> global id)
  relation_map.pt    new-of-old relation relabel
  entities.tsv/relations.tsv   name -> new id (raw string inputs only)

Entities are RELABELED so each part owns one contiguous id range (what the
range PartitionBook/ShardedEmbedding shard by); the assignment is greedy
degree-balanced (heaviest entity to the lightest part — the same
load-balance idea as DGL-KE's SoftRelationPartition applied to entities).
Relations are relabeled with distributed.kge.relation_partition_order.
"""
from __future__ import annotations

import argparse
import json
import os
from typing import Dict, List, Optional, Tuple

import torch


def parse_format(fmt: str) -> Tuple[bool, str]:
    """'raw_udd_hrt' -> (raw_strings=True, 'hrt'); 'udd_htr' -> (False,
    'htr'); plain 'hrt' defaults to int ids."""
    raw = False
    f = (fmt or "hrt").strip()
    if f.startswith("raw_udd_"):
        raw, f = True, f[len("raw_udd_"):]
    elif f.startswith("udd_"):
        f = f[len("udd_"):]
    elif f.startswith("raw_"):
        raw, f = True, f[len("raw_"):]
    assert sorted(f) == ["h", "r", "t"], f"bad triple format {fmt}"
    return raw, f


def read_triples(path: str, order: str, raw: bool,
                 ent_vocab: Optional[Dict[str, int]] = None,
                 rel_vocab: Optional[Dict[str, int]] = None) -> torch.Tensor:
    """Read one whitespace/TSV triple file -> int64 [n, 3] (h, r, t)."""
    hs: List[int] = []
    rs: List[int] = []
    ts: List[int] = []
    hi, ri, ti = order.index("h"), order.index("r"), order.index("t")
    with open(path) as f:
        for line in f:
            parts = line.split()
            if len(parts) < 3:
                continue
            h, r, t = parts[hi], parts[ri], parts[ti]
            if raw:
                hs.append(ent_vocab.setdefault(h, len(ent_vocab)))
                ts.append(ent_vocab.setdefault(t, len(ent_vocab)))
                rs.append(rel_vocab.setdefault(r, len(rel_vocab)))
            else:
                hs.append(int(h))
                rs.append(int(r))
                ts.append(int(t))
    return torch.stack([torch.tensor(hs, dtype=torch.int64),
                        torch.tensor(rs, dtype=torch.int64),
                        torch.tensor(ts, dtype=torch.int64)], dim=1)


def balanced_entity_partition(triples: torch.Tensor, num_entities: int,
                              num_parts: int):
    """Greedy degree-balanced entity->part assignment, then contiguous
    relabel. Returns (new_of_old [num_entities], boundaries [P+1])."""
    deg = torch.zeros(num_entities, dtype=torch.int64)
    deg.index_add_(0, triples[:, 0],
                   torch.ones(triples.shape[0], dtype=torch.int64))
    deg.index_add_(0, triples[:, 2],
                   torch.ones(triples.shape[0], dtype=torch.int64))
    order = torch.argsort(deg, descending=True)
    load = [0] * num_parts
    members: List[List[int]] = [[] for _ in range(num_parts)]
    # heaviest first to the lightest part; zero-degree entities round-robin
    for e in order.tolist():
        p = min(range(num_parts), key=lambda i: (load[i], len(members[i])))
        members[p].append(e)
        load[p] += int(deg[e]) + 1
    new_of_old = torch.empty(num_entities, dtype=torch.int64)
    boundaries = [0]
    nxt = 0
    for p in range(num_parts):
        for e in members[p]:
            new_of_old[e] = nxt
            nxt += 1
        boundaries.append(nxt)
    return new_of_old, boundaries


def partition_kg(
    triples: torch.Tensor,
    num_parts: int,
    out_dir: str,
    name: str = "kg",
    num_entities: Optional[int] = None,
    num_relations: Optional[int] = None,
    ent_names: Optional[List[str]] = None,
    rel_names: Optional[List[str]] = None,
    valid: Optional[torch.Tensor] = None,
    test: Optional[torch.Tensor] = None,
) -> dict:
    from ..distributed.kge import relation_partition_order

    num_entities = num_entities or int(triples[:, [0, 2]].max()) + 1
    num_relations = num_relations or int(triples[:, 1].max()) + 1
    ent_map, ent_bounds = balanced_entity_partition(
        triples, num_entities, num_parts)
    rel_map, rel_bounds = relation_partition_order(
        triples[:, 1], num_relations, num_parts, mode="soft")

    root = os.path.join(out_dir, name)
    os.makedirs(root, exist_ok=True)

    def relabel(t):
        out = t.clone()
        out[:, 0] = ent_map[t[:, 0]]
        out[:, 1] = rel_map[t[:, 1]]
        out[:, 2] = ent_map[t[:, 2]]
        return out

    triples = relabel(triples)
    ent_bounds_t = torch.tensor(ent_bounds)
    # a triple lives with the part that OWNS ITS HEAD entity (the pull of
    # h is then always shard-local; t and negatives go over the wire)
    part_of = torch.bucketize(triples[:, 0].contiguous(), ent_bounds_t[1:-1],
                              right=True)
    meta = {
        "graph_name": name,
        "num_parts": num_parts,
        "num_entities": num_entities,
        "num_relations": num_relations,
        "entity_boundaries": [int(b) for b in ent_bounds],
        "relation_boundaries": [int(b) for b in rel_bounds],
        "parts": {},
    }
    for k in range(num_parts):
        pdir = os.path.join(root, f"part{k}")
        os.makedirs(pdir, exist_ok=True)
        part_triples = triples[part_of == k]
        torch.save(part_triples, os.path.join(pdir, "train.pt"))
        meta["parts"][str(k)] = {
            "train": f"part{k}/train.pt",
            "num_triples": int(part_triples.shape[0]),
        }
    torch.save(ent_map, os.path.join(root, "entity_map.pt"))
    torch.save(rel_map, os.path.join(root, "relation_map.pt"))
    for split, t in (("valid", valid), ("test", test)):
        if t is not None:
            torch.save(relabel(t), os.path.join(root, f"{split}.pt"))
            meta[split] = f"{split}.pt"
    if ent_names is not None:
        with open(os.path.join(root, "entities.tsv"), "w") as f:
            for nm, old in sorted(
                    ((n, i) for n, i in ent_names), key=lambda x: x[1]):
                f.write(f"{nm}\t{int(ent_map[old])}\n")
    if rel_names is not None:
        with open(os.path.join(root, "relations.tsv"), "w") as f:
            for nm, old in sorted(
                    ((n, i) for n, i in rel_names), key=lambda x: x[1]):
                f.write(f"{nm}\t{int(rel_map[old])}\n")
    with open(os.path.join(root, f"{name}.json"), "w") as f:
        json.dump(meta, f, indent=2)
    return meta


def synthetic_triples(num_entities: int, num_relations: int,
                      num_triples: int, seed: int = 0) -> torch.Tensor:
    g = torch.Generator().manual_seed(seed)
    return torch.stack([
        torch.randint(0, num_entities, (num_triples,), generator=g),
        torch.randint(0, num_relations, (num_triples,), generator=g),
        torch.randint(0, num_entities, (num_triples,), generator=g),
    ], dim=1)


def main(argv=None):
    p = argparse.ArgumentParser(prog="kg_partition")
    p.add_argument("--dataset", default="synthetic",
                   help="dataset name (synthetic = generate in-process; "
                        "anything else requires --data-files)")
    p.add_argument("-k", "--num-partitions", type=int, required=True)
    p.add_argument("--data-path", "--data_path", default="dataset",
                   help="output root (dglke_partition --data_path parity)")
    p.add_argument("--format", default="hrt",
                   help="column order, optionally raw_udd_/udd_ prefixed "
                        "(DGL-KE --format parity)")
    p.add_argument("--data-files", "--data_files", nargs="*", default=[],
                   help="train [valid [test]] triple files")
    p.add_argument("--num-entities", type=int, default=10_000)
    p.add_argument("--num-relations", type=int, default=100)
    p.add_argument("--num-triples", type=int, default=100_000)
    p.add_argument("--seed", type=int, default=0)
    args = p.parse_args(argv)

    raw, order = parse_format(args.format)
    ent_names = rel_names = None
    valid = test = None
    if args.data_files:
        ent_vocab: Dict[str, int] = {}
        rel_vocab: Dict[str, int] = {}
        train = read_triples(args.data_files[0], order, raw, ent_vocab,
                             rel_vocab)
        if len(args.data_files) > 1:
            valid = read_triples(args.data_files[1], order, raw, ent_vocab,
                                 rel_vocab)
        if len(args.data_files) > 2:
            test = read_triples(args.data_files[2], order, raw, ent_vocab,
                                rel_vocab)
        ne = len(ent_vocab) if raw else None
        nr = len(rel_vocab) if raw else None
        if raw:
            ent_names = list(ent_vocab.items())
            rel_names = list(rel_vocab.items())
    else:
        assert args.dataset == "synthetic", (
            f"dataset {args.dataset!r} needs --data-files (no network for "
            "downloads in this environment)")
        train = synthetic_triples(args.num_entities, args.num_relations,
                                  args.num_triples, args.seed)
        ne, nr = args.num_entities, args.num_relations
    meta = partition_kg(train, args.num_partitions, args.data_path,
                        name=args.dataset, num_entities=ne,
                        num_relations=nr, ent_names=ent_names,
                        rel_names=rel_names, valid=valid, test=test)
    sizes = [meta["parts"][str(k)]["num_triples"]
             for k in range(args.num_partitions)]
    print(f"[kg_partition] {meta['num_entities']} entities / "
          f"{meta['num_relations']} relations / {int(train.shape[0])} "
          f"triples -> {args.num_partitions} parts {sizes}")


if __name__ == "__main__":
    main()
