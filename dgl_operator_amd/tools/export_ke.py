"""Merge sharded KGE checkpoints into DGL-KE-style .npy artifacts.

The reference's dglke saves `<dataset>_<model>_entity.npy` /
`..._relation.npy` under --save_path (dglkerun:91-94,113,303). Our trainers
save one shard per rank (`entity_shard{r}.pt`); this tool concatenates them
back into the reference's flat layout so downstream consumers of dglke
checkpoints keep working.

Usage: python -m dgl_operator_amd.tools.export_ke --save-path ckpts \
           --dataset mykg --model-name ComplEx
"""
from __future__ import annotations

import argparse
import glob
import os

import numpy as np
import torch


def merge_shards(save_path: str, prefix: str) -> np.ndarray:
    files = sorted(
        glob.glob(os.path.join(save_path, f"{prefix}_shard*.pt")),
        key=lambda p: int(p.rsplit("shard", 1)[1].split(".")[0]),
    )
    assert files, f"no {prefix}_shard*.pt under {save_path}"
    rows = []
    expect_lo = 0
    for f in files:
        d = torch.load(f, weights_only=True)
        assert d["lo"] == expect_lo, f"non-contiguous shards at {f}"
        expect_lo = d["hi"]
        rows.append(d["emb"].numpy())
    return np.concatenate(rows, axis=0)


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--save-path", required=True)
    p.add_argument("--dataset", default="kg")
    p.add_argument("--model-name", default="ComplEx")
    args = p.parse_args(argv)
    for kind in ("entity", "relation"):
        arr = merge_shards(args.save_path, kind)
        out = os.path.join(
            args.save_path, f"{args.dataset}_{args.model_name}_{kind}.npy"
        )
        np.save(out, arr)
        print(f"wrote {out} shape={arr.shape}")


if __name__ == "__main__":
    main()
