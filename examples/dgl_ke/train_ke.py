#!/usr/bin/env python3
"""Distributed KGE training entry — the dglke_dist_train workload
(reference anchor: ComplEx d=400, gamma=143, lr 0.1, batch 1024, neg 256,
1000 steps, -adv — /root/reference/examples/v1alpha1/DGL-KE.yaml:20-39 +
python/dglrun/exec/dglkerun:284-304). Synthetic KG (no network for FB15k).

Launch single-process or under torch.distributed.run (one rank per GPU);
entity + relation embeddings are sharded across ranks (kvstore)."""
from __future__ import annotations

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.join(_os.path.dirname(_os.path.abspath(__file__)), "..", ".."))



import argparse
import json
import os
import time

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model-name", default="ComplEx")
    p.add_argument("--hidden-dim", type=int, default=400)
    p.add_argument("--gamma", type=float, default=143.0)
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--batch-size", type=int, default=1024)
    p.add_argument("--neg-sample-size", type=int, default=256)
    p.add_argument("--chunk-size", type=int, default=64)
    p.add_argument("--max-step", type=int, default=1000)
    p.add_argument("--log-interval", type=int, default=100)
    p.add_argument("--num-entities", type=int, default=1_000_000)
    p.add_argument("--num-relations", type=int, default=1000)
    p.add_argument("--num-triples", type=int, default=5_000_000)
    p.add_argument("--save-path", default="")
    p.add_argument("--no-save-emb", action="store_true")
    p.add_argument("--json", action="store_true",
                   help="print a bench-style JSON line (triples/s) at the end")
    p.add_argument("--eval", action="store_true",
                   help="report raw MRR/MR/Hits@K on held-out triples")
    p.add_argument("--num-eval", type=int, default=500)
    args = p.parse_args()

    from dgl_operator_amd.distributed import DistKGEModel, KGEdgeSampler, comm

    rank, ws = comm.init_from_env()
    if torch.cuda.is_available():
        device = torch.device(f"cuda:{int(os.environ.get('LOCAL_RANK', 0))}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    gen = torch.Generator(device=device)
    gen.manual_seed(0)
    h = torch.randint(0, args.num_entities, (args.num_triples,),
                      generator=gen, device=device)
    r = torch.randint(0, args.num_relations, (args.num_triples,),
                      generator=gen, device=device)
    t = torch.randint(0, args.num_entities, (args.num_triples,),
                      generator=gen, device=device)

    model = DistKGEModel(
        args.num_entities, args.num_relations, args.hidden_dim,
        score_func=args.model_name, gamma=args.gamma, rank=rank,
        world_size=ws, device=device,
    )
    sampler = KGEdgeSampler(
        (h, r, t), args.num_entities, batch_size=args.batch_size,
        neg_sample_size=args.neg_sample_size, chunk_size=args.chunk_size,
        seed=rank + 1, device=device,
    )
    t0 = time.time()
    for step in range(1, args.max_step + 1):
        hh, rr, tt, negs, neg_head = sampler.next_batch()
        loss = model.train_step(hh, rr, tt, negs, args.chunk_size, args.lr,
                                neg_head=neg_head)
        if step % args.log_interval == 0 and rank == 0:
            elapsed = time.time() - t0
            tps = step * args.batch_size * ws / elapsed
            print(f"step {step} loss {loss:.4f} {tps:,.0f} triples/s",
                  flush=True)
    if args.json and rank == 0:
        elapsed = time.time() - t0
        tps = args.max_step * args.batch_size * ws / elapsed
        print(json.dumps({
            "metric": "triples/sec (whole node) KGE",
            "value": tps,
            "unit": "triples/s",
            "n_gpus": ws,
            "steps": args.max_step,
            "ms_per_step": elapsed / args.max_step * 1000.0,
            "higher_is_better": True,
            "dtype": "fp32",
            "data": "synthetic KG (uniform triples), random-init embeddings",
            "config": {
                "model": f"{args.model_name} d{args.hidden_dim} g{args.gamma}",
                "global_batch": args.batch_size * ws,
                "neg_sample_size": args.neg_sample_size,
                "parallelism": f"sharded-kvstore ep{ws}",
            },
        }), flush=True)
    if args.eval:
        from dgl_operator_amd.distributed.kge import evaluate_kge

        m = evaluate_kge(model, h[: args.num_eval], r[: args.num_eval],
                         t[: args.num_eval])
        if rank == 0:
            print("eval:", {k: round(v, 4) for k, v in m.items()}, flush=True)
    if args.save_path and not args.no_save_emb:
        os.makedirs(args.save_path, exist_ok=True)
        model.entities.save_shard(
            os.path.join(args.save_path, f"entity_shard{rank}.pt"))
        model.relations.save_shard(
            os.path.join(args.save_path, f"relation_shard{rank}.pt"))
        if rank == 0:
            with open(os.path.join(args.save_path, "config.json"), "w") as f:
                json.dump(vars(args), f, indent=2)
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
