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


def _capture_ke_step(model, sampler, args, device):
    """Build two captured KE steps (neg-tail / neg-head) over static id
    buffers; returns a replay(step) callable exposing the static loss."""
    import torch as th

    from dgl_operator_amd.ops import kge_loss

    B, C = args.batch_size, args.batch_size // args.chunk_size
    s_h = th.zeros(B, dtype=th.int64, device=device)
    s_r = th.zeros(B, dtype=th.int64, device=device)
    s_t = th.zeros(B, dtype=th.int64, device=device)
    s_n = th.zeros(C, args.neg_sample_size, dtype=th.int64, device=device)
    losses = {}

    def body(neg_head):
        ent_ids = th.cat([s_h, s_t, s_n.reshape(-1)])
        rows = model.entities.pull(ent_ids).requires_grad_(True)
        h, t = rows[:B], rows[B : 2 * B]
        n = rows[2 * B :]
        r = model.relations.pull(s_r).requires_grad_(True)
        pos = model.score.edge(h, r, t)
        hc = (t if neg_head else h).reshape(C, args.chunk_size, -1)
        rc = r.view(C, args.chunk_size, -1)
        nc = n.reshape(C, args.neg_sample_size, -1)
        neg = model.score.neg(hc, rc, nc, neg_head=neg_head)
        loss = kge_loss(pos, neg, args.adversarial_temperature)
        if args.regularization_coef:
            loss = loss + args.regularization_coef * (
                rows.abs().pow(args.regularization_norm).sum()
                / max(rows.shape[0], 1))
        loss.backward()
        with th.no_grad():
            model.entities.push_grad(ent_ids, rows.grad, args.lr)
            model.relations.push_grad(s_r, r.grad, args.lr)
        losses[neg_head] = losses.get(neg_head, th.zeros((), device=device))
        losses[neg_head].copy_(loss.detach())
        return loss

    def fill():
        hh, rr, tt, negs, neg_head = sampler.next_batch()
        s_h.copy_(hh)
        s_r.copy_(rr)
        s_t.copy_(tt)
        s_n.copy_(negs)
        return neg_head

    graphs = {}
    side = th.cuda.Stream()
    side.wait_stream(th.cuda.current_stream())
    with th.cuda.stream(side):
        for _ in range(4):  # warm both corruption sides
            body(fill())
    th.cuda.current_stream().wait_stream(side)
    # capture-mode ladder (same as bench.py): "global" forbids ANY thread's
    # unsafe HIP calls during capture and backward() runs on autograd
    # worker threads — thread_local/relaxed scope the check to this thread
    for nh in (False, True):
        last = None
        for mode in ("global", "thread_local", "relaxed"):
            g = th.cuda.CUDAGraph()
            try:
                with th.cuda.graph(g, capture_error_mode=mode):
                    body(nh)
                graphs[nh] = g
                break
            except Exception as e:  # noqa: BLE001
                last = e
                th.cuda.synchronize()
        if nh not in graphs:
            raise last
    # sanity replay both sides: losses finite AND the sharded embeddings
    # actually moved (the captured push+Adagrad ran — a graph that skipped
    # the update would bench an invalid step)
    ent_before = model.entities.local.detach().clone()
    for _ in range(2):
        nh = fill()
        graphs[nh].replay()
    th.cuda.synchronize()
    for nh in (False, True):
        assert bool(th.isfinite(losses[nh])), "non-finite captured loss"
    assert not th.equal(ent_before, model.entities.local), \
        "captured KE step did not update the entity shard"

    def replay(step):
        nh = fill()
        graphs[nh].replay()
        return losses[nh]

    return replay


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model-name", "--model_name", default="ComplEx")
    p.add_argument("--hidden-dim", "--hidden_dim", type=int, default=400)
    p.add_argument("--gamma", type=float, default=143.0)
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--batch-size", "--batch_size", type=int, default=1024)
    p.add_argument("--neg-sample-size", "--neg_sample_size", type=int,
                   default=256)
    p.add_argument("--chunk-size", type=int, default=64)
    p.add_argument("--adversarial-temperature", "-adv-temp", type=float,
                   default=1.0, help="self-adversarial sampling temperature "
                                     "(dglke -adv is always on here)")
    p.add_argument("--max-step", "--max_step", type=int, default=1000)
    p.add_argument("--regularization-coef", "--regularization_coef",
                   type=float, default=0.0,
                   help="Lp regularization of batch entity embeddings "
                        "(reference dglkerun passes 1e-9)")
    p.add_argument("--regularization-norm", "--regularization_norm",
                   type=int, default=3)
    p.add_argument("--log-interval", "--log_interval", type=int, default=100)
    p.add_argument("--num-entities", type=int, default=1_000_000)
    p.add_argument("--num-relations", type=int, default=1000)
    p.add_argument("--num-triples", type=int, default=5_000_000)
    p.add_argument("--data-path", default="",
                   help="partitioned dataset root (tools/kg_partition.py "
                        "output); rank r trains part-r's triples and the "
                        "shard boundaries come from the partition json")
    p.add_argument("--dataset-name", default="synthetic",
                   help="dataset dir name under --data-path")
    p.add_argument("--save-path", "--save_path", default="")
    p.add_argument("--no-save-emb", "--no_save_emb", action="store_true")
    p.add_argument("--no-capture", dest="capture", action="store_false",
                   default=True,
                   help="disable hipGraph capture of the train step "
                        "(1-GPU runs capture by default; shapes are static)")
    p.add_argument("--json", action="store_true",
                   help="print a bench-style JSON line (triples/s) at the end")
    p.add_argument("--eval", action="store_true",
                   help="report raw MRR/MR/Hits@K on held-out triples")
    p.add_argument("--num-eval", type=int, default=500)
    p.add_argument("--batch-size-eval", "--batch_size_eval", type=int,
                   default=128, help="eval scoring batch (dglke parity)")
    args = p.parse_args()

    from dgl_operator_amd.distributed import DistKGEModel, KGEdgeSampler, comm

    rank, ws = comm.init_from_env()
    if torch.cuda.is_available():
        device = torch.device(
            f"cuda:{int(os.environ.get('LOCAL_RANK', 0)) % torch.cuda.device_count()}"
        )  # rank % num_gpus, reference train_dist.py:282-285
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    ent_bounds = rel_bounds = None
    if args.data_path:
        # partitioned dataset (dglkerun Phases 1-3 output): rank r owns
        # part-r's triples; entity/relation shard boundaries from the json
        root = os.path.join(args.data_path, args.dataset_name)
        with open(os.path.join(root, f"{args.dataset_name}.json")) as f:
            meta = json.load(f)
        assert meta["num_parts"] == ws, (
            f"partitioned for {meta['num_parts']} ranks but WORLD_SIZE={ws}")
        args.num_entities = meta["num_entities"]
        args.num_relations = meta["num_relations"]
        ent_bounds = meta["entity_boundaries"]
        rel_bounds = meta["relation_boundaries"]
        triples = torch.load(
            os.path.join(root, meta["parts"][str(rank)]["train"]),
            weights_only=True).to(device)
        h, r, t = triples[:, 0], triples[:, 1], triples[:, 2]
        print(f"[train_ke] rank {rank}: {h.numel()} triples, "
              f"entities {ent_bounds[rank]}..{ent_bounds[rank + 1]}",
              flush=True)
    else:
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
        entity_boundaries=ent_bounds, relation_boundaries=rel_bounds,
    )
    sampler = KGEdgeSampler(
        (h, r, t), args.num_entities, batch_size=args.batch_size,
        neg_sample_size=args.neg_sample_size, chunk_size=args.chunk_size,
        seed=rank + 1, device=device,
    )
    # hipGraph capture (1-GPU): the KE step is fully static-shaped — ids go
    # into static buffers, one graph replay per step, loss read only at log
    # points. Head/tail corruption alternates per step, so TWO graphs are
    # captured (one per corruption side) and replayed alternately.
    captured = None
    if args.capture and ws == 1 and device.type == "cuda":
        try:
            captured = _capture_ke_step(model, sampler, args, device)
            print("# capture: enabled (hipGraph KE step, 2 graphs)",
                  flush=True)
        except Exception as e:  # noqa: BLE001
            print(f"# capture: disabled ({type(e).__name__}: {e})",
                  flush=True)

    t0 = time.time()
    for step in range(1, args.max_step + 1):
        if captured is not None:
            loss_t = captured(step)
            loss = None
        else:
            hh, rr, tt, negs, neg_head = sampler.next_batch()
            loss = model.train_step(
                hh, rr, tt, negs, args.chunk_size, args.lr,
                neg_head=neg_head,
                adversarial_temperature=args.adversarial_temperature,
                regularization_coef=args.regularization_coef,
                regularization_norm=args.regularization_norm,
            )
        if step % args.log_interval == 0 and rank == 0:
            if loss is None:
                loss = float(loss_t.detach().cpu())
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

        eh, er, et = h, r, t
        if args.data_path:
            # every rank must evaluate IDENTICAL triples (the eval pulls are
            # collectives): use the global valid split when present,
            # otherwise rank 0's local slice broadcast to everyone
            root = os.path.join(args.data_path, args.dataset_name)
            vpath = os.path.join(root, "valid.pt")
            if os.path.exists(vpath):
                v = torch.load(vpath, weights_only=True).to(device)
                eh, er, et = v[:, 0], v[:, 1], v[:, 2]
            elif ws > 1:
                import torch.distributed as dist

                n = torch.tensor([min(args.num_eval, eh.numel())],
                                 device=device)
                dist.broadcast(n, src=0)
                buf = torch.zeros(3, int(n[0]), dtype=torch.int64,
                                  device=device)
                if rank == 0:
                    buf.copy_(torch.stack([eh[: int(n[0])], er[: int(n[0])],
                                           et[: int(n[0])]]))
                dist.broadcast(buf, src=0)
                eh, er, et = buf[0], buf[1], buf[2]
        m = evaluate_kge(model, eh[: args.num_eval], er[: args.num_eval],
                         et[: args.num_eval],
                         batch_size=args.batch_size_eval)
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
