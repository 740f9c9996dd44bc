#!/usr/bin/env python3
"""Flagship benchmark: distributed GraphSAGE minibatch training on a synthetic
R-MAT graph of ogbn-products shape (BASELINE.json metric: edges/sec, whole
node).

Reference config anchors (SURVEY.md §6 / BASELINE.md): 2-layer GraphSAGE
hidden 16, batch 1000 seeds/rank, fanout [10,25], Adam lr 3e-3, dropout 0.5 —
/root/reference/examples/GraphSAGE_dist/code/train_dist.py defaults; graph =
2,449,029 nodes / 61,859,140 edges / 100 feats / 47 classes (ogbn-products
shape), synthetic R-MAT with random-init weights (no dataset network access).

Run (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

One process per GPU over RCCL; each rank owns one range partition
(weak scaling: per-rank batch fixed at 1000 seeds). Multi-rank runs
replicate the ghost-zone halo once at setup (--no-halo for per-step
alltoallv pulls instead). The timed step includes GPU neighbor sampling,
the feature gather, forward, backward, gradient all-reduce and the Adam
update; 1-GPU runs replay the whole step as a hipGraph when capture
succeeds (--no-capture for eager). Metric value = SUM over ranks of
message-passing edges in the sampled blocks / elapsed (max over ranks).
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist
import torch.nn.functional as F


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--nodes", type=int, default=2_449_029)
    p.add_argument("--edges", type=int, default=61_859_140)
    p.add_argument("--feat", type=int, default=100)
    p.add_argument("--classes", type=int, default=47)
    p.add_argument("--hidden", type=int, default=16)
    p.add_argument("--layers", type=int, default=2)
    p.add_argument("--batch", type=int, default=1000)
    p.add_argument("--fanout", type=str, default="10,25")
    p.add_argument("--lr", type=float, default=3e-3)
    p.add_argument("--dropout", type=float, default=0.5)
    p.add_argument("--partition", type=str, default="range",
                   choices=["range", "ldg"],
                   help="in-bench sharding: contiguous ranges (default) or "
                        "the LDG streaming partitioner (smaller halos at "
                        "ws>1; adds a CPU partitioning pass at setup)")
    p.add_argument("--no-halo", action="store_true",
                   help="disable ghost-zone replication (fall back to "
                        "per-step alltoallv sampling + feature pulls)")
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--profile", type=str, default="",
                   help="write a torch.profiler chrome trace of 3 steps here")
    p.add_argument("--phase-timing", action="store_true",
                   help="print a sample/gather/fwd/bwd/opt breakdown "
                        "(eager mode, CUDA events)")
    p.add_argument("--capture", action="store_true", default=None,
                   help="force hipGraph capture of the whole train step")
    p.add_argument("--no-prefetch", dest="prefetch", action="store_false",
                   default=True,
                   help="disable side-stream sampling prefetch in eager "
                        "mode (prefetch overlaps step k+1's sampling chain "
                        "+ host syncs with step k's fwd/bwd/opt)")
    p.add_argument("--no-capture", dest="capture", action="store_false",
                   help="disable the capture attempt (eager stepping)")
    p.add_argument("--capture-dist", action="store_true",
                   help="EXPERIMENTAL: capture multi-rank steps too (halo "
                        "mode; the graph records the RCCL all-reduce). "
                        "Requires uniform capture success across ranks.")
    return p.parse_args()


def flat_allreduce_grads(model):
    if not dist.is_initialized():
        return
    grads = [p.grad for p in model.parameters() if p.grad is not None]
    flat = torch.cat([g.reshape(-1) for g in grads])
    dist.all_reduce(flat)
    flat /= dist.get_world_size()
    off = 0
    for g in grads:
        n = g.numel()
        g.copy_(flat[off : off + n].view_as(g))
        off += n


def _build_captured_step(args, dg, model, opt, device, fanouts, next_seeds,
                         rank):
    """Capture the whole train step as a hipGraph; returns the replay fn and
    exposes the device-side edge accumulator as an attribute."""
    from dgl_operator_amd.ops.gather_mm import GatherView
    from dgl_operator_amd.ops.sampling import sample_block_capture

    static_seeds = torch.zeros(args.batch, dtype=torch.int64, device=device)
    seed_dev = torch.zeros(1, dtype=torch.int64, device=device)
    edge_accum = torch.zeros(1, dtype=torch.float64, device=device)
    if dg.halo is not None:
        # halo mode: sample over the extended structure (global-id keyed),
        # gather features through the halo map
        feat_t, label_t = dg.halo.feats["feat"], dg.halo.feats["label"]
        row_map, feat_map = dg.halo.row_map, dg.halo.feat_map
        samp_indptr, samp_indices = dg.halo.indptr, dg.halo.indices
    else:
        feat_t, label_t = dg.ndata["feat"], dg.ndata["label"]
        row_map = feat_map = None
        samp_indptr, samp_indices = dg.csc_indptr, dg.csc_indices

    def capture_body():
        cur = static_seeds
        blocks, counters = [], []
        for layer, fanout in enumerate(reversed(fanouts)):
            blk, ctr = sample_block_capture(
                samp_indptr, samp_indices, dg.workspace, cur, fanout,
                7777 + layer, seed_dev,
                rows=row_map[cur] if row_map is not None else None,
            )
            blocks.insert(0, blk)
            counters.insert(0, ctr)
            cur = blk.srcdata_nids
        x = GatherView(feat_t, feat_map[cur] if feat_map is not None else cur)
        y = label_t[feat_map[static_seeds] if feat_map is not None
                    else static_seeds]
        loss = F.cross_entropy(model(blocks, x), y)
        opt.zero_grad(set_to_none=False)
        loss.backward()
        flat_allreduce_grads(model)  # recorded into the graph when ws > 1
        opt.step()
        # valid-edge metric: outer block fully valid; the inner block's
        # valid dst prefix = actual src count of the outer block
        v = torch.full((1,), args.batch, dtype=torch.int64, device=device)
        for blk, ctr in zip(reversed(blocks), reversed(counters)):
            edge_accum.add_(blk.csc_indptr[v[0]].to(torch.float64))
            v = v + ctr
        return loss

    def fill_seeds(step):
        static_seeds.copy_(next_seeds())
        seed_dev.fill_(step + 1)

    # warm up (allocates grads + Adam state) on a side stream, then capture
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for s in range(3):
            fill_seeds(s)
            capture_body()
    torch.cuda.current_stream().wait_stream(side)
    # capture-mode ladder: "global" errors if ANY thread touches the HIP API
    # unsafely during capture — backward() runs on autograd worker threads,
    # which trips hipErrorStreamCaptureUnsupported on some ROCm runtimes.
    # thread_local/relaxed scope the restriction to the capturing thread.
    graph = None
    last_err = None
    for mode in ("global", "thread_local", "relaxed"):
        gr = torch.cuda.CUDAGraph()
        try:
            with torch.cuda.graph(gr, capture_error_mode=mode):
                capture_body()
            graph = gr
            if rank == 0 and mode != "global":
                print(f"# capture: mode={mode}")
            break
        except Exception as e:  # noqa: BLE001
            last_err = e
            torch.cuda.synchronize()
    if graph is None:
        raise last_err
    # sanity: replays must count edges, run backward (nonzero grads), step
    # the optimizer (weights move) and keep everything finite — a captured
    # graph that silently skipped work would otherwise bench an invalid step
    edge_accum.zero_()
    before = [p_.detach().clone() for p_ in model.parameters()]
    for s in range(2):
        fill_seeds(90_000 + s)
        graph.replay()
    torch.cuda.synchronize()
    assert float(edge_accum[0]) > 0, "captured step counted no edges"
    grad_mag = sum(float(p_.grad.abs().sum()) for p_ in model.parameters()
                   if p_.grad is not None)
    assert grad_mag > 0, "captured step produced no gradients"
    moved = any(not torch.equal(b, p_.detach())
                for b, p_ in zip(before, model.parameters()))
    assert moved, "captured step did not update weights"
    for p_ in model.parameters():
        assert bool(torch.isfinite(p_).all()), "non-finite weights"
    if rank == 0:
        print("# capture: enabled (hipGraph whole-step replay)")

    def one_step(step: int) -> int:
        fill_seeds(step)
        graph.replay()
        return 0  # edges tracked on device in edge_accum

    one_step.edge_accum = edge_accum
    return one_step


def main():
    args = parse_args()
    from dgl_operator_amd.distributed import DistGraph, PartitionBook, comm
    from dgl_operator_amd.graph import rmat_graph
    from dgl_operator_amd.models import GraphSAGE

    rank, ws = comm.init_from_env()
    if ws > 1 and args.gpus != ws and rank == 0:
        print(f"# note: --gpus {args.gpus} overridden by WORLD_SIZE={ws}")
    if args.capture_dist:
        # fail fast on unsupported combinations instead of silently
        # benching a different configuration than asked
        if args.no_halo:
            raise SystemExit("--capture-dist requires the halo path "
                             "(remove --no-halo): the captured step records "
                             "only the gradient all-reduce collective")
        if len(args.fanout.split(",")) != 2:
            raise SystemExit("--capture-dist supports the 2-layer flagship "
                             "config only (fanout must have 2 entries)")
    if args.device:
        device = torch.device(args.device)
    elif torch.cuda.is_available():
        device = torch.device(
            f"cuda:{int(os.environ.get('LOCAL_RANK', 0)) % torch.cuda.device_count()}"
        )  # rank % num_gpus, reference train_dist.py:282-285
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    fanouts = [int(x) for x in args.fanout.split(",")]

    # Every rank generates the same graph deterministically, then shards it.
    g = rmat_graph(
        args.nodes, args.edges, num_feats=args.feat, num_classes=args.classes,
        seed=0, device=device,
    )
    n = g.num_nodes
    if args.partition == "ldg" and ws > 1:
        # deterministic on every rank: same graph -> same assignment ->
        # same relabel; shards then follow LDG ownership (smaller halos)
        from dgl_operator_amd.graph.partition import (
            ldg_assignment, relabel_by_assignment,
        )

        t0 = time.time()
        g_cpu = g.to("cpu") if device.type != "cpu" else g
        assign = ldg_assignment(g_cpu, ws)
        g2, bounds, _ = relabel_by_assignment(g_cpu, assign, num_parts=ws)
        g = g2.to(device) if device.type != "cpu" else g2
        if rank == 0:
            s_, d_ = g_cpu.edges()
            cut = float((assign[s_] != assign[d_]).float().mean())
            print(f"# ldg partition: {time.time() - t0:.1f}s, "
                  f"edge-cut {cut:.3f}")
        del g_cpu, g2
    else:
        bounds = [n * p // ws for p in range(ws + 1)]
    book = PartitionBook(bounds, device=device)
    dg = DistGraph.from_full_graph(g, book, rank)
    # free the full graph copies we no longer need (features stay sharded)
    del g
    if ws > 1 and not args.no_halo:
        # one-time ghost-zone replication: per-step sampling + feature pulls
        # become communication-free (only the gradient all-reduce remains)
        dg.build_halo_cache(args.layers, feat_keys=("feat", "label"))

    model = GraphSAGE(args.feat, args.hidden, args.classes,
                      n_layers=args.layers, dropout=args.dropout).to(device)
    # identical init on every rank (same torch seed), so no broadcast needed;
    # make it explicit anyway for robustness
    if ws > 1:
        for p in model.parameters():
            dist.broadcast(p.data, src=0)
    # capture by default on 1-GPU CUDA runs; build failures fall back to
    # eager stepping (a sanity replay checks the captured step first)
    want_capture = (args.capture if args.capture is not None else True)
    # capture's device-side valid-edge accounting is exact for the 2-layer
    # flagship config; deeper samplers fall back to eager stepping
    multi_ok = (ws > 1 and args.capture_dist and not args.no_halo)
    use_capture = (want_capture and (ws == 1 or multi_ok)
                   and device.type == "cuda" and len(fanouts) == 2)
    opt = torch.optim.Adam(model.parameters(), lr=args.lr,
                           capturable=use_capture)

    seed_gen = torch.Generator(device=device)
    seed_gen.manual_seed(12345 + rank)
    lo, hi = dg.lo, dg.hi
    n_owned = hi - lo
    # DataLoader semantics (and no per-step unique/sync): shuffle the owned
    # nodes once per epoch, take contiguous distinct-seed slices
    epoch_perm = torch.randperm(n_owned, generator=seed_gen, device=device) + lo
    cursor = [0]

    def next_seeds():
        if cursor[0] + args.batch > n_owned:
            epoch_perm.copy_(
                torch.randperm(n_owned, generator=seed_gen, device=device) + lo
            )
            cursor[0] = 0
        s = epoch_perm[cursor[0] : cursor[0] + args.batch]
        cursor[0] += args.batch
        return s

    phase_ms = {k: 0.0 for k in ("sample", "gather", "fwd", "bwd", "opt")}
    phase_events = []
    if args.phase_timing and device.type == "cuda":
        phase_events = [torch.cuda.Event(enable_timing=True)
                        for _ in range(6)]

    def one_step(step: int) -> int:
        ev = phase_events
        if ev:
            ev[0].record()
        seeds = next_seeds()
        input_nodes, output_nodes, blocks = dg.sample_blocks(
            seeds, fanouts, seed=step + 1
        )
        if ev:
            ev[1].record()
        x = dg.pull_view("feat", input_nodes)
        y = dg.pull("label", output_nodes)
        if ev:
            ev[2].record()
        logits = model(blocks, x)
        loss = F.cross_entropy(logits, y)
        if ev:
            ev[3].record()
        opt.zero_grad(set_to_none=True)
        loss.backward()
        flat_allreduce_grads(model)
        if ev:
            ev[4].record()
        opt.step()
        if ev:
            ev[5].record()
            torch.cuda.synchronize()
            for i, k in enumerate(phase_ms):
                phase_ms[k] += ev[i].elapsed_time(ev[i + 1])
        return sum(b.num_edges for b in blocks)

    # -- eager-mode sampling prefetch: step k+1's sampling chain (and its
    # per-hop host syncs) runs on a SIDE stream while step k's fwd/bwd/opt
    # execute on the main stream. Comm/compute overlap per the CDNA guide;
    # sampler state is safe because ALL sampling runs on the side stream
    # (serialized there) and training never touches the workspace.
    # ws>1 without halo would issue sampling alltoallv collectives on the
    # side stream concurrently with the main-stream grad all-reduce —
    # cross-stream collective interleaving is a deadlock hazard, so
    # prefetch stays off there; halo-mode sampling is communication-free
    prefetch_ok = (args.prefetch and device.type == "cuda"
                   and not args.phase_timing
                   and (ws == 1 or not args.no_halo))

    def _make_prefetched_step(base_step_edges_fn):
        side = torch.cuda.Stream()
        box = {}

        def record_all(blocks, y, stream):
            for b in blocks:
                for t in (b.csc_indptr, b.csc_indices, b.srcdata_nids):
                    if t is not None:
                        t.record_stream(stream)
            y.record_stream(stream)

        def sample_next(step):
            with torch.cuda.stream(side):
                seeds = next_seeds()
                inp, out_nodes, blocks = dg.sample_blocks(
                    seeds, fanouts, seed=step + 1)
                y = dg.pull("label", out_nodes)
                ev = torch.cuda.Event()
                ev.record(side)
                box[step] = (inp, blocks, y, ev)

        def one_step_prefetched(step: int) -> int:
            if step not in box:
                sample_next(step)
            inp, blocks, y, ev = box.pop(step)
            cur = torch.cuda.current_stream()
            cur.wait_event(ev)
            # tensors were allocated on the side stream; pin their reuse
            # to the consuming stream for the caching allocator
            record_all(blocks, y, cur)
            x = dg.pull_view("feat", inp)
            logits = model(blocks, x)
            loss = F.cross_entropy(logits, y)
            opt.zero_grad(set_to_none=True)
            loss.backward()
            flat_allreduce_grads(model)
            opt.step()
            # overlap: sample step+1 while the enqueued fwd/bwd/opt run
            sample_next(step + 1)
            return sum(b.num_edges for b in blocks)

        return one_step_prefetched

    # -- hipGraph-captured step (1-GPU): sampling, compaction, gather, fwd,
    # bwd and Adam replay as ONE graph launch. Worst-case shapes keep every
    # tensor static; actual sizes live on device
    # (ops.sampling.sample_block_capture). Failures fall back to eager.
    if use_capture:
        try:
            one_step = _build_captured_step(
                args, dg, model, opt, device, fanouts, next_seeds, rank
            )
        except Exception as e:  # noqa: BLE001
            use_capture = False
            msg = str(e).splitlines()[0] if str(e) else type(e).__name__
            print(f"# capture: disabled ({type(e).__name__}: {msg}); "
                  "eager stepping")
            opt = torch.optim.Adam(model.parameters(), lr=args.lr)

    use_prefetch = prefetch_ok and not use_capture
    if use_prefetch:
        base_step = one_step
        one_step = _make_prefetched_step(base_step)
        # prove the prefetched path on this box before relying on it
        try:
            one_step(0)
            torch.cuda.synchronize()
            if rank == 0:
                print("# prefetch: enabled (side-stream sampling overlap)")
        except Exception as e:  # noqa: BLE001
            use_prefetch = False
            one_step = base_step
            msg = str(e).splitlines()[0] if str(e) else type(e).__name__
            print(f"# prefetch: disabled ({type(e).__name__}: {msg})")

    # warmup
    for s in range(1 if use_prefetch else 0, args.warmup):
        one_step(s)

    if args.profile and rank == 0:
        from torch.profiler import ProfilerActivity, profile

        acts = [ProfilerActivity.CPU]
        if device.type == "cuda":
            acts.append(ProfilerActivity.CUDA)
        with profile(activities=acts) as prof:
            for s in range(3):
                one_step(10_000 + s)
        prof.export_chrome_trace(args.profile)
        print(f"# wrote profiler trace to {args.profile}")

    if device.type == "cuda":
        torch.cuda.synchronize()
    if use_capture:
        one_step.edge_accum.zero_()  # count only the timed steps
    comm.barrier()
    t0 = time.perf_counter()
    edges = 0
    for s in range(args.warmup, args.warmup + args.steps):
        edges += one_step(s)
    if device.type == "cuda":
        torch.cuda.synchronize()
    comm.barrier()
    elapsed = time.perf_counter() - t0
    if use_capture:
        edges = float(one_step.edge_accum.cpu()[0])

    t = torch.tensor([elapsed], dtype=torch.float64)
    e = torch.tensor([edges], dtype=torch.float64)
    if ws > 1:
        # max elapsed over ranks; sum of processed edges over ranks
        te = t.to(device) if dist.get_backend() == "nccl" else t
        ee = e.to(device) if dist.get_backend() == "nccl" else e
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        dist.all_reduce(ee, op=dist.ReduceOp.SUM)
        elapsed = float(te.cpu()[0])
        edges = float(ee.cpu()[0])

    if rank == 0:
        if device.type == "cuda":
            print(f"# peak_mem_gb {torch.cuda.max_memory_allocated() / 1e9:.2f}")
        if args.phase_timing and phase_events and not use_capture:
            total_steps = args.warmup + args.steps + (3 if args.profile else 0)
            breakdown = {k: round(v / total_steps, 3)
                         for k, v in phase_ms.items()}
            print(f"# phase_ms_per_step {breakdown}")
        value = edges / elapsed
        print(json.dumps({
            "metric": "edges/sec (whole node) GraphSAGE ogbn-products",
            "value": value,
            "unit": "edges/s",
            "n_gpus": ws,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic R-MAT, ogbn-products shape (2.45M nodes / 61.9M edges / 100 feats / 47 classes), random-init weights",
            "config": {
                "model": f"GraphSAGE L{args.layers} h{args.hidden} fanout[{args.fanout}] dropout{args.dropout}",
                "global_batch": args.batch * ws,
                "seq_len": None,
                "parallelism": (
                    "dp1 (single GPU)" if ws == 1 else
                    f"dp{ws} (graph-partition data parallel, "
                    + ("ghost-zone halo replication)" if not args.no_halo
                       else "alltoallv halo pulls)")
                ),
                "step_mode": ("hipGraph-captured" if use_capture
                              else ("eager+prefetch" if use_prefetch
                                    else "eager")),
                # BASELINE metric's companion number: one epoch = every
                # owned node seeded once (ceil(nodes/ws/batch) steps/rank)
                "epoch_time_s": round(
                    (elapsed / args.steps)
                    * ((args.nodes // ws + args.batch - 1) // args.batch),
                    3),
            },
        }))

    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
