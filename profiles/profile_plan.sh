#!/bin/bash
# rocprofv3 capture plan for the GNN hot path — run on an MI355X box:
#   /usr/local/graft/bin/gpurun --timeout 1500 -- 'bash profiles/profile_plan.sh'
# Writes summaries under gpurun_out/prof; copy the kept CSVs into profiles/.
#
# Per the pool rules: PMC counter runs are SEPARATE from trace runs (never
# combine --pmc with -s/-r or hip/hsa/memory-copy trace domains).
set -x
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo

OUT=gpurun_out/prof
mkdir -p "$OUT"

# 0. step-mode A/B: plain eager vs eager+prefetch vs captured
timeout 300 python bench.py --steps 50 --warmup 10 --no-capture --no-prefetch \
    --phase-timing > "$OUT/bench_eager.log" 2>&1
timeout 300 python bench.py --steps 50 --warmup 10 --no-capture \
    > "$OUT/bench_prefetch.log" 2>&1
timeout 300 python bench.py --steps 50 --warmup 10 \
    > "$OUT/bench_capture.log" 2>&1

# 1. kernel-trace + stats over a short bench run (per-kernel wall time)
rocprofv3 --kernel-trace --stats -d "$OUT/bench_trace" -- \
    timeout 300 python bench.py --steps 10 --warmup 3 --no-capture \
    --no-prefetch > "$OUT/bench_trace.log" 2>&1

# 2. PMC counters for the SpMM/sampler kernels (separate run, counters only)
rocprofv3 --pmc SQ_WAVES,SQ_BUSY_CYCLES,TCC_HIT_sum,TCC_MISS_sum \
    -d "$OUT/bench_pmc" -- \
    timeout 300 python bench.py --steps 5 --warmup 2 --no-capture > "$OUT/bench_pmc.log" 2>&1

# 3. kernel-trace of the KGE path
rocprofv3 --kernel-trace --stats -d "$OUT/ke_trace" -- \
    timeout 300 python examples/dgl_ke/train_ke.py --max-step 50 \
    --log-interval 25 --num-entities 1000000 --num-relations 1000 \
    --num-triples 2000000 > "$OUT/ke_trace.log" 2>&1

# 4. per-kernel A/B microbenchmarks (HIP vs eager torch compositions)
timeout 600 python profiles/kernel_bench.py --iters 100 \
    > "$OUT/kernel_bench.log" 2>&1

ls -R "$OUT"
