// GPU neighbor sampler (K7) — replaces the reference's CPU sampler processes
// (launch.py --num-samplers) with an on-device kernel.
//
// One thread per seed. Without replacement we use Floyd's algorithm to draw
// `fanout` DISTINCT edge positions from the seed's in-edge segment (fanout is
// small — the reference uses 10/25 — so the O(k^2) membership scan stays in
// registers/L1). With replacement it is a plain k-draw. RNG is stateless
// splitmix64 keyed on (run seed, seed index, draw), so results are
// reproducible for tests.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace doa {

template <bool REPLACE>
__global__ void sample_kernel(const int64_t* __restrict__ indptr,
                              const int64_t* __restrict__ indices,
                              const int64_t* __restrict__ rows,  // CSC row ids
                              int64_t* __restrict__ out,   // [n, fanout] padded
                              int64_t* __restrict__ counts,  // [n]
                              int64_t n, int fanout, uint64_t rngseed_scalar,
                              const int64_t* __restrict__ rngseed_dev) {
  // hipGraph-replayable RNG: when a device seed buffer is given, mix it in —
  // the host updates the buffer between replays, so a captured graph draws
  // fresh samples every step.
  const uint64_t rngseed =
      rngseed_dev ? (rngseed_scalar ^ (uint64_t)rngseed_dev[0])
                  : rngseed_scalar;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t v = rows[i];
    const int64_t p0 = indptr[v];
    const int64_t deg = indptr[v + 1] - p0;
    int64_t* mine = out + i * fanout;
    if (deg == 0) {
      counts[i] = 0;
      continue;
    }
    if (REPLACE) {
      for (int j = 0; j < fanout; ++j) {
        const uint64_t r = rand_below(rngseed, (uint64_t)i * fanout + j, deg);
        mine[j] = indices[p0 + r];
      }
      counts[i] = fanout;
    } else {
      if (deg <= fanout) {
        for (int64_t j = 0; j < deg; ++j) mine[j] = indices[p0 + j];
        counts[i] = deg;
      } else {
        // Floyd's sampling of `fanout` distinct positions in [0, deg).
        // Selected positions live in the output row itself (global memory,
        // L1-hot, <= fanout entries) — a runtime-indexed local array would
        // be allocated in scratch.
        int cnt = 0;
        for (int64_t j = deg - fanout; j < deg; ++j) {
          const uint64_t t =
              rand_below(rngseed, (uint64_t)i * fanout + (j - (deg - fanout)),
                         (uint64_t)(j + 1));
          bool seen = false;
          for (int k = 0; k < cnt; ++k)
            if (mine[k] == (int64_t)t) { seen = true; break; }
          mine[cnt++] = seen ? j : (int64_t)t;
        }
        for (int k = 0; k < cnt; ++k) mine[k] = indices[p0 + mine[k]];
        counts[i] = cnt;
      }
    }
  }
}

std::tuple<at::Tensor, at::Tensor> sample_neighbors(at::Tensor indptr,
                                                    at::Tensor indices,
                                                    at::Tensor seeds,
                                                    int64_t fanout, bool replace,
                                                    int64_t seed) {
  TORCH_CHECK(seeds.is_cuda(), "sample_neighbors: seeds must be on GPU");
  TORCH_CHECK(fanout >= 1 && fanout <= 256, "fanout must be in [1, 256]");
  const int64_t n = seeds.numel();
  auto padded = at::empty({n, fanout}, seeds.options());
  auto counts = at::empty({n}, seeds.options());
  const int block = 256;
  const int grid = grid_for(n, block);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (replace) {
    hipLaunchKernelGGL((sample_kernel<true>), dim3(grid), dim3(block), 0,
                       stream, indptr.data_ptr<int64_t>(),
                       indices.data_ptr<int64_t>(), seeds.data_ptr<int64_t>(),
                       padded.data_ptr<int64_t>(), counts.data_ptr<int64_t>(),
                       n, (int)fanout, (uint64_t)seed, (const int64_t*)nullptr);
  } else {
    hipLaunchKernelGGL((sample_kernel<false>), dim3(grid), dim3(block), 0,
                       stream, indptr.data_ptr<int64_t>(),
                       indices.data_ptr<int64_t>(), seeds.data_ptr<int64_t>(),
                       padded.data_ptr<int64_t>(), counts.data_ptr<int64_t>(),
                       n, (int)fanout, (uint64_t)seed, (const int64_t*)nullptr);
  }
  DOA_CHECK_HIP(hipGetLastError());
  // pack the padded matrix into a flat neighbor list
  auto mask = at::arange(fanout, seeds.options()).unsqueeze(0) <
              counts.unsqueeze(1);
  auto flat = padded.masked_select(mask);
  return std::make_tuple(flat, counts);
}

}  // namespace doa

namespace doa {

// ---------------------------------------------------------------------------
// Block compaction (the dgl.to_block relabel step, fused):
// given a -1-filled translation table, seeds (assumed unique) and the sampled
// neighbor list, assign block-local ids — seeds take [0, n_seed), unseen
// neighbors claim ids n_seed + atomic counter — translate the neighbor list,
// and hand back srcdata_nids. Ordering of new ids is nondeterministic
// (atomic claim order), which is fine: features are gathered by nid.
// Replaces a sort-based torch.unique + scatter chain (5+ kernels) with 4
// tiny kernels and no sort.
// ---------------------------------------------------------------------------

__global__ void compact_seed_kernel(int64_t* __restrict__ table,
                                    const int64_t* __restrict__ seeds,
                                    int64_t* __restrict__ srcdata,
                                    int64_t n_seed) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_seed;
       i += (int64_t)gridDim.x * blockDim.x) {
    table[seeds[i]] = i;
    srcdata[i] = seeds[i];
  }
}

__global__ void compact_claim_kernel(int64_t* __restrict__ table,
                                     const int64_t* __restrict__ nbrs,
                                     int64_t* __restrict__ srcdata,
                                     unsigned long long* __restrict__ counter,
                                     int64_t n_seed, int64_t n_nbr) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_nbr;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t v = nbrs[i];
    // claim unseen nodes: CAS -1 -> -2, winner assigns the real id
    const long long old = atomicCAS(
        reinterpret_cast<unsigned long long*>(&table[v]),
        (unsigned long long)(-1LL), (unsigned long long)(-2LL));
    if ((long long)old == -1LL) {
      const int64_t id = n_seed + (int64_t)atomicAdd(counter, 1ull);
      srcdata[id] = v;
      atomicExch(reinterpret_cast<unsigned long long*>(&table[v]),
                 (unsigned long long)id);
    }
  }
}

__global__ void compact_translate_kernel(const int64_t* __restrict__ table,
                                         const int64_t* __restrict__ nbrs,
                                         int64_t* __restrict__ out,
                                         int64_t n_nbr) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_nbr;
       i += (int64_t)gridDim.x * blockDim.x) {
    // runs in a separate launch after the claim kernel, so every table
    // entry is a published id by now
    out[i] = table[nbrs[i]];
  }
}

__global__ void compact_reset_kernel(int64_t* __restrict__ table,
                                     const int64_t* __restrict__ srcdata,
                                     int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    table[srcdata[i]] = -1;
  }
}

std::tuple<at::Tensor, at::Tensor> compact_ids(at::Tensor table,
                                               at::Tensor seeds,
                                               at::Tensor neighbors) {
  TORCH_CHECK(table.is_cuda(), "compact_ids: GPU tensors expected");
  const int64_t n_seed = seeds.numel();
  const int64_t n_nbr = neighbors.numel();
  // worst case: every neighbor is new
  auto srcdata = at::empty({n_seed + n_nbr}, seeds.options());
  auto counter = at::zeros({1}, seeds.options());
  auto local = at::empty({n_nbr}, seeds.options());
  const int block = 256;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(compact_seed_kernel, dim3(grid_for(n_seed, block)),
                     dim3(block), 0, stream, table.data_ptr<int64_t>(),
                     seeds.data_ptr<int64_t>(), srcdata.data_ptr<int64_t>(),
                     n_seed);
  if (n_nbr > 0) {
    hipLaunchKernelGGL(compact_claim_kernel, dim3(grid_for(n_nbr, block)),
                       dim3(block), 0, stream, table.data_ptr<int64_t>(),
                       neighbors.data_ptr<int64_t>(),
                       srcdata.data_ptr<int64_t>(),
                       reinterpret_cast<unsigned long long*>(
                           counter.data_ptr<int64_t>()),
                       n_seed, n_nbr);
    hipLaunchKernelGGL(compact_translate_kernel,
                       dim3(grid_for(n_nbr, block)), dim3(block), 0, stream,
                       table.data_ptr<int64_t>(),
                       neighbors.data_ptr<int64_t>(),
                       local.data_ptr<int64_t>(), n_nbr);
  }
  DOA_CHECK_HIP(hipGetLastError());
  const int64_t n_new = counter.item<int64_t>();  // syncs the stream
  srcdata = srcdata.narrow(0, 0, n_seed + n_new);
  hipLaunchKernelGGL(compact_reset_kernel,
                     dim3(grid_for(n_seed + n_new, block)), dim3(block), 0,
                     stream, table.data_ptr<int64_t>(),
                     srcdata.data_ptr<int64_t>(), n_seed + n_new);
  DOA_CHECK_HIP(hipGetLastError());
  return std::make_tuple(srcdata, local);
}

}  // namespace doa

namespace doa {

// ---------------------------------------------------------------------------
// Fused sample+compact block builder — ONE host sync per hop.
//
// The separate sample_neighbors -> masked_select -> compact_ids chain costs
// two device syncs (masked_select's output sizing + the compaction counter
// read) plus ~4 extra kernels per hop; in the minibatch regime the step is
// launch/latency-bound, so this path samples into a padded [n, fanout]
// buffer, compacts/translates in place, and returns device-side totals the
// caller fetches with a single .cpu() copy.
// ---------------------------------------------------------------------------

__global__ void compact_claim_padded_kernel(
    int64_t* __restrict__ table, const int64_t* __restrict__ padded,
    const int64_t* __restrict__ counts, int64_t* __restrict__ srcdata,
    unsigned long long* __restrict__ counter, int64_t n_seed, int64_t n,
    int fanout) {
  const int64_t total = n * fanout;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = i / fanout;
    if ((i % fanout) >= counts[row]) continue;
    const int64_t v = padded[i];
    const long long old = atomicCAS(
        reinterpret_cast<unsigned long long*>(&table[v]),
        (unsigned long long)(-1LL), (unsigned long long)(-2LL));
    if ((long long)old == -1LL) {
      const int64_t id = n_seed + (int64_t)atomicAdd(counter, 1ull);
      srcdata[id] = v;
      atomicExch(reinterpret_cast<unsigned long long*>(&table[v]),
                 (unsigned long long)id);
    }
  }
}

__global__ void translate_padded_kernel(const int64_t* __restrict__ table,
                                        int64_t* __restrict__ padded,
                                        const int64_t* __restrict__ counts,
                                        int64_t n, int fanout) {
  const int64_t total = n * fanout;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = i / fanout;
    if ((i % fanout) >= counts[row]) continue;
    padded[i] = table[padded[i]];
  }
}

__global__ void reset_from_srcdata_kernel(int64_t* __restrict__ table,
                                          const int64_t* __restrict__ srcdata,
                                          const unsigned long long* counter,
                                          int64_t n_seed) {
  const int64_t n = n_seed + (int64_t)*counter;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    table[srcdata[i]] = -1;
  }
}

__global__ void pack_padded_kernel(const int64_t* __restrict__ padded,
                                   const int64_t* __restrict__ counts,
                                   const int64_t* __restrict__ offsets,
                                   int64_t* __restrict__ out, int64_t n,
                                   int fanout) {
  const int64_t total = n * fanout;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = i / fanout;
    const int64_t j = i % fanout;
    if (j < counts[row]) out[offsets[row] + j] = padded[i];
  }
}

// returns (padded_local [n,fanout], counts [n], srcdata [n_seed+n*fanout],
//          totals [1] = n_new)  — all device tensors, no host sync here.
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> sample_block(
    at::Tensor indptr, at::Tensor indices, at::Tensor table, at::Tensor seeds,
    int64_t fanout, bool replace, int64_t seed,
    c10::optional<at::Tensor> seed_dev, c10::optional<at::Tensor> rows_opt) {
  TORCH_CHECK(seeds.is_cuda(), "sample_block: GPU tensors expected");
  // `seeds` are GLOBAL node ids (table/srcdata keys); `rows` are their CSC
  // row indices — identical unless sampling a halo-extended local structure.
  at::Tensor rows = rows_opt.has_value() ? *rows_opt : seeds;
  const int64_t n = seeds.numel();
  auto padded = at::empty({n, fanout}, seeds.options());
  auto counts = at::empty({n}, seeds.options());
  // zero-filled so the unclaimed tail is a VALID node id (0) — capture mode
  // consumes the worst-case buffer without reading the claim counter
  auto srcdata = at::zeros({n + n * fanout}, seeds.options());
  auto counter = at::zeros({1}, seeds.options());
  const int64_t* sdev =
      seed_dev.has_value() ? seed_dev->data_ptr<int64_t>() : nullptr;
  const int block = 256;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  if (replace) {
    hipLaunchKernelGGL((sample_kernel<true>), dim3(grid_for(n, block)),
                       dim3(block), 0, stream, indptr.data_ptr<int64_t>(),
                       indices.data_ptr<int64_t>(), rows.data_ptr<int64_t>(),
                       padded.data_ptr<int64_t>(), counts.data_ptr<int64_t>(),
                       n, (int)fanout, (uint64_t)seed, sdev);
  } else {
    hipLaunchKernelGGL((sample_kernel<false>), dim3(grid_for(n, block)),
                       dim3(block), 0, stream, indptr.data_ptr<int64_t>(),
                       indices.data_ptr<int64_t>(), rows.data_ptr<int64_t>(),
                       padded.data_ptr<int64_t>(), counts.data_ptr<int64_t>(),
                       n, (int)fanout, (uint64_t)seed, sdev);
  }
  hipLaunchKernelGGL(compact_seed_kernel, dim3(grid_for(n, block)),
                     dim3(block), 0, stream, table.data_ptr<int64_t>(),
                     seeds.data_ptr<int64_t>(), srcdata.data_ptr<int64_t>(),
                     n);
  hipLaunchKernelGGL(compact_claim_padded_kernel,
                     dim3(grid_for(n * fanout, block)), dim3(block), 0,
                     stream, table.data_ptr<int64_t>(),
                     padded.data_ptr<int64_t>(), counts.data_ptr<int64_t>(),
                     srcdata.data_ptr<int64_t>(),
                     reinterpret_cast<unsigned long long*>(
                         counter.data_ptr<int64_t>()),
                     n, n, (int)fanout);
  hipLaunchKernelGGL(translate_padded_kernel,
                     dim3(grid_for(n * fanout, block)), dim3(block), 0,
                     stream, table.data_ptr<int64_t>(),
                     padded.data_ptr<int64_t>(), counts.data_ptr<int64_t>(),
                     n, (int)fanout);
  hipLaunchKernelGGL(reset_from_srcdata_kernel,
                     dim3(grid_for(n + n * fanout, block)), dim3(block), 0,
                     stream, table.data_ptr<int64_t>(),
                     srcdata.data_ptr<int64_t>(),
                     reinterpret_cast<const unsigned long long*>(
                         counter.data_ptr<int64_t>()),
                     n);
  DOA_CHECK_HIP(hipGetLastError());
  return std::make_tuple(padded, counts, srcdata, counter);
}

at::Tensor pack_padded(at::Tensor padded, at::Tensor counts,
                       at::Tensor offsets, int64_t total) {
  const int64_t n = counts.numel();
  const int fanout = padded.size(1);
  auto out = at::empty({total}, padded.options());
  const int block = 256;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(pack_padded_kernel,
                     dim3(grid_for(n * fanout, block)), dim3(block), 0,
                     stream, padded.data_ptr<int64_t>(),
                     counts.data_ptr<int64_t>(), offsets.data_ptr<int64_t>(),
                     out.data_ptr<int64_t>(), n, fanout);
  DOA_CHECK_HIP(hipGetLastError());
  return out;
}

}  // namespace doa
