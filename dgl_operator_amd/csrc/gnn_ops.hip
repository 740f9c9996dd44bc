// g-SpMM / g-SDDMM / edge-softmax / segment-reduce HIP kernels for gfx950.
//
// These are the message-passing hot ops (K1-K5 in SURVEY.md §2.4) the
// reference delegates to DGL's CUDA build; here they are written natively for
// CDNA4: 64-lane wavefronts, float4 (16 B/lane) vectorized feature rows,
// thread-per-(row, feature-chunk) mapping so a gather of one neighbor row is
// fully coalesced across the consecutive threads that own the row.
//
// All kernels are memory-bound (HBM3E ~8 TB/s is the roofline); design goal
// is full-line utilization on the feature gathers, not MFMA.

#include <torch/extension.h>
#include <ATen/Parallel.h>
#include <limits>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace doa {

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// ---------------------------------------------------------------------------
// SpMM: out[v, :] = reduce_{p in [indptr[v], indptr[v+1])} w[p] * feat[indices[p], :]
//   weight: W_NONE (copy_u) | W_SCALAR (w[p]) | W_HEAD (w[p*H + h])
// ---------------------------------------------------------------------------
enum WeightMode { W_NONE = 0, W_SCALAR = 1, W_HEAD = 2 };

template <typename scalar_t, int VEC, WeightMode WM>
__global__ void spmm_kernel(
    const int64_t* __restrict__ indptr, const int64_t* __restrict__ indices,
    const scalar_t* __restrict__ feat, const scalar_t* __restrict__ ew,
    scalar_t* __restrict__ out, int64_t num_rows, int F, int D, bool mean,
    int64_t thresh) {
  const int chunks = F / VEC;
  const int64_t total = num_rows * chunks;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; tid < total;
       tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = tid / chunks;
    const int c = (int)(tid % chunks);
    const int f0 = c * VEC;
    const int h = (WM == W_HEAD) ? (f0 / D) : 0;
    const int64_t p0 = indptr[row], p1 = indptr[row + 1];
    if (p1 - p0 > thresh) continue;  // hub rows: block-parallel kernel
    using acc_t = typename AccT<scalar_t>::type;
    acc_t acc[VEC];
#pragma unroll
    for (int i = 0; i < VEC; ++i) acc[i] = acc_t(0);
    for (int64_t p = p0; p < p1; ++p) {
      const int64_t u = indices[p];
      acc_t w = acc_t(1);
      if (WM == W_SCALAR) w = (acc_t)ew[p];
      if (WM == W_HEAD) w = (acc_t)ew[p * (F / D) + h];
      const scalar_t* src = feat + u * F + f0;
      if (VEC == 4 && sizeof(scalar_t) == 4) {
        const float4 v = *reinterpret_cast<const float4*>(src);
        acc[0] += w * (acc_t)((const scalar_t*)&v)[0];
        acc[1] += w * (acc_t)((const scalar_t*)&v)[1];
        acc[2] += w * (acc_t)((const scalar_t*)&v)[2];
        acc[3] += w * (acc_t)((const scalar_t*)&v)[3];
      } else if (VEC == 8 && sizeof(scalar_t) == 2) {
        // 8 x bf16/half = 16 B per lane
        const uint4 v = *reinterpret_cast<const uint4*>(src);
        const scalar_t* e = (const scalar_t*)&v;
#pragma unroll
        for (int i = 0; i < 8; ++i) acc[i] += w * (acc_t)e[i];
      } else {
#pragma unroll
        for (int i = 0; i < VEC; ++i) acc[i] += w * (acc_t)src[i];
      }
    }
    if (mean && p1 > p0) {
      const acc_t inv = acc_t(1) / acc_t(p1 - p0);
#pragma unroll
      for (int i = 0; i < VEC; ++i) acc[i] *= inv;
    }
    scalar_t* dst = out + row * F + f0;
#pragma unroll
    for (int i = 0; i < VEC; ++i) dst[i] = (scalar_t)acc[i];
  }
}

// Hub rows (power-law full-graph aggregation / layer-wise inference):
// one 256-thread block per long row; threads split (neighbor stripe x
// feature chunk), partial sums reduced through an LDS tile. Parallelizes
// the neighbor loop 256/chunks-way instead of serializing 10k+ iterations
// per thread.
constexpr int64_t kSpmmLongRow = 256;

template <typename scalar_t, int VEC, WeightMode WM>
__global__ void spmm_long_kernel(
    const int64_t* __restrict__ indptr, const int64_t* __restrict__ indices,
    const scalar_t* __restrict__ feat, const scalar_t* __restrict__ ew,
    scalar_t* __restrict__ out, int64_t num_rows, int F, int D, bool mean,
    int64_t thresh) {
  using acc_t = typename AccT<scalar_t>::type;
  extern __shared__ float sacc[];  // [F] fp32 partial tile
  const int chunks = F / VEC;
  const int tpc = blockDim.x / chunks;  // threads cooperating per chunk
  for (int64_t row = blockIdx.x; row < num_rows; row += gridDim.x) {
    const int64_t p0 = indptr[row], p1 = indptr[row + 1];
    const int64_t deg = p1 - p0;
    if (deg <= thresh) continue;
    for (int i = threadIdx.x; i < F; i += blockDim.x) sacc[i] = 0.f;
    __syncthreads();
    const int c = threadIdx.x % chunks;
    const int stripe = threadIdx.x / chunks;
    const int f0 = c * VEC;
    const int h = (WM == W_HEAD) ? (f0 / D) : 0;
    if (stripe < tpc) {
      float acc[VEC];
#pragma unroll
      for (int i = 0; i < VEC; ++i) acc[i] = 0.f;
      for (int64_t p = p0 + stripe; p < p1; p += tpc) {
        const int64_t u = indices[p];
        float w = 1.f;
        if (WM == W_SCALAR) w = (float)ew[p];
        if (WM == W_HEAD) w = (float)ew[p * (F / D) + h];
        const scalar_t* src = feat + u * F + f0;
#pragma unroll
        for (int i = 0; i < VEC; ++i) acc[i] += w * (float)src[i];
      }
#pragma unroll
      for (int i = 0; i < VEC; ++i) atomicAdd(&sacc[f0 + i], acc[i]);
    }
    __syncthreads();
    const float inv = mean ? 1.f / (float)deg : 1.f;
    for (int i = threadIdx.x; i < F; i += blockDim.x)
      out[row * F + i] = (scalar_t)(sacc[i] * inv);
    __syncthreads();
  }
}

template <typename scalar_t>
static void spmm_launch(const at::Tensor& indptr, const at::Tensor& indices,
                        const at::Tensor& feat, const c10::optional<at::Tensor>& ew,
                        at::Tensor& out, int F, int D, bool mean) {
  const int64_t num_rows = indptr.numel() - 1;
  WeightMode wm = W_NONE;
  const scalar_t* ewp = nullptr;
  if (ew.has_value()) {
    ewp = ew->data_ptr<scalar_t>();
    wm = (ew->dim() == 2) ? W_HEAD : W_SCALAR;
  }
  const int block = 256;
  const bool vec4 = (F % 4 == 0) && (sizeof(scalar_t) == 4) &&
                    (wm != W_HEAD || (D % 4 == 0));
  const bool vec8 = (F % 8 == 0) && (sizeof(scalar_t) == 2) &&
                    (wm != W_HEAD || (D % 8 == 0));
  const int chunks = vec4 ? F / 4 : (vec8 ? F / 8 : F);
  const int grid = grid_for(num_rows * chunks, block);
  // the hub-row kernel needs >=1 thread per feature chunk and an F-float
  // LDS tile; outside that envelope the per-thread kernel takes every row
  const bool long_ok =
      (chunks <= block) && ((size_t)F * sizeof(float) <= 64 * 1024);
  const int64_t thresh =
      long_ok ? kSpmmLongRow : std::numeric_limits<int64_t>::max();
  auto stream = cur_stream();
// NOTE: must be a single statement (do/while) — an unbraced two-statement
// expansion under `if (wm==…) DOA_SPMM(…); else …` re-binds the else to the
// inner `if (long_ok)` and launches the W_SCALAR/W_HEAD long kernels with a
// null edge-weight pointer (GPU memory fault on any >thresh-degree row).
#define DOA_SPMM(V, W)                                                        \
  do {                                                                        \
    hipLaunchKernelGGL((spmm_kernel<scalar_t, V, W>), dim3(grid),             \
                       dim3(block), 0, stream, indptr.data_ptr<int64_t>(),    \
                       indices.data_ptr<int64_t>(),                           \
                       feat.data_ptr<scalar_t>(), ewp,                        \
                       out.data_ptr<scalar_t>(), num_rows, F, D, mean,        \
                       thresh);                                               \
    if (long_ok)                                                              \
      hipLaunchKernelGGL((spmm_long_kernel<scalar_t, V, W>),                  \
                         dim3(grid_for(num_rows * 256, block)), dim3(block),  \
                         F * sizeof(float), stream,                           \
                         indptr.data_ptr<int64_t>(),                          \
                         indices.data_ptr<int64_t>(),                         \
                         feat.data_ptr<scalar_t>(), ewp,                      \
                         out.data_ptr<scalar_t>(), num_rows, F, D, mean,      \
                         kSpmmLongRow);                                       \
  } while (0)
  if (vec4) {
    if (wm == W_NONE) DOA_SPMM(4, W_NONE);
    else if (wm == W_SCALAR) DOA_SPMM(4, W_SCALAR);
    else DOA_SPMM(4, W_HEAD);
  } else if (vec8) {
    if (wm == W_NONE) DOA_SPMM(8, W_NONE);
    else if (wm == W_SCALAR) DOA_SPMM(8, W_SCALAR);
    else DOA_SPMM(8, W_HEAD);
  } else {
    if (wm == W_NONE) DOA_SPMM(1, W_NONE);
    else if (wm == W_SCALAR) DOA_SPMM(1, W_SCALAR);
    else DOA_SPMM(1, W_HEAD);
  }
#undef DOA_SPMM
  DOA_CHECK_HIP(hipGetLastError());
}

at::Tensor spmm(at::Tensor indptr, at::Tensor indices, at::Tensor feat,
                c10::optional<at::Tensor> eweight, bool mean) {
  TORCH_CHECK(feat.is_cuda(), "spmm: feat must be on GPU");
  TORCH_CHECK(indptr.scalar_type() == at::kLong && indices.scalar_type() == at::kLong,
              "spmm: int64 structure expected");
  auto featc = feat.contiguous();
  const int64_t num_rows = indptr.numel() - 1;
  int F = 1, D = 1;
  for (int i = 1; i < featc.dim(); ++i) F *= featc.size(i);
  D = F;
  if (eweight.has_value() && eweight->dim() == 2) {
    const int H = eweight->size(1);
    TORCH_CHECK(F % H == 0, "spmm: feat width not divisible by heads");
    D = F / H;
  }
  std::vector<int64_t> osz;
  osz.push_back(num_rows);
  for (int i = 1; i < featc.dim(); ++i) osz.push_back(featc.size(i));
  auto out = at::empty(osz, featc.options());
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, featc.scalar_type(), "spmm", [&] {
    c10::optional<at::Tensor> ewc;
    if (eweight.has_value()) ewc = eweight->contiguous();
    spmm_launch<scalar_t>(indptr, indices, featc, ewc, out, F, D, mean);
  });
  return out;
}

// ---------------------------------------------------------------------------
// Transposed SpMM as scatter (backward over a Block): walks the SAME CSC
// structure as forward and atomically accumulates grad into the source rows
// — no per-step CSR transpose (the argsort it needs costs more than the
// low-contention fp32 atomics on sampled blocks, where each source node is
// referenced only a handful of times).
//   out[u, :] += w[p] * grad[row, :]   for every csc position p with
//                                      indices[p] == u
// ---------------------------------------------------------------------------
template <typename scalar_t, int VEC, WeightMode WM>
__global__ void spmm_scatter_kernel(
    const int64_t* __restrict__ indptr, const int64_t* __restrict__ indices,
    const scalar_t* __restrict__ grad, const scalar_t* __restrict__ ew,
    float* __restrict__ out, int64_t num_rows, int F, int D) {
  const int chunks = F / VEC;
  const int64_t total = num_rows * chunks;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; tid < total;
       tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = tid / chunks;
    const int c = (int)(tid % chunks);
    const int f0 = c * VEC;
    const int h = (WM == W_HEAD) ? (f0 / D) : 0;
    const int64_t p0 = indptr[row], p1 = indptr[row + 1];
    float g[VEC];
    const scalar_t* grow = grad + row * F + f0;
#pragma unroll
    for (int i = 0; i < VEC; ++i) g[i] = (float)grow[i];
    for (int64_t p = p0; p < p1; ++p) {
      const int64_t u = indices[p];
      float w = 1.f;
      if (WM == W_SCALAR) w = (float)ew[p];
      if (WM == W_HEAD) w = (float)ew[p * (F / D) + h];
      float* dst = out + u * F + f0;
#pragma unroll
      for (int i = 0; i < VEC; ++i) atomicAdd(&dst[i], w * g[i]);
    }
  }
}

template <typename scalar_t>
static void spmm_scatter_launch(const at::Tensor& indptr,
                                const at::Tensor& indices,
                                const at::Tensor& gradc,
                                const c10::optional<at::Tensor>& eweight,
                                at::Tensor& out, int F, int D) {
  const int64_t num_rows = indptr.numel() - 1;
  WeightMode wm = W_NONE;
  const scalar_t* ewp = nullptr;
  c10::optional<at::Tensor> ewc;
  if (eweight.has_value()) {
    ewc = eweight->contiguous();
    ewp = ewc->data_ptr<scalar_t>();
    wm = (ewc->dim() == 2) ? W_HEAD : W_SCALAR;
  }
  const int block = 256;
  const bool vec4 = (F % 4 == 0) && (wm != W_HEAD || (D % 4 == 0));
  const int chunks = vec4 ? F / 4 : F;
  const int grid = grid_for(num_rows * chunks, block);
  auto stream = cur_stream();
#define DOA_SCAT(V, W)                                                        \
  hipLaunchKernelGGL((spmm_scatter_kernel<scalar_t, V, W>), dim3(grid),       \
                     dim3(block), 0, stream, indptr.data_ptr<int64_t>(),      \
                     indices.data_ptr<int64_t>(),                             \
                     gradc.data_ptr<scalar_t>(), ewp,                         \
                     out.data_ptr<float>(), num_rows, F, D)
  if (vec4) {
    if (wm == W_NONE) DOA_SCAT(4, W_NONE);
    else if (wm == W_SCALAR) DOA_SCAT(4, W_SCALAR);
    else DOA_SCAT(4, W_HEAD);
  } else {
    if (wm == W_NONE) DOA_SCAT(1, W_NONE);
    else if (wm == W_SCALAR) DOA_SCAT(1, W_SCALAR);
    else DOA_SCAT(1, W_HEAD);
  }
#undef DOA_SCAT
  DOA_CHECK_HIP(hipGetLastError());
}

at::Tensor spmm_scatter(at::Tensor indptr, at::Tensor indices, at::Tensor grad,
                        c10::optional<at::Tensor> eweight, int64_t num_src) {
  TORCH_CHECK(grad.is_cuda(), "spmm_scatter: grad must be on GPU");
  auto gradc = grad.contiguous();
  int F = 1, D = 1;
  for (int i = 1; i < gradc.dim(); ++i) F *= gradc.size(i);
  D = F;
  if (eweight.has_value() && eweight->dim() == 2) {
    const int H = eweight->size(1);
    D = F / H;
  }
  std::vector<int64_t> osz;
  osz.push_back(num_src);
  for (int i = 1; i < gradc.dim(); ++i) osz.push_back(gradc.size(i));
  // fp32 accumulator (atomicAdd); cast to grad dtype at the end
  auto out = at::zeros(osz, gradc.options().dtype(at::kFloat));
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, gradc.scalar_type(), "spmm_scatter", [&] {
    spmm_scatter_launch<scalar_t>(indptr, indices, gradc, eweight, out, F, D);
  });
  return out.scalar_type() == gradc.scalar_type() ? out
                                                  : out.to(gradc.scalar_type());
}

// ---------------------------------------------------------------------------
// SDDMM u_dot_v: out[e, h] = sum_d fu[src[e], h, d] * fv[dst[e], h, d]
// ---------------------------------------------------------------------------
template <typename scalar_t, int VEC>
__global__ void sddmm_dot_kernel(
    const int64_t* __restrict__ src, const int64_t* __restrict__ dst,
    const scalar_t* __restrict__ fu, const scalar_t* __restrict__ fv,
    scalar_t* __restrict__ out, int64_t E, int H, int D) {
  const int64_t total = E * H;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; tid < total;
       tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t e = tid / H;
    const int h = (int)(tid % H);
    const scalar_t* a = fu + (src[e] * H + h) * D;
    const scalar_t* b = fv + (dst[e] * H + h) * D;
    using acc_t = typename AccT<scalar_t>::type;
    acc_t acc = acc_t(0);
    int d = 0;
    if (VEC == 4 && sizeof(scalar_t) == 4) {
      for (; d + 4 <= D; d += 4) {
        const float4 va = *reinterpret_cast<const float4*>(a + d);
        const float4 vb = *reinterpret_cast<const float4*>(b + d);
        acc += va.x * vb.x + va.y * vb.y + va.z * vb.z + va.w * vb.w;
      }
    }
    for (; d < D; ++d) acc += (acc_t)a[d] * (acc_t)b[d];
    out[tid] = (scalar_t)acc;
  }
}

at::Tensor sddmm_dot(at::Tensor src, at::Tensor dst, at::Tensor feat_u,
                     at::Tensor feat_v) {
  TORCH_CHECK(feat_u.is_cuda(), "sddmm: tensors must be on GPU");
  auto fu = feat_u.contiguous();
  auto fv = feat_v.contiguous();
  const int64_t E = src.numel();
  int H = 1, D = fu.size(-1);
  if (fu.dim() == 3) H = fu.size(1);
  at::Tensor out = (H == 1) ? at::empty({E}, fu.options())
                            : at::empty({E, H}, fu.options());
  const int block = 256;
  const int grid = grid_for(E * H, block);
  auto stream = cur_stream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, fu.scalar_type(), "sddmm_dot", [&] {
    if (D % 4 == 0 && sizeof(scalar_t) == 4) {
      hipLaunchKernelGGL((sddmm_dot_kernel<scalar_t, 4>), dim3(grid),
                         dim3(block), 0, stream, src.data_ptr<int64_t>(),
                         dst.data_ptr<int64_t>(), fu.data_ptr<scalar_t>(),
                         fv.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(), E,
                         H, D);
    } else {
      hipLaunchKernelGGL((sddmm_dot_kernel<scalar_t, 1>), dim3(grid),
                         dim3(block), 0, stream, src.data_ptr<int64_t>(),
                         dst.data_ptr<int64_t>(), fu.data_ptr<scalar_t>(),
                         fv.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(), E,
                         H, D);
    }
  });
  DOA_CHECK_HIP(hipGetLastError());
  return out;
}

// ---------------------------------------------------------------------------
// Edge softmax over in-edge segments (scores in CSC order), [E] or [E, H].
// One thread per (dst, head): online max/sum pass, then a normalize pass.
// ---------------------------------------------------------------------------
template <typename scalar_t>
__global__ void edge_softmax_fwd_kernel(const int64_t* __restrict__ indptr,
                                        const scalar_t* __restrict__ s,
                                        scalar_t* __restrict__ out,
                                        int64_t num_rows, int H,
                                        int64_t thresh) {
  const int64_t total = num_rows * H;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; tid < total;
       tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = tid / H;
    const int h = (int)(tid % H);
    const int64_t p0 = indptr[row], p1 = indptr[row + 1];
    if (p1 - p0 > thresh) continue;  // long rows: wave kernel
    float m = -INFINITY, sum = 0.f;
    for (int64_t p = p0; p < p1; ++p) {
      const float v = (float)s[p * H + h];
      if (v > m) {
        sum = sum * __expf(m - v) + 1.f;
        m = v;
      } else {
        sum += __expf(v - m);
      }
    }
    const float inv = (sum > 0.f) ? 1.f / sum : 0.f;
    for (int64_t p = p0; p < p1; ++p) {
      const float v = (float)s[p * H + h];
      out[p * H + h] = (scalar_t)(__expf(v - m) * inv);
    }
  }
}

template <typename scalar_t>
__global__ void edge_softmax_bwd_kernel(const int64_t* __restrict__ indptr,
                                        const scalar_t* __restrict__ a,
                                        const scalar_t* __restrict__ g,
                                        scalar_t* __restrict__ out,
                                        int64_t num_rows, int H,
                                        int64_t thresh) {
  const int64_t total = num_rows * H;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; tid < total;
       tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = tid / H;
    const int h = (int)(tid % H);
    const int64_t p0 = indptr[row], p1 = indptr[row + 1];
    if (p1 - p0 > thresh) continue;  // long rows: wave kernel
    float acc = 0.f;
    for (int64_t p = p0; p < p1; ++p)
      acc += (float)a[p * H + h] * (float)g[p * H + h];
    for (int64_t p = p0; p < p1; ++p)
      out[p * H + h] =
          (scalar_t)((float)a[p * H + h] * ((float)g[p * H + h] - acc));
  }
}

// Wave-per-(row,head) variants for LONG segments (power-law hubs in
// full-graph attention): lanes stride the segment, online (max, sum) pairs
// merge across lanes with shfl_xor. The thread-per-row kernels above skip
// rows longer than the threshold; these skip the short ones.
template <typename scalar_t>
__global__ void edge_softmax_fwd_long_kernel(
    const int64_t* __restrict__ indptr, const scalar_t* __restrict__ s,
    scalar_t* __restrict__ out, int64_t num_rows, int H, int64_t thresh) {
  const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / kWave;
  const int lane = threadIdx.x % kWave;
  const int64_t nwaves = (int64_t)gridDim.x * blockDim.x / kWave;
  const int64_t total = num_rows * H;
  for (int64_t t = wid; t < total; t += nwaves) {
    const int64_t row = t / H;
    const int h = (int)(t % H);
    const int64_t p0 = indptr[row], p1 = indptr[row + 1];
    if (p1 - p0 <= thresh) continue;
    float m = -INFINITY, sum = 0.f;
    for (int64_t p = p0 + lane; p < p1; p += kWave) {
      const float v = (float)s[p * H + h];
      if (v > m) {
        sum = sum * __expf(m - v) + 1.f;
        m = v;
      } else {
        sum += __expf(v - m);
      }
    }
#pragma unroll
    for (int off = kWave / 2; off > 0; off >>= 1) {
      const float om = __shfl_xor(m, off, kWave);
      const float os = __shfl_xor(sum, off, kWave);
      const float mn = fmaxf(m, om);
      sum = sum * __expf(m - mn) + os * __expf(om - mn);
      m = mn;
    }
    const float inv = (sum > 0.f) ? 1.f / sum : 0.f;
    for (int64_t p = p0 + lane; p < p1; p += kWave)
      out[p * H + h] = (scalar_t)(__expf((float)s[p * H + h] - m) * inv);
  }
}

template <typename scalar_t>
__global__ void edge_softmax_bwd_long_kernel(
    const int64_t* __restrict__ indptr, const scalar_t* __restrict__ a,
    const scalar_t* __restrict__ g, scalar_t* __restrict__ out,
    int64_t num_rows, int H, int64_t thresh) {
  const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / kWave;
  const int lane = threadIdx.x % kWave;
  const int64_t nwaves = (int64_t)gridDim.x * blockDim.x / kWave;
  const int64_t total = num_rows * H;
  for (int64_t t = wid; t < total; t += nwaves) {
    const int64_t row = t / H;
    const int h = (int)(t % H);
    const int64_t p0 = indptr[row], p1 = indptr[row + 1];
    if (p1 - p0 <= thresh) continue;
    float acc = 0.f;
    for (int64_t p = p0 + lane; p < p1; p += kWave)
      acc += (float)a[p * H + h] * (float)g[p * H + h];
#pragma unroll
    for (int off = kWave / 2; off > 0; off >>= 1)
      acc += __shfl_xor(acc, off, kWave);
    for (int64_t p = p0 + lane; p < p1; p += kWave)
      out[p * H + h] =
          (scalar_t)((float)a[p * H + h] * ((float)g[p * H + h] - acc));
  }
}

constexpr int64_t kSoftmaxLongRow = 64;

at::Tensor edge_softmax_fwd(at::Tensor indptr, at::Tensor scores) {
  TORCH_CHECK(scores.is_cuda(), "edge_softmax: scores must be on GPU");
  auto s = scores.contiguous();
  const int64_t num_rows = indptr.numel() - 1;
  const int64_t E = s.size(0);
  if (E == 0) return s.clone();
  const int H = (s.dim() > 1) ? (int)(s.numel() / E) : 1;
  auto out = at::empty_like(s);
  const int block = 256;
  const int grid = grid_for(num_rows * H, block);
  auto stream = cur_stream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, s.scalar_type(), "edge_softmax_fwd", [&] {
    hipLaunchKernelGGL((edge_softmax_fwd_kernel<scalar_t>), dim3(grid),
                       dim3(block), 0, stream, indptr.data_ptr<int64_t>(),
                       s.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(),
                       num_rows, H, kSoftmaxLongRow);
    hipLaunchKernelGGL((edge_softmax_fwd_long_kernel<scalar_t>),
                       dim3(grid_for(num_rows * H * kWave, block)),
                       dim3(block), 0, stream, indptr.data_ptr<int64_t>(),
                       s.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(),
                       num_rows, H, kSoftmaxLongRow);
  });
  DOA_CHECK_HIP(hipGetLastError());
  return out;
}

at::Tensor edge_softmax_bwd(at::Tensor indptr, at::Tensor out,
                            at::Tensor grad_out) {
  auto a = out.contiguous();
  auto g = grad_out.contiguous();
  const int64_t num_rows = indptr.numel() - 1;
  const int64_t E = a.size(0);
  if (E == 0) return a.clone();
  const int H = (a.dim() > 1) ? (int)(a.numel() / E) : 1;
  auto gin = at::empty_like(a);
  const int block = 256;
  const int grid = grid_for(num_rows * H, block);
  auto stream = cur_stream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, a.scalar_type(), "edge_softmax_bwd", [&] {
    hipLaunchKernelGGL((edge_softmax_bwd_kernel<scalar_t>), dim3(grid),
                       dim3(block), 0, stream, indptr.data_ptr<int64_t>(),
                       a.data_ptr<scalar_t>(), g.data_ptr<scalar_t>(),
                       gin.data_ptr<scalar_t>(), num_rows, H,
                       kSoftmaxLongRow);
    hipLaunchKernelGGL((edge_softmax_bwd_long_kernel<scalar_t>),
                       dim3(grid_for(num_rows * H * kWave, block)),
                       dim3(block), 0, stream, indptr.data_ptr<int64_t>(),
                       a.data_ptr<scalar_t>(), g.data_ptr<scalar_t>(),
                       gin.data_ptr<scalar_t>(), num_rows, H,
                       kSoftmaxLongRow);
  });
  DOA_CHECK_HIP(hipGetLastError());
  return gin;
}

// ---------------------------------------------------------------------------
// Segment reduce over contiguous row ranges (mean_nodes readout).
// ---------------------------------------------------------------------------
template <typename scalar_t, int VEC>
__global__ void segment_reduce_kernel(const int64_t* __restrict__ offsets,
                                      const scalar_t* __restrict__ feat,
                                      scalar_t* __restrict__ out,
                                      int64_t num_segs, int F, bool mean) {
  const int chunks = F / VEC;
  const int64_t total = num_segs * chunks;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; tid < total;
       tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t seg = tid / chunks;
    const int f0 = (int)(tid % chunks) * VEC;
    const int64_t r0 = offsets[seg], r1 = offsets[seg + 1];
    using acc_t = typename AccT<scalar_t>::type;
    acc_t acc[VEC];
#pragma unroll
    for (int i = 0; i < VEC; ++i) acc[i] = acc_t(0);
    for (int64_t r = r0; r < r1; ++r) {
      const scalar_t* srcp = feat + r * F + f0;
#pragma unroll
      for (int i = 0; i < VEC; ++i) acc[i] += (acc_t)srcp[i];
    }
    if (mean && r1 > r0) {
      const acc_t inv = acc_t(1) / acc_t(r1 - r0);
#pragma unroll
      for (int i = 0; i < VEC; ++i) acc[i] *= inv;
    }
    scalar_t* dst = out + seg * F + f0;
#pragma unroll
    for (int i = 0; i < VEC; ++i) dst[i] = (scalar_t)acc[i];
  }
}

at::Tensor segment_reduce(at::Tensor offsets, at::Tensor feat, bool mean) {
  TORCH_CHECK(feat.is_cuda(), "segment_reduce: feat must be on GPU");
  auto f = feat.contiguous();
  const int64_t num_segs = offsets.numel() - 1;
  int F = 1;
  for (int i = 1; i < f.dim(); ++i) F *= f.size(i);
  std::vector<int64_t> osz;
  osz.push_back(num_segs);
  for (int i = 1; i < f.dim(); ++i) osz.push_back(f.size(i));
  auto out = at::empty(osz, f.options());
  const int block = 256;
  auto stream = cur_stream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, f.scalar_type(), "segment_reduce", [&] {
    if (F % 4 == 0 && sizeof(scalar_t) == 4) {
      const int grid = grid_for(num_segs * (F / 4), block);
      hipLaunchKernelGGL((segment_reduce_kernel<scalar_t, 4>), dim3(grid),
                         dim3(block), 0, stream, offsets.data_ptr<int64_t>(),
                         f.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(),
                         num_segs, F, mean);
    } else {
      const int grid = grid_for(num_segs * F, block);
      hipLaunchKernelGGL((segment_reduce_kernel<scalar_t, 1>), dim3(grid),
                         dim3(block), 0, stream, offsets.data_ptr<int64_t>(),
                         f.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(),
                         num_segs, F, mean);
    }
  });
  DOA_CHECK_HIP(hipGetLastError());
  return out;
}

}  // namespace doa

namespace doa {

// ---------------------------------------------------------------------------
// Fused row gather: out[i, :] = feat[(map ? map[gids[i]] : gids[i] - offset), :]
// One kernel replaces the two-gather chain of the halo-cache feature pull
// (feat[feat_map[gids]]) and the sub+index_select of the owned-range pull.
// ---------------------------------------------------------------------------
template <typename scalar_t, int VEC>
__global__ void gather_rows_kernel(const scalar_t* __restrict__ feat,
                                   const int64_t* __restrict__ gids,
                                   const int64_t* __restrict__ map,
                                   scalar_t* __restrict__ out, int64_t n,
                                   int F, int64_t offset) {
  const int chunks = F / VEC;
  const int64_t total = n * chunks;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       tid < total; tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t i = tid / chunks;
    const int f0 = (int)(tid % chunks) * VEC;
    const int64_t g = gids[i];
    const int64_t row = map ? map[g] : (g - offset);
    const scalar_t* src = feat + row * F + f0;
    scalar_t* dst = out + i * F + f0;
#pragma unroll
    for (int k = 0; k < VEC; ++k) dst[k] = src[k];
  }
}

at::Tensor gather_rows(at::Tensor feat, at::Tensor gids,
                       c10::optional<at::Tensor> map, int64_t offset) {
  TORCH_CHECK(feat.is_cuda(), "gather_rows: feat must be on GPU");
  auto f = feat.contiguous();
  const int64_t n = gids.numel();
  int F = 1;
  for (int i = 1; i < f.dim(); ++i) F *= f.size(i);
  std::vector<int64_t> osz;
  osz.push_back(n);
  for (int i = 1; i < f.dim(); ++i) osz.push_back(f.size(i));
  auto out = at::empty(osz, f.options());
  const int block = 256;
  auto stream = cur_stream();
  const int64_t* mp = map.has_value() ? map->data_ptr<int64_t>() : nullptr;
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half, f.scalar_type(), "gather_rows", [&] {
    if (F % 4 == 0 && sizeof(scalar_t) == 4) {
      hipLaunchKernelGGL((gather_rows_kernel<scalar_t, 4>),
                         dim3(grid_for(n * (F / 4), block)), dim3(block), 0,
                         stream, f.data_ptr<scalar_t>(),
                         gids.data_ptr<int64_t>(), mp,
                         out.data_ptr<scalar_t>(), n, F, offset);
    } else if (F % 8 == 0 && sizeof(scalar_t) == 2) {
      hipLaunchKernelGGL((gather_rows_kernel<scalar_t, 8>),
                         dim3(grid_for(n * (F / 8), block)), dim3(block), 0,
                         stream, f.data_ptr<scalar_t>(),
                         gids.data_ptr<int64_t>(), mp,
                         out.data_ptr<scalar_t>(), n, F, offset);
    } else {
      hipLaunchKernelGGL((gather_rows_kernel<scalar_t, 1>),
                         dim3(grid_for(n * F, block)), dim3(block), 0,
                         stream, f.data_ptr<scalar_t>(),
                         gids.data_ptr<int64_t>(), mp,
                         out.data_ptr<scalar_t>(), n, F, offset);
    }
  });
  DOA_CHECK_HIP(hipGetLastError());
  return out;
}

}  // namespace doa

namespace doa {

// ---------------------------------------------------------------------------
// Fused GAT attention score (K4-family): s[p,h] = LeakyReLU(el[src[p],h] +
// er[dst[p],h]) over CSC positions — one kernel instead of the
// gather + gather + add + leaky chain; backward scatters into el/er with
// fp32 atomics (the sign is recomputed from el/er, nothing extra saved).
// ---------------------------------------------------------------------------
__global__ void gat_score_fwd_kernel(const int64_t* __restrict__ src,
                                     const int64_t* __restrict__ dst,
                                     const float* __restrict__ el,
                                     const float* __restrict__ er,
                                     float* __restrict__ out, int64_t E,
                                     int H, float slope) {
  const int64_t total = E * H;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       tid < total; tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t p = tid / H;
    const int h = (int)(tid % H);
    const float v = el[src[p] * H + h] + er[dst[p] * H + h];
    out[tid] = v > 0.f ? v : slope * v;
  }
}

__global__ void gat_score_bwd_kernel(const int64_t* __restrict__ src,
                                     const int64_t* __restrict__ dst,
                                     const float* __restrict__ el,
                                     const float* __restrict__ er,
                                     const float* __restrict__ gout,
                                     float* __restrict__ gel,
                                     float* __restrict__ ger, int64_t E,
                                     int H, float slope) {
  const int64_t total = E * H;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       tid < total; tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t p = tid / H;
    const int h = (int)(tid % H);
    const float v = el[src[p] * H + h] + er[dst[p] * H + h];
    const float g = gout[tid] * (v > 0.f ? 1.f : slope);
    atomicAdd(&gel[src[p] * H + h], g);
    atomicAdd(&ger[dst[p] * H + h], g);
  }
}

at::Tensor gat_score_fwd(at::Tensor src, at::Tensor dst, at::Tensor el,
                         at::Tensor er, double slope) {
  TORCH_CHECK(el.is_cuda() && el.scalar_type() == at::kFloat,
              "gat_score: fp32 GPU tensors expected");
  auto elc = el.contiguous();
  auto erc = er.contiguous();
  const int64_t E = src.numel();
  const int H = elc.dim() > 1 ? elc.size(1) : 1;
  auto out = (H == 1) ? at::empty({E}, elc.options())
                      : at::empty({E, H}, elc.options());
  const int block = 256;
  auto stream = cur_stream();
  hipLaunchKernelGGL(gat_score_fwd_kernel, dim3(grid_for(E * H, block)),
                     dim3(block), 0, stream, src.data_ptr<int64_t>(),
                     dst.data_ptr<int64_t>(), elc.data_ptr<float>(),
                     erc.data_ptr<float>(), out.data_ptr<float>(), E, H,
                     (float)slope);
  DOA_CHECK_HIP(hipGetLastError());
  return out;
}

std::tuple<at::Tensor, at::Tensor> gat_score_bwd(at::Tensor src,
                                                 at::Tensor dst,
                                                 at::Tensor el, at::Tensor er,
                                                 at::Tensor gout,
                                                 double slope) {
  auto elc = el.contiguous();
  auto erc = er.contiguous();
  auto g = gout.contiguous();
  const int64_t E = src.numel();
  const int H = elc.dim() > 1 ? elc.size(1) : 1;
  auto gel = at::zeros_like(elc);
  auto ger = at::zeros_like(erc);
  const int block = 256;
  auto stream = cur_stream();
  hipLaunchKernelGGL(gat_score_bwd_kernel, dim3(grid_for(E * H, block)),
                     dim3(block), 0, stream, src.data_ptr<int64_t>(),
                     dst.data_ptr<int64_t>(), elc.data_ptr<float>(),
                     erc.data_ptr<float>(), g.data_ptr<float>(),
                     gel.data_ptr<float>(), ger.data_ptr<float>(), E, H,
                     (float)slope);
  DOA_CHECK_HIP(hipGetLastError());
  return std::make_tuple(gel, ger);
}

}  // namespace doa
