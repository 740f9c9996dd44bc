// Fused KGE negative-score kernels (K9) for gfx950.
//
// DGL-KE's chunked corruption scores every (positive, negative) pair inside a
// chunk: with batch 1024 / neg 256 / d 400 the naive broadcast materializes
// [C, c, n, d] tensors (~1.7 GB each in fwd + more in bwd). These kernels
// compute the translational-distance scores pairwise without materializing:
//
//   pdist:   out[ci,i,j]  = gamma - || base[ci,i,:] - neg[ci,j,:] ||_p
//   cpdist:  out[ci,i,j]  = gamma - sum_k sqrt((br-nr)^2 + (bi-ni)^2)
//
// `base` is the torch-side precomputed translation/rotation of the positive
// side (h+r for TransE neg-tail, t-r for neg-head, h rotated by r for
// RotatE), so ONE kernel family serves both corruption sides and gradients
// flow through base by plain autograd. GEMM-shaped scores (DistMult/ComplEx/
// RESCAL) stay on rocBLAS (torch.bmm) where MFMA already serves them.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace doa {

static hipStream_t kge_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// ---------------------------------------------------------------------------
// real L1/L2 pairwise distance
// ---------------------------------------------------------------------------
template <int P>
__global__ void pdist_fwd_kernel(const float* __restrict__ base,
                                 const float* __restrict__ neg,
                                 float* __restrict__ out, int64_t C, int c,
                                 int n, int D, float gamma) {
  const int64_t total = C * c * n;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       tid < total; tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t ci = tid / (c * n);
    const int i = (int)((tid / n) % c);
    const int j = (int)(tid % n);
    const float* b = base + (ci * c + i) * D;
    const float* t = neg + (ci * n + j) * D;
    float acc = 0.f;
    int d = 0;
    for (; d + 4 <= D; d += 4) {
      const float4 vb = *reinterpret_cast<const float4*>(b + d);
      const float4 vt = *reinterpret_cast<const float4*>(t + d);
      const float d0 = vb.x - vt.x, d1 = vb.y - vt.y, d2 = vb.z - vt.z,
                  d3 = vb.w - vt.w;
      if (P == 2)
        acc += d0 * d0 + d1 * d1 + d2 * d2 + d3 * d3;
      else
        acc += fabsf(d0) + fabsf(d1) + fabsf(d2) + fabsf(d3);
    }
    for (; d < D; ++d) {
      const float df = b[d] - t[d];
      acc += (P == 2) ? df * df : fabsf(df);
    }
    out[tid] = gamma - ((P == 2) ? sqrtf(acc) : acc);
  }
}

// grad wrt base: gb[ci,i,d] = sum_j gout[ci,i,j] * dOut/dBase
//   P==2: dOut/dBase = -(b-t)/norm ; P==1: -sign(b-t)
template <int P>
__global__ void pdist_bwd_base_kernel(
    const float* __restrict__ base, const float* __restrict__ neg,
    const float* __restrict__ out, const float* __restrict__ gout,
    float* __restrict__ gbase, int64_t C, int c, int n, int D, float gamma) {
  const int64_t total = C * c * D;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       tid < total; tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t ci = tid / (c * D);
    const int i = (int)((tid / D) % c);
    const int d = (int)(tid % D);
    const float bv = base[(ci * c + i) * D + d];
    float acc = 0.f;
    for (int j = 0; j < n; ++j) {
      const float tv = neg[(ci * n + j) * D + d];
      const float g = gout[(ci * c + i) * n + j];
      if (P == 2) {
        const float norm = gamma - out[(ci * c + i) * n + j];
        acc += -g * (bv - tv) / fmaxf(norm, 1e-12f);
      } else {
        acc += -g * ((bv > tv) ? 1.f : ((bv < tv) ? -1.f : 0.f));
      }
    }
    gbase[tid] = acc;
  }
}

template <int P>
__global__ void pdist_bwd_neg_kernel(
    const float* __restrict__ base, const float* __restrict__ neg,
    const float* __restrict__ out, const float* __restrict__ gout,
    float* __restrict__ gneg, int64_t C, int c, int n, int D, float gamma) {
  const int64_t total = C * n * D;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       tid < total; tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t ci = tid / (n * D);
    const int j = (int)((tid / D) % n);
    const int d = (int)(tid % D);
    const float tv = neg[(ci * n + j) * D + d];
    float acc = 0.f;
    for (int i = 0; i < c; ++i) {
      const float bv = base[(ci * c + i) * D + d];
      const float g = gout[(ci * c + i) * n + j];
      if (P == 2) {
        const float norm = gamma - out[(ci * c + i) * n + j];
        acc += g * (bv - tv) / fmaxf(norm, 1e-12f);
      } else {
        acc += g * ((bv > tv) ? 1.f : ((bv < tv) ? -1.f : 0.f));
      }
    }
    gneg[tid] = acc;
  }
}

at::Tensor pdist_neg_fwd(at::Tensor base, at::Tensor neg, int64_t p,
                         double gamma) {
  TORCH_CHECK(base.is_cuda() && base.scalar_type() == at::kFloat);
  const int64_t C = base.size(0);
  const int c = base.size(1), D = base.size(2), n = neg.size(1);
  auto out = at::empty({C, c, n}, base.options());
  const int block = 256;
  const int grid = grid_for(C * c * n, block);
  if (p == 2)
    hipLaunchKernelGGL((pdist_fwd_kernel<2>), dim3(grid), dim3(block), 0,
                       kge_stream(), base.data_ptr<float>(),
                       neg.data_ptr<float>(), out.data_ptr<float>(), C, c, n,
                       D, (float)gamma);
  else
    hipLaunchKernelGGL((pdist_fwd_kernel<1>), dim3(grid), dim3(block), 0,
                       kge_stream(), base.data_ptr<float>(),
                       neg.data_ptr<float>(), out.data_ptr<float>(), C, c, n,
                       D, (float)gamma);
  DOA_CHECK_HIP(hipGetLastError());
  return out;
}

std::tuple<at::Tensor, at::Tensor> pdist_neg_bwd(at::Tensor base,
                                                 at::Tensor neg,
                                                 at::Tensor out,
                                                 at::Tensor gout, int64_t p,
                                                 double gamma) {
  const int64_t C = base.size(0);
  const int c = base.size(1), D = base.size(2), n = neg.size(1);
  auto gbase = at::empty_like(base);
  auto gneg = at::empty_like(neg);
  const int block = 256;
  auto s = kge_stream();
#define DOA_PD_BWD(P)                                                         \
  hipLaunchKernelGGL((pdist_bwd_base_kernel<P>),                              \
                     dim3(grid_for(C* c* D, block)), dim3(block), 0, s,       \
                     base.data_ptr<float>(), neg.data_ptr<float>(),           \
                     out.data_ptr<float>(), gout.data_ptr<float>(),           \
                     gbase.data_ptr<float>(), C, c, n, D, (float)gamma);      \
  hipLaunchKernelGGL((pdist_bwd_neg_kernel<P>),                               \
                     dim3(grid_for(C* n* D, block)), dim3(block), 0, s,       \
                     base.data_ptr<float>(), neg.data_ptr<float>(),           \
                     out.data_ptr<float>(), gout.data_ptr<float>(),           \
                     gneg.data_ptr<float>(), C, c, n, D, (float)gamma)
  if (p == 2) { DOA_PD_BWD(2); } else { DOA_PD_BWD(1); }
#undef DOA_PD_BWD
  DOA_CHECK_HIP(hipGetLastError());
  return std::make_tuple(gbase, gneg);
}

// ---------------------------------------------------------------------------
// complex modulus distance (RotatE): neg rows are [re | im] halves of width 2*D2
// ---------------------------------------------------------------------------
__global__ void cpdist_fwd_kernel(const float* __restrict__ br,
                                  const float* __restrict__ bi,
                                  const float* __restrict__ neg,
                                  float* __restrict__ out, int64_t C, int c,
                                  int n, int D2, float gamma) {
  const int64_t total = C * c * n;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       tid < total; tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t ci = tid / (c * n);
    const int i = (int)((tid / n) % c);
    const int j = (int)(tid % n);
    const float* r = br + (ci * c + i) * D2;
    const float* im = bi + (ci * c + i) * D2;
    const float* t = neg + (ci * n + j) * 2 * D2;
    float acc = 0.f;
    for (int k = 0; k < D2; ++k) {
      const float dr = r[k] - t[k];
      const float di = im[k] - t[D2 + k];
      acc += sqrtf(dr * dr + di * di);
    }
    out[tid] = gamma - acc;
  }
}

__global__ void cpdist_bwd_base_kernel(
    const float* __restrict__ br, const float* __restrict__ bi,
    const float* __restrict__ neg, const float* __restrict__ gout,
    float* __restrict__ gbr, float* __restrict__ gbi, int64_t C, int c, int n,
    int D2) {
  const int64_t total = C * c * D2;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       tid < total; tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t ci = tid / (c * D2);
    const int i = (int)((tid / D2) % c);
    const int k = (int)(tid % D2);
    const float rv = br[(ci * c + i) * D2 + k];
    const float iv = bi[(ci * c + i) * D2 + k];
    float ar = 0.f, ai = 0.f;
    for (int j = 0; j < n; ++j) {
      const float tr = neg[(ci * n + j) * 2 * D2 + k];
      const float ti = neg[(ci * n + j) * 2 * D2 + D2 + k];
      const float dr = rv - tr, di = iv - ti;
      const float s = fmaxf(sqrtf(dr * dr + di * di), 1e-12f);
      const float g = gout[(ci * c + i) * n + j];
      ar += -g * dr / s;
      ai += -g * di / s;
    }
    gbr[tid] = ar;
    gbi[tid] = ai;
  }
}

__global__ void cpdist_bwd_neg_kernel(
    const float* __restrict__ br, const float* __restrict__ bi,
    const float* __restrict__ neg, const float* __restrict__ gout,
    float* __restrict__ gneg, int64_t C, int c, int n, int D2) {
  const int64_t total = C * n * D2;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       tid < total; tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t ci = tid / (n * D2);
    const int j = (int)((tid / D2) % n);
    const int k = (int)(tid % D2);
    const float tr = neg[(ci * n + j) * 2 * D2 + k];
    const float ti = neg[(ci * n + j) * 2 * D2 + D2 + k];
    float ar = 0.f, ai = 0.f;
    for (int i = 0; i < c; ++i) {
      const float dr = br[(ci * c + i) * D2 + k] - tr;
      const float di = bi[(ci * c + i) * D2 + k] - ti;
      const float s = fmaxf(sqrtf(dr * dr + di * di), 1e-12f);
      const float g = gout[(ci * c + i) * n + j];
      ar += g * dr / s;
      ai += g * di / s;
    }
    gneg[(ci * n + j) * 2 * D2 + k] = ar;
    gneg[(ci * n + j) * 2 * D2 + D2 + k] = ai;
  }
}

at::Tensor cpdist_neg_fwd(at::Tensor base_r, at::Tensor base_i, at::Tensor neg,
                          double gamma) {
  TORCH_CHECK(base_r.is_cuda() && base_r.scalar_type() == at::kFloat);
  const int64_t C = base_r.size(0);
  const int c = base_r.size(1), D2 = base_r.size(2), n = neg.size(1);
  TORCH_CHECK(neg.size(2) == 2 * D2, "neg width must be 2*D2");
  auto out = at::empty({C, c, n}, base_r.options());
  const int block = 256;
  hipLaunchKernelGGL(cpdist_fwd_kernel, dim3(grid_for(C * c * n, block)),
                     dim3(block), 0, kge_stream(), base_r.data_ptr<float>(),
                     base_i.data_ptr<float>(), neg.data_ptr<float>(),
                     out.data_ptr<float>(), C, c, n, D2, (float)gamma);
  DOA_CHECK_HIP(hipGetLastError());
  return out;
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> cpdist_neg_bwd(
    at::Tensor base_r, at::Tensor base_i, at::Tensor neg, at::Tensor gout) {
  const int64_t C = base_r.size(0);
  const int c = base_r.size(1), D2 = base_r.size(2), n = neg.size(1);
  auto gbr = at::empty_like(base_r);
  auto gbi = at::empty_like(base_i);
  auto gneg = at::empty_like(neg);
  const int block = 256;
  auto s = kge_stream();
  hipLaunchKernelGGL(cpdist_bwd_base_kernel,
                     dim3(grid_for(C * c * D2, block)), dim3(block), 0, s,
                     base_r.data_ptr<float>(), base_i.data_ptr<float>(),
                     neg.data_ptr<float>(), gout.data_ptr<float>(),
                     gbr.data_ptr<float>(), gbi.data_ptr<float>(), C, c, n,
                     D2);
  hipLaunchKernelGGL(cpdist_bwd_neg_kernel,
                     dim3(grid_for(C * n * D2, block)), dim3(block), 0, s,
                     base_r.data_ptr<float>(), base_i.data_ptr<float>(),
                     neg.data_ptr<float>(), gout.data_ptr<float>(),
                     gneg.data_ptr<float>(), C, c, n, D2);
  DOA_CHECK_HIP(hipGetLastError());
  return std::make_tuple(gbr, gbi, gneg);
}

}  // namespace doa
