// Fused gather + tall-skinny GEMM on the fp32 matrix cores (gfx950).
//
//   out[i, n] = sum_k feat[rows[i], k] * W[k, n]  (+ bias[n])
//
// The GNN input projection (SAGEConv layer 0: K=feat_dim~100 -> N=hidden~16)
// reads gathered feature rows exactly once if the gather is fused into the
// GEMM — the eager path materializes x = feat[rows] (write + re-read, the
// single largest HBM stream of the minibatch step; docs/perf_model.md).
//
// MFMA: v_mfma_f32_16x16x4_f32 (exact fp32, 16x16 tile, K-step 4;
// /opt/skills/guides/cdna_hip_programming.md §3: A frag lane l holds
// A[l&15][l>>4], B frag B[l>>4][l&15], C/D col=lane&15,
// row=(lane>>4)*4+reg). Each 256-thread block stages 64 gathered rows
// (K padded to a multiple of 4) + the whole W panel in LDS; each of the
// 4 waves owns a 16-row tile and iterates the K loop in steps of 4.
// N <= 16 (one tile column) — the GNN hidden sizes this fuses for.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace doa {

using f32x4 = __attribute__((__vector_size__(4 * sizeof(float)))) float;

constexpr int GM_ROWS = 64;   // rows per block (4 waves x 16)
constexpr int GM_N = 16;      // output tile width (N padded to 16)

__global__ __launch_bounds__(256) void gather_mm_kernel(
    const float* __restrict__ feat, const int64_t* __restrict__ rows,
    const float* __restrict__ W,    // [K, N] row-major
    const float* __restrict__ bias,  // [N] or nullptr
    float* __restrict__ out,         // [M, N] row-major
    int64_t M, int K, int N) {
  const int Kp = (K + 3) & ~3;   // K padded to a multiple of 4
  const int As = Kp + 1;          // A row stride: odd*4B => conflict-free
                                  // column reads across the 16-row group
  extern __shared__ float lds[];
  // lds layout: A [GM_ROWS][As]  then  W [Kp][GM_N]
  float* a_lds = lds;
  float* w_lds = lds + GM_ROWS * As;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  // stage W once (zero-padded to [Kp][GM_N])
  for (int idx = tid; idx < Kp * GM_N; idx += blockDim.x) {
    const int k = idx / GM_N, n = idx % GM_N;
    w_lds[idx] = (k < K && n < N) ? W[k * N + n] : 0.f;
  }

  for (int64_t base = (int64_t)blockIdx.x * GM_ROWS; base < M;
       base += (int64_t)gridDim.x * GM_ROWS) {
    __syncthreads();  // W ready / previous tile's reads done
    // stage up to 64 gathered rows; 4 threads cooperate per row so global
    // reads stay coalesced across the row's K elements
    for (int r = tid / 4; r < GM_ROWS; r += blockDim.x / 4) {
      const int64_t i = base + r;
      const float* src =
          (i < M) ? feat + rows[i] * K : nullptr;
      for (int k = (tid & 3); k < Kp; k += 4) {
        a_lds[r * As + k] = (src != nullptr && k < K) ? src[k] : 0.f;
      }
    }
    __syncthreads();

    // wave computes its 16-row tile: acc[r] covers rows
    // (wave*16 + (lane>>4)*4 + r), col (lane&15)
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const int a_row = wave * 16 + (lane & 15);  // A frag row (i = l&15)
    const int a_k = lane >> 4;                  // A frag k   (k = l>>4)
    for (int kk = 0; kk < Kp; kk += 4) {
      const float a = a_lds[a_row * As + kk + a_k];
      const float b = w_lds[(kk + a_k) * GM_N + (lane & 15)];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
    }

    const int out_col = lane & 15;
    if (out_col < N) {
      const float badd = bias ? bias[out_col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int64_t i = base + wave * 16 + (lane >> 4) * 4 + r;
        if (i < M) out[i * N + out_col] = acc[r] + badd;
      }
    }
  }
}

at::Tensor gather_mm(at::Tensor feat, at::Tensor rows, at::Tensor weight,
                     c10::optional<at::Tensor> bias) {
  TORCH_CHECK(feat.is_cuda() && feat.scalar_type() == at::kFloat,
              "gather_mm: fp32 GPU feat expected");
  TORCH_CHECK(weight.dim() == 2, "gather_mm: weight must be [K, N]");
  auto f = feat.contiguous();
  auto w = weight.contiguous();
  const int64_t M = rows.numel();
  const int K = f.size(1);
  const int N = w.size(1);
  TORCH_CHECK(f.dim() == 2, "gather_mm: feat must be [num_nodes, K]");
  TORCH_CHECK(w.size(0) == K, "gather_mm: weight K mismatch");
  TORCH_CHECK(N <= GM_N, "gather_mm: N must be <= 16");
  auto out = at::empty({M, N}, f.options());
  if (M == 0) return out;
  const int Kp = (K + 3) & ~3;
  const size_t lds_bytes =
      (GM_ROWS * (Kp + 1) + Kp * GM_N) * sizeof(float);
  // HIP caps dynamic LDS at 64 KiB unless the max-dynamic-shared attribute
  // is raised; 80*Kp + 64 floats <= 16384 => Kp <= 204 (callers guard at
  // K <= 204 and fall back to gather_rows + rocBLAS GEMM beyond)
  TORCH_CHECK(lds_bytes <= 64 * 1024, "gather_mm: K too large for LDS");
  const int grid = grid_for(ceil_div(M, GM_ROWS) * 256, 256);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  at::Tensor bc;  // keep the contiguous bias alive across the async launch
  const float* bp = nullptr;
  if (bias.has_value()) {
    bc = bias->contiguous();
    bp = bc.data_ptr<float>();
  }
  hipLaunchKernelGGL(gather_mm_kernel, dim3(grid), dim3(256), lds_bytes,
                     stream, f.data_ptr<float>(), rows.data_ptr<int64_t>(),
                     w.data_ptr<float>(), bp, out.data_ptr<float>(), M, K, N);
  DOA_CHECK_HIP(hipGetLastError());
  return out;
}

}  // namespace doa
