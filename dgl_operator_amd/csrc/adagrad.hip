// Row-sparse Adagrad (K10) — semantics of the reference's KGEServer push
// handler (/root/reference/examples/DGL-KE/hotfix/kvserver.py:41-51):
//   state[ids] += mean(grad^2, dim=1)   (duplicates accumulate)
//   emb[ids]   -= lr * grad / (sqrt(state[ids]) + eps)   (state read AFTER
//                                                        the full state pass)
// Two kernels back-to-back on the same stream give the two-phase ordering.
// Wave-per-row squared-mean reduction; fp32 global atomics for duplicate ids.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace doa {

__global__ void adagrad_state_kernel(const int64_t* __restrict__ ids,
                                     const float* __restrict__ grad,
                                     float* __restrict__ state, int64_t B,
                                     int D) {
  // one wave per row
  const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / kWave;
  const int lane = threadIdx.x % kWave;
  const int64_t nwaves = (int64_t)gridDim.x * blockDim.x / kWave;
  for (int64_t i = wid; i < B; i += nwaves) {
    const float* g = grad + i * D;
    float acc = 0.f;
    for (int d = lane; d < D; d += kWave) acc += g[d] * g[d];
#pragma unroll
    for (int off = kWave / 2; off > 0; off >>= 1)
      acc += __shfl_down(acc, off, kWave);
    if (lane == 0) atomicAdd(&state[ids[i]], acc / D);
  }
}

__global__ void adagrad_apply_kernel(const int64_t* __restrict__ ids,
                                     const float* __restrict__ grad,
                                     const float* __restrict__ state,
                                     float* __restrict__ emb, int64_t B, int D,
                                     float lr, float eps) {
  const int64_t total = B * D;
  for (int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       tid < total; tid += (int64_t)gridDim.x * blockDim.x) {
    const int64_t i = tid / D;
    const int d = (int)(tid % D);
    const int64_t row = ids[i];
    const float std = sqrtf(state[row]) + eps;
    atomicAdd(&emb[row * D + d], -lr * grad[i * D + d] / std);
  }
}

void sparse_adagrad(at::Tensor emb, at::Tensor state, at::Tensor ids,
                    at::Tensor grad, double lr, double eps) {
  TORCH_CHECK(emb.is_cuda() && grad.is_cuda(), "sparse_adagrad: GPU tensors expected");
  TORCH_CHECK(emb.scalar_type() == at::kFloat, "sparse_adagrad: fp32 only");
  const int64_t B = ids.numel();
  const int D = emb.size(1);
  const int block = 256;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  {
    const int grid = grid_for(B * kWave, block);
    hipLaunchKernelGGL(adagrad_state_kernel, dim3(grid), dim3(block), 0,
                       stream, ids.data_ptr<int64_t>(), grad.data_ptr<float>(),
                       state.data_ptr<float>(), B, D);
  }
  {
    const int grid = grid_for(B * D, block);
    hipLaunchKernelGGL(adagrad_apply_kernel, dim3(grid), dim3(block), 0,
                       stream, ids.data_ptr<int64_t>(), grad.data_ptr<float>(),
                       state.data_ptr<float>(), emb.data_ptr<float>(), B, D,
                       (float)lr, (float)eps);
  }
  DOA_CHECK_HIP(hipGetLastError());
}

}  // namespace doa
