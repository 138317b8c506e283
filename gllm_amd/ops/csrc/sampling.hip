// Sampling-path kernels. Reference: layers/repetition_penalty.py
// (_scaling_penalty_kernel Triton) — per-row scaling penalty against a
// persistent uint8 seen-token mask pool (core/penalty.py). Memory
// bound: [B, V] fp32 logits read+write plus [B, V] uint8 mask reads;
// a single pass fused with the slot gather.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

namespace {

__global__ void rep_penalty_kernel(float* __restrict__ logits,
                                   const unsigned char* __restrict__ pool,
                                   long pool_stride,
                                   const long* __restrict__ slots,
                                   const float* __restrict__ penalties,
                                   int V) {
  int b = blockIdx.y;
  long slot = slots[b];
  float p = penalties[b];
  if (slot < 0 || p == 1.f) return;
  const unsigned char* mask = pool + slot * pool_stride;
  float* row = logits + (long)b * V;
  float inv = 1.f / p;
  int i = (blockIdx.x * blockDim.x + threadIdx.x) * 4;
  int stride = gridDim.x * blockDim.x * 4;
  for (; i < V; i += stride) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int k = i + j;
      if (k < V && mask[k]) {
        float v = row[k];
        row[k] = v > 0.f ? v * inv : v * p;
      }
    }
  }
}

}  // namespace

void apply_repetition_penalty(torch::Tensor logits, torch::Tensor pool,
                              torch::Tensor slots,
                              torch::Tensor penalties) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == at::kFloat &&
              logits.is_contiguous());
  TORCH_CHECK(pool.dtype() == at::kByte && pool.is_contiguous());
  TORCH_CHECK(slots.dtype() == at::kLong);
  int B = logits.size(0), V = logits.size(1);
  TORCH_CHECK(pool.size(1) >= V, "mask pool narrower than vocab");
  auto stream = at::cuda::getCurrentCUDAStream();
  dim3 grid((V + 4 * 256 - 1) / (4 * 256), B);
  rep_penalty_kernel<<<grid, 256, 0, stream>>>(
      logits.data_ptr<float>(), pool.data_ptr<unsigned char>(),
      pool.stride(0), slots.data_ptr<long>(),
      penalties.data_ptr<float>(), V);
}
