// Gated DeltaNet (GDN) decode kernels for gfx950.
//
// Reference parity: the vendored FLA Triton kernels'
// fused_recurrent_gated_delta_rule decode path + causal_conv1d_update +
// gated RMSNorm (layers/ops/fla/fused_recurrent.py:16,
// layers/ops/mamba/causal_conv1d_triton.py:584, fused_norm_gate.py).
// The r1 torch path looped PER SEQUENCE with ~8 launches each; these
// kernels process the whole decode batch in one launch per op and
// stream each recurrent state S [Dv, Dk] exactly once (read + write).
//
// Recurrence per (seq, v-head), single token:
//   Sk   = exp(g) * (S_old @ k)
//   u    = beta * (v - Sk)            (per v-row)
//   S    = exp(g) * S_old + u k^T
//   o    = S @ q
// One pass per S row: dot against k (wave reduce), update, dot against
// q. Wave = one Dv row (64 lanes x float2 over Dk=128), 4 waves/WG.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

// ---------------------------------------------------------- conv update
// x [B, C] input token; weight [C, K]; conv_state [slots, C, K-1]
// (rolls in place); out [B, C] = silu(conv).
template <int K>
__global__ void gdn_conv_update_kernel(
    __hip_bfloat16 *__restrict__ out, const __hip_bfloat16 *__restrict__ x,
    const __hip_bfloat16 *__restrict__ weight,
    __hip_bfloat16 *__restrict__ conv_state,
    const long *__restrict__ slots, int C, long state_stride) {
  const int b = blockIdx.x;
  const long slot = slots[b];
  __hip_bfloat16 *st = conv_state + slot * state_stride;
  for (int c = blockIdx.y * blockDim.x + threadIdx.x; c < C;
       c += gridDim.y * blockDim.x) {
    float ctx[K];
#pragma unroll
    for (int j = 0; j < K - 1; ++j)
      ctx[j] = __bfloat162float(st[(long)c * (K - 1) + j]);
    ctx[K - 1] = __bfloat162float(x[(long)b * C + c]);
    float acc = 0.f;
#pragma unroll
    for (int j = 0; j < K; ++j)
      acc += ctx[j] * __bfloat162float(weight[(long)c * K + j]);
#pragma unroll
    for (int j = 0; j < K - 1; ++j)
      st[(long)c * (K - 1) + j] = __float2bfloat16(ctx[j + 1]);
    const float s = acc / (1.f + __expf(-acc));
    out[(long)b * C + c] = __float2bfloat16(s);
  }
}

// ---------------------------------------------------------- delta decode
// qn/kn [B, Hv, Dk] fp32 (L2-normalized, q pre-scaled); v [B, Hv, Dv]
// fp32; g, beta [B, Hv] fp32; state pool [slots, Hv, Dv, Dk] fp32.
template <int DK>
__global__ __launch_bounds__(256) void gdn_decode_kernel(
    __hip_bfloat16 *__restrict__ o,        // [B, Hv, Dv]
    const float *__restrict__ qn, const float *__restrict__ kn,
    const float *__restrict__ v, const float *__restrict__ g,
    const float *__restrict__ beta, float *__restrict__ state,
    const long *__restrict__ slots, int Hv, int Dv, long slot_stride,
    long head_stride) {
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  const long slot = slots[b];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  constexpr int PER_LANE = DK / 64;  // float2 for DK=128

  float *S = state + slot * slot_stride + h * head_stride;
  const float eg = __expf(g[(long)b * Hv + h]);
  const float bt = beta[(long)b * Hv + h];

  float kf[PER_LANE], qf[PER_LANE];
#pragma unroll
  for (int j = 0; j < PER_LANE; ++j) {
    kf[j] = kn[((long)b * Hv + h) * DK + lane * PER_LANE + j];
    qf[j] = qn[((long)b * Hv + h) * DK + lane * PER_LANE + j];
  }

  for (int row = wave; row < Dv; row += 4) {
    float *srow = S + (long)row * DK + lane * PER_LANE;
    float s[PER_LANE];
    float dot = 0.f;
#pragma unroll
    for (int j = 0; j < PER_LANE; ++j) {
      s[j] = srow[j];
      dot += s[j] * kf[j];
    }
    dot = wave_reduce_sum(dot);
    const float u = bt * (v[((long)b * Hv + h) * Dv + row] - eg * dot);
    float od = 0.f;
#pragma unroll
    for (int j = 0; j < PER_LANE; ++j) {
      s[j] = eg * s[j] + u * kf[j];
      srow[j] = s[j];
      od += s[j] * qf[j];
    }
    od = wave_reduce_sum(od);
    if (lane == 0)
      o[((long)b * Hv + h) * Dv + row] = __float2bfloat16(od);
  }
}

// ---------------------------------------------------------- gated norm
// out = rmsnorm(x) * w * silu(z); rows [N, D], one wave per row.
__global__ void rmsnorm_gated_kernel(__hip_bfloat16 *__restrict__ out,
                                     const __hip_bfloat16 *__restrict__ x,
                                     const __hip_bfloat16 *__restrict__ z,
                                     const __hip_bfloat16 *__restrict__ w,
                                     float eps, long N, int D) {
  const long row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= N) return;
  const int lane = threadIdx.x & 63;
  const __hip_bfloat16 *xr = x + row * D;
  const __hip_bfloat16 *zr = z + row * D;
  float ss = 0.f;
  for (int i = lane; i < D; i += 64) {
    const float xv = __bfloat162float(xr[i]);
    ss += xv * xv;
  }
  ss = wave_reduce_sum(ss);
  const float inv = rsqrtf(ss / D + eps);
  for (int i = lane; i < D; i += 64) {
    const float xv = __bfloat162float(xr[i]);
    const float zv = __bfloat162float(zr[i]);
    const float sz = zv / (1.f + __expf(-zv));
    out[row * D + i] =
        __float2bfloat16(xv * inv * __bfloat162float(w[i]) * sz);
  }
}

}  // namespace

void gdn_conv_update(torch::Tensor out, torch::Tensor x,
                     torch::Tensor weight, torch::Tensor conv_state,
                     torch::Tensor slots) {
  const int B = x.size(0), C = x.size(1);
  const int K = weight.size(1);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(conv_state.scalar_type() == at::kBFloat16);
  TORCH_CHECK(conv_state.stride(1) == K - 1,
              "conv_state rows must be contiguous");
  TORCH_CHECK(slots.scalar_type() == at::kLong);
  auto stream = at::cuda::getCurrentCUDAStream();
  const int ygrid = std::min(16, (C + 255) / 256);
  TORCH_CHECK(K == 4, "gdn conv: kernel width 4 expected");
  hipLaunchKernelGGL((gdn_conv_update_kernel<4>), dim3(B, ygrid),
                     dim3(256), 0, stream,
                     (__hip_bfloat16 *)out.data_ptr(),
                     (const __hip_bfloat16 *)x.data_ptr(),
                     (const __hip_bfloat16 *)weight.data_ptr(),
                     (__hip_bfloat16 *)conv_state.data_ptr(),
                     slots.data_ptr<long>(), C, conv_state.stride(0));
  HIP_CHECK_KERNEL();
}

void gdn_decode(torch::Tensor o, torch::Tensor qn, torch::Tensor kn,
                torch::Tensor v, torch::Tensor g, torch::Tensor beta,
                torch::Tensor state, torch::Tensor slots) {
  const int B = qn.size(0), Hv = qn.size(1), Dk = qn.size(2);
  const int Dv = v.size(2);
  TORCH_CHECK(qn.scalar_type() == at::kFloat && qn.is_contiguous());
  TORCH_CHECK(state.scalar_type() == at::kFloat);
  TORCH_CHECK(state.stride(3) == 1 && state.stride(2) == Dk,
              "state head must be contiguous [Dv, Dk]");
  TORCH_CHECK(Dk == 128, "gdn decode: Dk=128 expected");
  TORCH_CHECK(slots.scalar_type() == at::kLong);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL((gdn_decode_kernel<128>), dim3(B, Hv), dim3(256), 0,
                     stream, (__hip_bfloat16 *)o.data_ptr(),
                     qn.data_ptr<float>(), kn.data_ptr<float>(),
                     v.data_ptr<float>(), g.data_ptr<float>(),
                     beta.data_ptr<float>(), state.data_ptr<float>(),
                     slots.data_ptr<long>(), Hv, Dv, state.stride(0),
                     state.stride(1));
  HIP_CHECK_KERNEL();
}

void rmsnorm_gated(torch::Tensor out, torch::Tensor x, torch::Tensor z,
                   torch::Tensor w, double eps) {
  const long N = x.size(0);
  const int D = x.size(1);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(z.is_contiguous() && out.is_contiguous());
  auto stream = at::cuda::getCurrentCUDAStream();
  const long grid = (N + 3) / 4;
  hipLaunchKernelGGL(rmsnorm_gated_kernel, dim3(grid), dim3(256), 0,
                     stream, (__hip_bfloat16 *)out.data_ptr(),
                     (const __hip_bfloat16 *)x.data_ptr(),
                     (const __hip_bfloat16 *)z.data_ptr(),
                     (const __hip_bfloat16 *)w.data_ptr(), (float)eps, N,
                     D);
  HIP_CHECK_KERNEL();
}
