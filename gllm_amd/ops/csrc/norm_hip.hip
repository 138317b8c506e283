#include "hip/hip_runtime.h"
// RMSNorm + fused residual-add RMSNorm for gfx950.
// Memory-bound: vectorized 16 B/lane loads (guide G13: scalar bf16 loads
// are ~2x slower). One workgroup per token row; fp32 accumulation.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

template <typename T, int BLOCK>
__global__ void rmsnorm_kernel(T *__restrict__ out, const T *__restrict__ in,
                               const T *__restrict__ weight, float eps,
                               int hidden) {
  __shared__ float red[BLOCK / WAVE_SIZE];
  const int row = blockIdx.x;
  const T *x = in + (long)row * hidden;
  T *o = out + (long)row * hidden;
  const int nvec = hidden / 8;

  float ss = 0.f;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    shortx8 p = reinterpret_cast<const shortx8 *>(x)[i];
    float v[8];
    unpack8<T>(p, v);
#pragma unroll
    for (int j = 0; j < 8; ++j) ss += v[j] * v[j];
  }
  ss = block_reduce_sum<BLOCK>(ss, red);
  const float inv = rsqrtf(ss / hidden + eps);

  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    shortx8 p = reinterpret_cast<const shortx8 *>(x)[i];
    shortx8 w = reinterpret_cast<const shortx8 *>(weight)[i];
    float v[8], wv[8];
    unpack8<T>(p, v);
    unpack8<T>(w, wv);
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = v[j] * inv * wv[j];
    reinterpret_cast<shortx8 *>(o)[i] = pack8<T>(v);
  }
}

// residual += x; x = rmsnorm(residual) — both in place.
// The summed row is staged in LDS (bf16, exactly what was stored to
// residual) so the normalize pass re-reads LDS instead of HBM.
template <typename T, int BLOCK>
__global__ void fused_add_rmsnorm_kernel(T *__restrict__ x,
                                         T *__restrict__ residual,
                                         const T *__restrict__ weight,
                                         float eps, int hidden) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  T *srow = reinterpret_cast<T *>(smem_raw);
  __shared__ float red[BLOCK / WAVE_SIZE];
  const int row = blockIdx.x;
  T *xr = x + (long)row * hidden;
  T *rr = residual + (long)row * hidden;
  const int nvec = hidden / 8;

  float ss = 0.f;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    shortx8 px = reinterpret_cast<const shortx8 *>(xr)[i];
    shortx8 pr = reinterpret_cast<const shortx8 *>(rr)[i];
    float vx[8], vr[8], vs[8];
    unpack8<T>(px, vx);
    unpack8<T>(pr, vr);
#pragma unroll
    for (int j = 0; j < 8; ++j) vs[j] = vx[j] + vr[j];
    shortx8 packed = pack8<T>(vs);
    // variance over the ROUNDED stored residual (matches the torch oracle,
    // which norms the bf16 residual it just stored)
    float vq[8];
    unpack8<T>(packed, vq);
#pragma unroll
    for (int j = 0; j < 8; ++j) ss += vq[j] * vq[j];
    reinterpret_cast<shortx8 *>(rr)[i] = packed;
    reinterpret_cast<shortx8 *>(srow)[i] = packed;
  }
  ss = block_reduce_sum<BLOCK>(ss, red);
  const float inv = rsqrtf(ss / hidden + eps);

  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    shortx8 p = reinterpret_cast<const shortx8 *>(srow)[i];
    shortx8 w = reinterpret_cast<const shortx8 *>(weight)[i];
    float v[8], wv[8];
    unpack8<T>(p, v);
    unpack8<T>(w, wv);
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = v[j] * inv * wv[j];
    reinterpret_cast<shortx8 *>(xr)[i] = pack8<T>(v);
  }
}

}  // namespace

#define DISPATCH_16BIT(TENSOR, NAME, ...)                                   \
  do {                                                                      \
    if ((TENSOR).scalar_type() == at::kBFloat16) {                          \
      using scalar_t = __hip_bfloat16;                                      \
      __VA_ARGS__;                                                          \
    } else if ((TENSOR).scalar_type() == at::kHalf) {                       \
      using scalar_t = __half;                                              \
      __VA_ARGS__;                                                          \
    } else {                                                                \
      TORCH_CHECK(false, NAME ": unsupported dtype ",                       \
                  (TENSOR).scalar_type());                                  \
    }                                                                       \
  } while (0)

void rmsnorm(torch::Tensor out, torch::Tensor in, torch::Tensor weight,
             double eps) {
  const int hidden = in.size(-1);
  const long rows = in.numel() / hidden;
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  TORCH_CHECK(in.is_contiguous() && out.is_contiguous());
  constexpr int BLOCK = 256;
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  DISPATCH_16BIT(in, "rmsnorm", {
    hipLaunchKernelGGL((rmsnorm_kernel<scalar_t, BLOCK>), dim3(rows),
                       dim3(BLOCK), 0, stream,
                       (scalar_t *)out.data_ptr(),
                       (const scalar_t *)in.data_ptr(),
                       (const scalar_t *)weight.data_ptr(), (float)eps,
                       hidden);
  });
  HIP_CHECK_KERNEL();
}

void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                       torch::Tensor weight, double eps) {
  const int hidden = x.size(-1);
  const long rows = x.numel() / hidden;
  TORCH_CHECK(hidden % 8 == 0);
  TORCH_CHECK(x.is_contiguous() && residual.is_contiguous());
  constexpr int BLOCK = 256;
  const int lds = hidden * x.element_size();
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  DISPATCH_16BIT(x, "fused_add_rmsnorm", {
    hipLaunchKernelGGL((fused_add_rmsnorm_kernel<scalar_t, BLOCK>),
                       dim3(rows), dim3(BLOCK), lds, stream,
                       (scalar_t *)x.data_ptr(),
                       (scalar_t *)residual.data_ptr(),
                       (const scalar_t *)weight.data_ptr(), (float)eps,
                       hidden);
  });
  HIP_CHECK_KERNEL();
}
