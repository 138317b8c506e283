// silu_and_mul + RoPE + reshape_and_cache for gfx950.
// All memory-bound: grid-stride, 16 B/lane vector loads.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

// out[t, i] = silu(x[t, i]) * x[t, d + i],  x: [T, 2d]
template <typename T>
__global__ void silu_and_mul_kernel(T *__restrict__ out,
                                    const T *__restrict__ x, long rows,
                                    int d) {
  const int nvec = d / 8;
  const long total = rows * nvec;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / nvec;
    const int i = idx % nvec;
    shortx8 a = reinterpret_cast<const shortx8 *>(x + row * 2 * d)[i];
    shortx8 b = reinterpret_cast<const shortx8 *>(x + row * 2 * d + d)[i];
    float va[8], vb[8];
    unpack8<T>(a, va);
    unpack8<T>(b, vb);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float s = va[j] / (1.f + __expf(-va[j]));
      va[j] = s * vb[j];
    }
    reinterpret_cast<shortx8 *>(out + row * d)[i] = pack8<T>(va);
  }
}

// gelu(x[:, :d]) * x[:, d:] — erf-based GELU (reference gelu_and_mul,
// the MoE activation="gelu" path in fused_moe.py:943).
template <typename T>
__global__ void gelu_and_mul_kernel(T *__restrict__ out,
                                    const T *__restrict__ x, long rows,
                                    int d) {
  const int nvec = d / 8;
  const long total = rows * nvec;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / nvec;
    const int i = idx % nvec;
    shortx8 a = reinterpret_cast<const shortx8 *>(x + row * 2 * d)[i];
    shortx8 b = reinterpret_cast<const shortx8 *>(x + row * 2 * d + d)[i];
    float va[8], vb[8];
    unpack8<T>(a, va);
    unpack8<T>(b, vb);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float g = 0.5f * va[j] * (1.f + erff(va[j] * 0.70710678f));
      va[j] = g * vb[j];
    }
    reinterpret_cast<shortx8 *>(out + row * d)[i] = pack8<T>(va);
  }
}

// In-place neox / interleaved RoPE on q [T, Hq*D] and k [T, Hk*D].
// cos_sin_cache: [max_pos, rot_dim] fp32, [cos | sin] halves (host-built
// — no device trig, guide App. B). One block per token.
template <typename T, bool NEOX>
__global__ void rope_kernel(const long *__restrict__ positions,
                            T *__restrict__ q, T *__restrict__ k,
                            const float *__restrict__ cache, int rot_dim,
                            int head_dim, int num_q_heads, int num_k_heads,
                            long q_stride, long k_stride) {
  const long t = blockIdx.x;
  const float *cs = cache + positions[t] * rot_dim;
  const int half = rot_dim / 2;
  const int total = (num_q_heads + num_k_heads) * half;
  for (int idx = threadIdx.x; idx < total; idx += blockDim.x) {
    const int h = idx / half;
    const int p = idx % half;           // rotation pair index
    T *base = (h < num_q_heads)
                  ? q + t * q_stride + h * head_dim
                  : k + t * k_stride + (h - num_q_heads) * head_dim;
    const float c = cs[p], s = cs[half + p];
    int i1, i2;
    if (NEOX) { i1 = p; i2 = p + half; }
    else      { i1 = 2 * p; i2 = 2 * p + 1; }
    const float x1 = ScalarOps<T>::to_f32(base[i1]);
    const float x2 = ScalarOps<T>::to_f32(base[i2]);
    base[i1] = ScalarOps<T>::from_f32(x1 * c - x2 * s);
    base[i2] = ScalarOps<T>::from_f32(x2 * c + x1 * s);
  }
}

// Scatter per-token K/V [T, H, D] into paged caches [P, page, H, D].
// One block per token; row copy is H*D*2 bytes, vectorized.
template <typename T>
__global__ void reshape_and_cache_kernel(
    const T *__restrict__ k, const T *__restrict__ v, T *__restrict__ k_cache,
    T *__restrict__ v_cache, const long *__restrict__ slot_mapping,
    int k_row_elems /* H*Dk */, int v_row_elems /* H*Dv (MLA: != k) */,
    int page_size, long k_stride, long v_stride) {
  const long t = blockIdx.x;
  const long slot = slot_mapping[t];
  const long page = slot / page_size, off = slot % page_size;
  const long tok = page * page_size + off;
  const int knvec = k_row_elems / 8, vnvec = v_row_elems / 8;
  const shortx8 *ks = reinterpret_cast<const shortx8 *>(k + t * k_stride);
  const shortx8 *vs = reinterpret_cast<const shortx8 *>(v + t * v_stride);
  shortx8 *kd = reinterpret_cast<shortx8 *>(k_cache + tok * (long)k_row_elems);
  shortx8 *vd = reinterpret_cast<shortx8 *>(v_cache + tok * (long)v_row_elems);
  for (int i = threadIdx.x; i < knvec; i += blockDim.x) kd[i] = ks[i];
  for (int i = threadIdx.x; i < vnvec; i += blockDim.x) vd[i] = vs[i];
}

// fp32 variant: floatx4 (16 B) vector loads instead of shortx8.
__global__ void silu_and_mul_f32_kernel(float *__restrict__ out,
                                        const float *__restrict__ x,
                                        long rows, int d) {
  const int nvec = d / 4;
  const long total = rows * nvec;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / nvec;
    const int i = idx % nvec;
    floatx4 a = reinterpret_cast<const floatx4 *>(x + row * 2 * d)[i];
    floatx4 b = reinterpret_cast<const floatx4 *>(x + row * 2 * d + d)[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float s = a[j] / (1.f + __expf(-a[j]));
      a[j] = s * b[j];
    }
    reinterpret_cast<floatx4 *>(out + row * d)[i] = a;
  }
}

}  // namespace

void silu_and_mul(torch::Tensor out, torch::Tensor x) {
  const int d = x.size(-1) / 2;
  const long rows = x.numel() / (2 * d);
  TORCH_CHECK(d % 8 == 0);
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous());
  const long total = rows * (d / 8);
  const int block = 256;
  const long grid = std::min<long>((total + block - 1) / block, 2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((silu_and_mul_kernel<__hip_bfloat16>), dim3(grid),
                       dim3(block), 0, stream,
                       (__hip_bfloat16 *)out.data_ptr(),
                       (const __hip_bfloat16 *)x.data_ptr(), rows, d);
  } else if (x.scalar_type() == at::kHalf) {
    hipLaunchKernelGGL((silu_and_mul_kernel<__half>), dim3(grid), dim3(block),
                       0, stream, (__half *)out.data_ptr(),
                       (const __half *)x.data_ptr(), rows, d);
  } else if (x.scalar_type() == at::kFloat) {
    hipLaunchKernelGGL((silu_and_mul_f32_kernel), dim3(grid), dim3(block),
                       0, stream, out.data_ptr<float>(),
                       x.data_ptr<float>(), rows, d);
  } else {
    TORCH_CHECK(false, "silu_and_mul: unsupported dtype");
  }
  HIP_CHECK_KERNEL();
}

void gelu_and_mul(torch::Tensor out, torch::Tensor x) {
  const int d = x.size(-1) / 2;
  const long rows = x.numel() / (2 * d);
  TORCH_CHECK(d % 8 == 0);
  TORCH_CHECK(x.is_contiguous() && out.is_contiguous());
  const long total = rows * (d / 8);
  const int block = 256;
  const long grid = std::min<long>((total + block - 1) / block, 2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((gelu_and_mul_kernel<__hip_bfloat16>), dim3(grid),
                       dim3(block), 0, stream,
                       (__hip_bfloat16 *)out.data_ptr(),
                       (const __hip_bfloat16 *)x.data_ptr(), rows, d);
  } else if (x.scalar_type() == at::kHalf) {
    hipLaunchKernelGGL((gelu_and_mul_kernel<__half>), dim3(grid), dim3(block),
                       0, stream, (__half *)out.data_ptr(),
                       (const __half *)x.data_ptr(), rows, d);
  } else {
    TORCH_CHECK(false, "gelu_and_mul: unsupported dtype");
  }
  HIP_CHECK_KERNEL();
}

void rotary_embedding(torch::Tensor positions, torch::Tensor q,
                      torch::Tensor k, long head_dim,
                      torch::Tensor cos_sin_cache, bool is_neox) {
  const long T = positions.size(0);
  const int rot_dim = cos_sin_cache.size(-1);
  const int hq = q.size(-1) / head_dim, hk = k.size(-1) / head_dim;
  TORCH_CHECK(q.stride(-1) == 1 && k.stride(-1) == 1,
              "rope: innermost dim must be contiguous");
  const long qs = q.stride(0), ks = k.stride(0);
  TORCH_CHECK(q.scalar_type() == at::kBFloat16,
              "rope: bf16 only for now");
  TORCH_CHECK(cos_sin_cache.scalar_type() == at::kFloat);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (is_neox) {
    hipLaunchKernelGGL((rope_kernel<__hip_bfloat16, true>), dim3(T),
                       dim3(256), 0, stream,
                       positions.data_ptr<long>(),
                       (__hip_bfloat16 *)q.data_ptr(),
                       (__hip_bfloat16 *)k.data_ptr(),
                       cos_sin_cache.data_ptr<float>(), rot_dim,
                       (int)head_dim, hq, hk, qs, ks);
  } else {
    hipLaunchKernelGGL((rope_kernel<__hip_bfloat16, false>), dim3(T),
                       dim3(256), 0, stream,
                       positions.data_ptr<long>(),
                       (__hip_bfloat16 *)q.data_ptr(),
                       (__hip_bfloat16 *)k.data_ptr(),
                       cos_sin_cache.data_ptr<float>(), rot_dim,
                       (int)head_dim, hq, hk, qs, ks);
  }
  HIP_CHECK_KERNEL();
}

void reshape_and_cache(torch::Tensor k, torch::Tensor v,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor slot_mapping) {
  const long T = k.size(0);
  if (T == 0) return;
  const int k_row = k.size(1) * k.size(2);
  const int v_row = v.size(1) * v.size(2);
  const int page_size = k_cache.size(1);
  TORCH_CHECK(k_row % 8 == 0 && v_row % 8 == 0);
  TORCH_CHECK(k.stride(-1) == 1 && v.stride(-1) == 1 &&
              k.stride(1) == k.size(2) && v.stride(1) == v.size(2),
              "reshape_and_cache: per-token row must be contiguous");
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  TORCH_CHECK(k_cache.size(2) * k_cache.size(3) == k_row &&
              v_cache.size(2) * v_cache.size(3) == v_row,
              "cache row dims must match inputs");
  TORCH_CHECK(slot_mapping.scalar_type() == at::kLong);
  TORCH_CHECK(k.scalar_type() == at::kBFloat16);
  auto stream = at::cuda::getCurrentCUDAStream();
  const int block = std::min(256, std::max(64, k_row / 8));
  hipLaunchKernelGGL((reshape_and_cache_kernel<__hip_bfloat16>), dim3(T),
                     dim3(block), 0, stream,
                     (const __hip_bfloat16 *)k.data_ptr(),
                     (const __hip_bfloat16 *)v.data_ptr(),
                     (__hip_bfloat16 *)k_cache.data_ptr(),
                     (__hip_bfloat16 *)v_cache.data_ptr(),
                     slot_mapping.data_ptr<long>(), k_row, v_row,
                     page_size, k.stride(0), v.stride(0));
  HIP_CHECK_KERNEL();
}
