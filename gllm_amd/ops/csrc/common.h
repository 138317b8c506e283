// Common helpers for gfx950 (MI355X / CDNA4) kernels.
// Wavefront = 64 lanes; block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE_SIZE 64
#define DEV_INLINE __device__ __forceinline__

// ---- vector types for wide loads (16 B / lane sweet spot, G13) ----
typedef __attribute__((ext_vector_type(4))) float floatx4;
typedef __attribute__((ext_vector_type(2))) float floatx2;
typedef __attribute__((ext_vector_type(8))) short shortx8;   // 8 x bf16/f16
typedef __attribute__((ext_vector_type(4))) short shortx4;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;

// ---- scalar dtype conversion ----
template <typename T> struct ScalarOps;

template <> struct ScalarOps<__hip_bfloat16> {
  static DEV_INLINE float to_f32(__hip_bfloat16 v) { return __bfloat162float(v); }
  static DEV_INLINE __hip_bfloat16 from_f32(float v) { return __float2bfloat16(v); }
};
template <> struct ScalarOps<__half> {
  static DEV_INLINE float to_f32(__half v) { return __half2float(v); }
  static DEV_INLINE __half from_f32(float v) { return __float2half(v); }
};
template <> struct ScalarOps<float> {
  static DEV_INLINE float to_f32(float v) { return v; }
  static DEV_INLINE float from_f32(float v) { return v; }
};

// unpack 8 packed 16-bit elements to float
template <typename T>
DEV_INLINE void unpack8(const shortx8 &p, float *out) {
  const T *e = reinterpret_cast<const T *>(&p);
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = ScalarOps<T>::to_f32(e[i]);
}

template <typename T>
DEV_INLINE shortx8 pack8(const float *in) {
  shortx8 p;
  T *e = reinterpret_cast<T *>(&p);
#pragma unroll
  for (int i = 0; i < 8; ++i) e[i] = ScalarOps<T>::from_f32(in[i]);
  return p;
}

// ---- wave reductions (64-wide) ----
DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

DEV_INLINE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// reduce within groups of N lanes (N power of two <= 64)
template <int N>
DEV_INLINE float group_reduce_sum(float v) {
#pragma unroll
  for (int off = N / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// block reduction via LDS (blockDim.x threads, <= 1024)
template <int BLOCK>
DEV_INLINE float block_reduce_sum(float v, float *lds_scratch) {
  constexpr int NWAVES = BLOCK / WAVE_SIZE;
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) lds_scratch[wave] = v;
  __syncthreads();
  if (wave == 0) {
    float x = (lane < NWAVES) ? lds_scratch[lane] : 0.f;
    x = wave_reduce_sum(x);
    if (lane == 0) lds_scratch[0] = x;
  }
  __syncthreads();
  return lds_scratch[0];
}

#define HIP_CHECK_KERNEL()                                                 \
  do {                                                                     \
    hipError_t e = hipGetLastError();                                      \
    if (e != hipSuccess)                                                   \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e)); \
  } while (0)
