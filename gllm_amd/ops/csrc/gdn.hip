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


// ------------------------------------------------- chunked WY prefill
// Fused chunk-parallel gated delta rule (WY form) — replaces the torch
// composite in ops/gdn_ref.py::gated_delta_rule_chunked_batched (the
// reference's vendored FLA chunk kernels, layers/ops/fla/chunk*.py):
// ~25 launches per 256-token chunk per layer collapse into ONE launch
// per layer covering the whole padded batch. All math fp32
// (V_MFMA_F32_16X16X4_F32) to track the fp32 oracle.
//
// Grid (B, Hv, 2): each block owns one (sequence, v-head, 64-wide Dv
// half); the state half S [64, 128] lives in LDS across the in-kernel
// chunk loop. C = 64-token chunks; padding rows carry beta = 0, g = 0
// and are exactly inert (U row 0). LDS ~150 KB -> 1 block/CU.
typedef __attribute__((ext_vector_type(4))) float gdn_f4;

__global__ __launch_bounds__(256) void gdn_chunk_prefill_kernel(
    __hip_bfloat16 *__restrict__ o,        // [B, T, Hv, 128]
    const __hip_bfloat16 *__restrict__ q,  // [B, T, Hk, 128]
    const __hip_bfloat16 *__restrict__ k,  // [B, T, Hk, 128]
    const __hip_bfloat16 *__restrict__ v,  // [B, T, Hv, 128]
    const float *__restrict__ g,           // [B, T, Hv]
    const float *__restrict__ beta,        // [B, T, Hv]
    float *__restrict__ states,            // [B, Hv, 128, 128]
    int T, int Hk, int Hv, float scale) {
  constexpr int C = 64, DK = 128, DVH = 64;
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  const int z = blockIdx.z;           // Dv half
  const int hk = h / (Hv / Hk);
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;

  __shared__ float Kt[C][DK + 1];
  __shared__ float Qt[C][DK + 1];
  __shared__ float Vt[C][DVH + 1];
  __shared__ float S[DVH][DK + 2];
  __shared__ float Am[C][C + 1];      // A, then reused for att
  __shared__ float Um[C][C + 1];
  __shared__ float bb[C], Bd[C], btv[C], wc[C];

  // state half -> LDS
  float *sg = states + (((long)b * Hv + h) * 128 + z * DVH) * DK;
  for (int i = tid; i < DVH * DK; i += 256)
    S[i / DK][i % DK] = sg[i];
  __syncthreads();

  const long qk_tok = (long)Hk * DK;  // q/k token stride
  const long v_tok = (long)Hv * DK;

  for (int tb = 0; tb < T; tb += C) {
    // ---- load + l2norm K and Q (each wave: 16 rows, serial) ----
    for (int r = 0; r < 16; ++r) {
      const int t = wave * 16 + r;
      const long tok = (long)b * T + tb + t;
      const __hip_bfloat16 *kp = k + tok * qk_tok + (long)hk * DK;
      const __hip_bfloat16 *qp = q + tok * qk_tok + (long)hk * DK;
      float k0 = __bfloat162float(kp[lane]);
      float k1 = __bfloat162float(kp[lane + 64]);
      float q0 = __bfloat162float(qp[lane]);
      float q1 = __bfloat162float(qp[lane + 64]);
      float nk = k0 * k0 + k1 * k1, nq = q0 * q0 + q1 * q1;
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) {
        nk += __shfl_xor(nk, off, 64);
        nq += __shfl_xor(nq, off, 64);
      }
      const float rk = rsqrtf(fmaxf(nk, 1e-12f));
      const float rq = rsqrtf(fmaxf(nq, 1e-12f)) * scale;
      Kt[t][lane] = k0 * rk;
      Kt[t][lane + 64] = k1 * rk;
      Qt[t][lane] = q0 * rq;
      Qt[t][lane + 64] = q1 * rq;
    }
    // ---- V half + per-token scalars ----
    for (int i = tid; i < C * DVH; i += 256) {
      const int t = i / DVH;
      const long tok = (long)b * T + tb + t;
      Vt[t][i % DVH] = __bfloat162float(
          v[tok * v_tok + (long)h * DK + z * DVH + i % DVH]);
    }
    if (wave == 0) {  // inclusive scan of g over the chunk
      float gv = g[((long)b * T + tb + lane) * Hv + h];
      float bv = beta[((long)b * T + tb + lane) * Hv + h];
#pragma unroll
      for (int off = 1; off < 64; off <<= 1) {
        const float prev = __shfl_up(gv, off, 64);
        if (lane >= off) gv += prev;
      }
      bb[lane] = gv;
      Bd[lane] = __expf(gv);
      btv[lane] = bv;
    }
    __syncthreads();
    const float bb63 = bb[C - 1];

    // ---- A = tril_strict(bt * exp(bb_t - bb_i) * K K^T) ----
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      gdn_f4 acc = {0, 0, 0, 0};
      for (int ks = 0; ks < DK / 4; ++ks)
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(
            Kt[wave * 16 + l16][ks * 4 + lhi],
            Kt[it * 16 + l16][ks * 4 + lhi], acc, 0, 0, 0);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int t = wave * 16 + lhi * 4 + r;
        const int i = it * 16 + l16;
        Am[t][i] = (i < t)
                       ? btv[t] * __expf(bb[t] - bb[i]) * acc[r]
                       : 0.f;
      }
    }
    // ---- U0 = bt * (V - Bd * K S^T) ----
#pragma unroll
    for (int vt = 0; vt < 4; ++vt) {
      gdn_f4 acc = {0, 0, 0, 0};
      for (int ks = 0; ks < DK / 4; ++ks)
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(
            Kt[wave * 16 + l16][ks * 4 + lhi],
            S[vt * 16 + l16][ks * 4 + lhi], acc, 0, 0, 0);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int t = wave * 16 + lhi * 4 + r;
        const int vv = vt * 16 + l16;
        Um[t][vv] = btv[t] * (Vt[t][vv] - Bd[t] * acc[r]);
      }
    }
    __syncthreads();

    // ---- blocked unit-lower solve (I + A) U = U0, in place ----
    for (int R = 0; R < 4; ++R) {
      if (R > 0) {
        // U[Rrows] -= A[Rrows, :R*16] @ U[:R*16]  (wave = v-tile)
        gdn_f4 acc = {0, 0, 0, 0};
        for (int ks = 0; ks < R * 4; ++ks)
          acc = __builtin_amdgcn_mfma_f32_16x16x4f32(
              Am[R * 16 + l16][ks * 4 + lhi],
              Um[ks * 4 + lhi][wave * 16 + l16], acc, 0, 0, 0);
        __syncthreads();
#pragma unroll
        for (int r = 0; r < 4; ++r)
          Um[R * 16 + lhi * 4 + r][wave * 16 + l16] -= acc[r];
      }
      __syncthreads();
      if (wave == 0) {  // in-block forward substitution, 15 steps
        for (int tt = 1; tt < 16; ++tt) {
          const int t = R * 16 + tt;
          float f = 0.f;
          for (int j = R * 16; j < t; ++j)
            f += Am[t][j] * Um[j][lane];
          Um[t][lane] -= f;
        }
      }
      __syncthreads();
    }

    // ---- att = tril(exp(bb_t - bb_i) * Q K^T) (reuse Am) ----
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      gdn_f4 acc = {0, 0, 0, 0};
      for (int ks = 0; ks < DK / 4; ++ks)
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(
            Qt[wave * 16 + l16][ks * 4 + lhi],
            Kt[it * 16 + l16][ks * 4 + lhi], acc, 0, 0, 0);
      __syncthreads();  // everyone past the solve's Am reads
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int t = wave * 16 + lhi * 4 + r;
        const int i = it * 16 + l16;
        Am[t][i] = (i <= t) ? __expf(bb[t] - bb[i]) * acc[r] : 0.f;
      }
    }
    __syncthreads();

    // ---- O = Bd * (Q S^T) + att @ U; write bf16 ----
#pragma unroll
    for (int vt = 0; vt < 4; ++vt) {
      gdn_f4 acc = {0, 0, 0, 0};
      for (int ks = 0; ks < DK / 4; ++ks)
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(
            Qt[wave * 16 + l16][ks * 4 + lhi],
            S[vt * 16 + l16][ks * 4 + lhi], acc, 0, 0, 0);
      gdn_f4 acc2 = {0, 0, 0, 0};
      for (int ks = 0; ks < C / 4; ++ks)
        acc2 = __builtin_amdgcn_mfma_f32_16x16x4f32(
            Am[wave * 16 + l16][ks * 4 + lhi],
            Um[ks * 4 + lhi][vt * 16 + l16], acc2, 0, 0, 0);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int t = wave * 16 + lhi * 4 + r;
        const int vv = vt * 16 + l16;
        const long tok = (long)b * T + tb + t;
        o[tok * v_tok + (long)h * DK + z * DVH + vv] =
            __float2bfloat16(Bd[t] * acc[r] + acc2[r]);
      }
    }
    // ---- state update: S = Bd[63]*S + (wc*U)^T @ K ----
    if (wave == 0) wc[lane] = __expf(bb63 - bb[lane]);
    __syncthreads();
    for (int i = tid; i < DVH * DK; i += 256)
      S[i / DK][i % DK] *= Bd[C - 1];
    __syncthreads();
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      gdn_f4 acc = {0, 0, 0, 0};
      for (int ks = 0; ks < C / 4; ++ks) {
        const int t = ks * 4 + lhi;
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(
            wc[t] * Um[t][wave * 16 + l16],
            Kt[t][dt * 16 + l16], acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r)
        S[wave * 16 + lhi * 4 + r][dt * 16 + l16] += acc[r];
    }
    __syncthreads();
  }

  for (int i = tid; i < DVH * DK; i += 256) sg[i] = S[i / DK][i % DK];
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

void gdn_chunk_prefill(torch::Tensor o, torch::Tensor q, torch::Tensor k,
                       torch::Tensor v, torch::Tensor g, torch::Tensor beta,
                       torch::Tensor states, double scale) {
  const int B = q.size(0), T = q.size(1), Hk = q.size(2);
  const int Hv = v.size(2);
  TORCH_CHECK(q.size(3) == 128 && v.size(3) == 128,
              "gdn prefill: Dk = Dv = 128");
  TORCH_CHECK(T % 64 == 0, "gdn prefill: T padded to 64");
  TORCH_CHECK(Hv % Hk == 0);
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  TORCH_CHECK(g.is_contiguous() && beta.is_contiguous());
  TORCH_CHECK(states.is_contiguous() &&
              states.scalar_type() == at::kFloat);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(gdn_chunk_prefill_kernel, dim3(B, Hv, 2), dim3(256),
                     0, stream, (__hip_bfloat16 *)o.data_ptr(),
                     (const __hip_bfloat16 *)q.data_ptr(),
                     (const __hip_bfloat16 *)k.data_ptr(),
                     (const __hip_bfloat16 *)v.data_ptr(),
                     g.data_ptr<float>(), beta.data_ptr<float>(),
                     states.data_ptr<float>(), T, Hk, Hv, (float)scale);
  HIP_CHECK_KERNEL();
}
