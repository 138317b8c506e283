// Varlen causal (chunked-)prefill attention over the paged KV cache,
// MFMA-based, for gfx950.
//
// Geometry: grid = (q_tiles, B, Hq), block = 256 threads (4 waves).
// Each workgroup computes a 64-row Q tile for one q head; each wave owns
// 16 q rows. K/V tiles of 32 tokens are staged from the paged cache into
// LDS (K XOR-swizzled for conflict-free ds_read_b128 — guide T2; V
// transposed at staging so the PV B-fragment reads contiguous kv).
// Scores via v_mfma_f32_16x16x32_bf16 (QK^T), online softmax per row in
// registers (C-fragment row groups reduced by 16-lane shfl), P routed
// through LDS to re-enter A-fragment layout, PV via MFMA into fp32
// accumulators.
//
// Capability parity: flash_attn_with_kvcache varlen semantics
// (reference layers/attention.py:77-141) — one kernel serves chunked
// prefill, mixed prefill+decode batches, and prefix-cache hits (context
// tokens already in cache; q covers only the new chunk).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 mfma_bf8;
typedef __attribute__((ext_vector_type(4))) float mfma_f4;

namespace {

constexpr int BLOCK = 256;
constexpr int BQ = 64;     // q rows per workgroup
constexpr int BKV = 32;    // kv tokens per tile
constexpr int VT_STRIDE = 40;  // padded kv stride of the V^T tile (elems)

template <int D>
DEV_INLINE int kswz(int row, int byte_off) {
  // XOR swizzle within a K row: the 16-lane b128 column read hits
  // distinct 16-B slots (T2). Mask keeps the swizzled offset inside the
  // D*2-byte row (D=128: row&15 -> conflict-free; D=64: row&7 -> <=2-way).
  constexpr int MASK = (D * 2 / 16 - 1) & 15;
  return byte_off ^ ((row & MASK) << 4);
}

template <int D>
__global__ __launch_bounds__(BLOCK) void paged_prefill_kernel(
    __hip_bfloat16 *__restrict__ out,            // [T, Hq, D]
    const __hip_bfloat16 *__restrict__ q,        // [T, Hq, D]
    const __hip_bfloat16 *__restrict__ k_cache,  // [P, ps, Hkv, D]
    const __hip_bfloat16 *__restrict__ v_cache,
    const int *__restrict__ block_table,         // [B, max_pages]
    const int *__restrict__ seq_lens,            // [B]
    const int *__restrict__ qsl,                 // [B+1]
    int max_pages, int page_size, int Hq, int num_kv_heads, float scale,
    long q_stride, int window) {
  const int tile = blockIdx.x;
  const int b = blockIdx.y;
  const int h = blockIdx.z;
  const int kvh = h / (Hq / num_kv_heads);

  const int q_start = qsl[b];
  const int q_len = qsl[b + 1] - q_start;
  if (tile * BQ >= q_len) return;
  const int seq_len = seq_lens[b];
  const int past = seq_len - q_len;

  constexpr int KT = D / 32;        // k-steps per QK^T fragment row
  constexpr int NT = D / 16;        // PV n-tiles
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;        // fragment col
  const int lhi = lane >> 4;        // fragment k-group (0..3)

  // ---- LDS ----
  __shared__ __hip_bfloat16 k_tile[BKV * D];              // swizzled rows
  __shared__ __hip_bfloat16 vt_tile[D * VT_STRIDE];       // [d][kv] padded
  __shared__ __hip_bfloat16 p_tile[4][16 * VT_STRIDE];    // per wave [row][kv]

  // ---- Q fragments in registers (A-operand layout) ----
  // lane holds Q[qrow = wave*16 + l16][k = kt*32 + lhi*8 .. +8]
  const int qrow_local = tile * BQ + wave * 16 + l16;
  mfma_bf8 qfrag[KT];
  const bool qrow_valid = qrow_local < q_len;
  {
    const __hip_bfloat16 *qp =
        q + (long)(q_start + qrow_local) * q_stride + h * D + lhi * 8;
#pragma unroll
    for (int kt = 0; kt < KT; ++kt) {
      if (qrow_valid)
        qfrag[kt] = *reinterpret_cast<const mfma_bf8 *>(qp + kt * 32);
      else
        qfrag[kt] = mfma_bf8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  // ---- online softmax state: 4 q rows per lane (rows lhi*4+r) ----
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -INFINITY; l_run[r] = 0.f; }
  mfma_f4 o_acc[NT];
#pragma unroll
  for (int nt = 0; nt < NT; ++nt) o_acc[nt] = mfma_f4{0, 0, 0, 0};

  const int *bt = block_table + (long)b * max_pages;
  // causal upper bound for this q tile
  const int kv_max = min(seq_len, past + min(q_len, (tile + 1) * BQ));
  // sliding window lower bound for the tile's FIRST row
  int kv_lo = 0;
  if (window > 0) {
    const int first_qpos = past + tile * BQ;
    kv_lo = max(0, first_qpos - window + 1) / BKV * BKV;
  }

  for (int kv0 = kv_lo; kv0 < kv_max; kv0 += BKV) {
    const int kv_len = min(BKV, kv_max - kv0);
    // ---------- stage K (swizzled) and V^T ----------
    __syncthreads();
    {
      // 256 threads: row = tid/16 (+16 per iter), chunk = tid%16
      const int chunk = tid & 15;
#pragma unroll
      for (int it = 0; it < BKV / 16; ++it) {
        const int row = it * 16 + (tid >> 4);
        if (row < kv_len) {
          const int tok = kv0 + row;
          const long crow =
              ((long)bt[tok / page_size] * page_size + tok % page_size);
          const __hip_bfloat16 *kp =
              k_cache + (crow * num_kv_heads + kvh) * D + chunk * 8;
          const __hip_bfloat16 *vp =
              v_cache + (crow * num_kv_heads + kvh) * D + chunk * 8;
          if (chunk * 8 < D) {
            shortx8 kv8 = *reinterpret_cast<const shortx8 *>(kp);
            shortx8 vv8 = *reinterpret_cast<const shortx8 *>(vp);
            // K: row-major with XOR swizzle on the byte offset
            *reinterpret_cast<shortx8 *>(
                reinterpret_cast<char *>(&k_tile[row * D]) +
                kswz<D>(row, chunk * 16)) = kv8;
            // V^T: scatter 8 d's
            const __hip_bfloat16 *ve =
                reinterpret_cast<const __hip_bfloat16 *>(&vv8);
#pragma unroll
            for (int j = 0; j < 8; ++j)
              vt_tile[(chunk * 8 + j) * VT_STRIDE + row] = ve[j];
          }
        }
      }
    }
    __syncthreads();

    // ---------- QK^T: S[16 q x 32 kv] per wave ----------
    mfma_f4 s_frag[2];
#pragma unroll
    for (int f = 0; f < 2; ++f) {
      s_frag[f] = mfma_f4{0, 0, 0, 0};
      // B fragment: lane holds K[kv = f*16 + l16][d = kt*32 + lhi*8..+8]
#pragma unroll
      for (int kt = 0; kt < KT; ++kt) {
        const int krow = f * 16 + l16;
        mfma_bf8 bfrag = *reinterpret_cast<const mfma_bf8 *>(
            reinterpret_cast<char *>(&k_tile[krow * D]) +
            kswz<D>(krow, (kt * 32 + lhi * 8) * 2));
        s_frag[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qfrag[kt], bfrag, s_frag[f], 0, 0, 0);
      }
    }

    // ---------- mask + online softmax (rows lhi*4+r per lane) ----------
    float p_vals[2][4];
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = tile * BQ + wave * 16 + lhi * 4 + r;
      const int qpos = past + qrow;
      float s0 = s_frag[0][r] * scale;
      float s1 = s_frag[1][r] * scale;
      const int kvp0 = kv0 + l16, kvp1 = kv0 + 16 + l16;
      if (qrow >= q_len || kvp0 > qpos || kvp0 >= seq_len) s0 = -INFINITY;
      if (qrow >= q_len || kvp1 > qpos || kvp1 >= seq_len) s1 = -INFINITY;
      if (window > 0) {
        if (kvp0 <= qpos - window) s0 = -INFINITY;
        if (kvp1 <= qpos - window) s1 = -INFINITY;
      }
      // row max over the 16 lanes of this row group, both fragments
      float mx = fmaxf(s0, s1);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, 64));
      const float m_new = fmaxf(m_run[r], mx);
      float p0 = 0.f, p1 = 0.f;
      if (m_new != -INFINITY) {
        p0 = (s0 == -INFINITY) ? 0.f : __expf(s0 - m_new);
        p1 = (s1 == -INFINITY) ? 0.f : __expf(s1 - m_new);
      }
      float psum = p0 + p1;
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        psum += __shfl_xor(psum, off, 64);
      alpha[r] = (m_run[r] == -INFINITY || m_new == -INFINITY)
                     ? 0.f : __expf(m_run[r] - m_new);
      l_run[r] = l_run[r] * alpha[r] + psum;
      m_run[r] = m_new;
      p_vals[0][r] = p0;
      p_vals[1][r] = p1;
    }

    // rescale O accumulators: o rows are also lhi*4+r
#pragma unroll
    for (int nt = 0; nt < NT; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[nt][r] *= alpha[r];

    // ---------- P -> LDS -> A-fragment ----------
    __hip_bfloat16 *pw = p_tile[wave];
#pragma unroll
    for (int f = 0; f < 2; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        pw[(lhi * 4 + r) * VT_STRIDE + f * 16 + l16] =
            __float2bfloat16(p_vals[f][r]);
    // wave-local LDS write then read: lgkmcnt ordering within the wave
    // is guaranteed by the compiler's dependency tracking (same address
    // space, same wave; no cross-wave sharing of p_tile[wave]).
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mfma_bf8 pfrag;
    {
      const __hip_bfloat16 *pp = &pw[l16 * VT_STRIDE + lhi * 8];
      pfrag = *reinterpret_cast<const mfma_bf8 *>(pp);
    }

    // ---------- PV ----------
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
      // B fragment: lane holds V^T[d = nt*16 + l16][kv = lhi*8 ..+8]
      mfma_bf8 vfrag = *reinterpret_cast<const mfma_bf8 *>(
          &vt_tile[(nt * 16 + l16) * VT_STRIDE + lhi * 8]);
      o_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          pfrag, vfrag, o_acc[nt], 0, 0, 0);
    }
  }

  // ---------- epilogue ----------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = tile * BQ + wave * 16 + lhi * 4 + r;
    if (qrow >= q_len) continue;
    const float inv_l = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
    __hip_bfloat16 *op = out + ((long)(q_start + qrow) * Hq + h) * D;
#pragma unroll
    for (int nt = 0; nt < NT; ++nt)
      op[nt * 16 + l16] = __float2bfloat16(o_acc[nt][r] * inv_l);
  }
}

}  // namespace

void paged_attention_prefill(torch::Tensor out, torch::Tensor q,
                             torch::Tensor k_cache, torch::Tensor v_cache,
                             torch::Tensor block_table,
                             torch::Tensor seq_lens,
                             torch::Tensor query_start_loc,
                             long max_query_len, double scale,
                             long sliding_window) {
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "prefill attn: bf16 only");
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2),
              "q per-token row must be contiguous");
  const int D = q.size(2);
  const int Hq = q.size(1);
  const int Hkv = k_cache.size(2);
  const int B = seq_lens.size(0);
  const long T = q.size(0);
  if (T == 0) return;
  const int q_tiles = (int)((max_query_len + BQ - 1) / BQ);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (D == 128) {
    hipLaunchKernelGGL((paged_prefill_kernel<128>),
                       dim3(q_tiles, B, Hq), dim3(BLOCK), 0, stream,
                       (__hip_bfloat16 *)out.data_ptr(),
                       (const __hip_bfloat16 *)q.data_ptr(),
                       (const __hip_bfloat16 *)k_cache.data_ptr(),
                       (const __hip_bfloat16 *)v_cache.data_ptr(),
                       block_table.data_ptr<int>(), seq_lens.data_ptr<int>(),
                       query_start_loc.data_ptr<int>(),
                       (int)block_table.size(1), (int)k_cache.size(1), Hq,
                       Hkv, (float)scale, q.stride(0),
                       (int)sliding_window);
  } else if (D == 64) {
    hipLaunchKernelGGL((paged_prefill_kernel<64>),
                       dim3(q_tiles, B, Hq), dim3(BLOCK), 0, stream,
                       (__hip_bfloat16 *)out.data_ptr(),
                       (const __hip_bfloat16 *)q.data_ptr(),
                       (const __hip_bfloat16 *)k_cache.data_ptr(),
                       (const __hip_bfloat16 *)v_cache.data_ptr(),
                       block_table.data_ptr<int>(), seq_lens.data_ptr<int>(),
                       query_start_loc.data_ptr<int>(),
                       (int)block_table.size(1), (int)k_cache.size(1), Hq,
                       Hkv, (float)scale, q.stride(0),
                       (int)sliding_window);
  } else {
    TORCH_CHECK(false, "prefill attn: head_dim ", D, " unsupported");
  }
  HIP_CHECK_KERNEL();
}
