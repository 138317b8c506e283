// Varlen causal (chunked-)prefill attention over the paged KV cache,
// MFMA-based, for gfx950.
//
// v2 structure: grid = (q_tiles, B, Hq), block = 256 threads (4 waves).
// Each workgroup computes a 64-row Q tile for one q head; each wave owns
// 16 q rows. K/V tiles of BKV=64 tokens double-buffered in LDS with the
// T14 issue-early/write-late split (guide par.6 G15): tile t+1's global
// loads issue before tile t's MFMAs, the LDS write lands after the
// barrier, so HBM latency hides under compute. K image XOR-swizzled for
// conflict-free ds_read_b128 (T2); V transposed at staging (padded
// stride) so the PV B-fragment reads contiguous kv; P re-enters
// A-fragment layout through a per-wave LDS bounce. s_setprio(1) wraps
// the MFMA clusters (T5). Online softmax per row in registers.
//
// Capability parity: flash_attn_with_kvcache varlen semantics
// (reference layers/attention.py:77-141) -- one kernel serves chunked
// prefill, mixed prefill+decode batches, prefix-cache hits and sliding
// windows.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 mfma_bf8;
typedef __attribute__((ext_vector_type(4))) float mfma_f4;

namespace {

constexpr int BLOCK = 256;
constexpr int BQ = 64;       // q rows per workgroup
constexpr int BKV = 64;      // kv tokens per tile
constexpr int NF = BKV / 16; // score fragments per wave (4)
constexpr int VT_STRIDE = BKV + 8;  // padded kv stride of V^T / P images

template <int D>
DEV_INLINE int kswz(int row, int byte_off) {
  // XOR swizzle within a K row (T2). Mask keeps the offset inside the
  // D*2-byte row (D=128: row&15 conflict-free; D=64: row&7 <=2-way).
  constexpr int MASK = (D * 2 / 16 - 1) & 15;
  return byte_off ^ ((row & MASK) << 4);
}

template <int D>
__global__ __launch_bounds__(BLOCK) void paged_prefill_kernel(
    __hip_bfloat16 *__restrict__ out,            // [T, Hq, D]
    const __hip_bfloat16 *__restrict__ q,        // [T, Hq, D] (row stride)
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

  constexpr int KT = D / 32;        // QK^T k-steps
  constexpr int NT = D / 16;        // PV n-tiles
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;

  // ---- LDS (double-buffered K + V^T; per-wave P bounce) ----
  __shared__ __hip_bfloat16 k_tile[2][BKV * D];
  __shared__ __hip_bfloat16 vt_tile[2][D * VT_STRIDE];
  __shared__ __hip_bfloat16 p_tile[4][16 * VT_STRIDE];

  // ---- Q fragments in registers (A-operand layout) ----
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

  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -INFINITY; l_run[r] = 0.f; }
  mfma_f4 o_acc[NT];
#pragma unroll
  for (int nt = 0; nt < NT; ++nt) o_acc[nt] = mfma_f4{0, 0, 0, 0};

  const int *bt = block_table + (long)b * max_pages;
  const int kv_max = min(seq_len, past + min(q_len, (tile + 1) * BQ));
  int kv_lo = 0;
  if (window > 0)
    kv_lo = max(0, past + tile * BQ - window + 1) / BKV * BKV;

  // ---- staging: per-thread slice of a K/V tile ----------------------
  constexpr int LPT = D / 8;            // 16-B chunks per token row
  constexpr int ROWS_PER_IT = BLOCK / LPT;
  constexpr int N_IT = BKV / ROWS_PER_IT;
  const int s_chunk = tid % LPT;
  const int s_row0 = tid / LPT;

  shortx8 kreg[N_IT], vreg[N_IT];
  auto stage_load = [&](int kv0) {
#pragma unroll
    for (int it = 0; it < N_IT; ++it) {
      const int row = s_row0 + it * ROWS_PER_IT;
      const int tok = kv0 + row;
      if (tok < seq_len) {
        const long crow =
            ((long)bt[tok / page_size] * page_size + tok % page_size);
        const __hip_bfloat16 *kp =
            k_cache + (crow * num_kv_heads + kvh) * D + s_chunk * 8;
        const __hip_bfloat16 *vp =
            v_cache + (crow * num_kv_heads + kvh) * D + s_chunk * 8;
        kreg[it] = *reinterpret_cast<const shortx8 *>(kp);
        vreg[it] = *reinterpret_cast<const shortx8 *>(vp);
      } else {
        kreg[it] = shortx8{0, 0, 0, 0, 0, 0, 0, 0};
        vreg[it] = shortx8{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int it = 0; it < N_IT; ++it) {
      const int row = s_row0 + it * ROWS_PER_IT;
      *reinterpret_cast<shortx8 *>(
          reinterpret_cast<char *>(&k_tile[buf][row * D]) +
          kswz<D>(row, s_chunk * 16)) = kreg[it];
      const __hip_bfloat16 *ve =
          reinterpret_cast<const __hip_bfloat16 *>(&vreg[it]);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        vt_tile[buf][(s_chunk * 8 + j) * VT_STRIDE + row] = ve[j];
    }
  };

  // prologue: stage the first tile
  stage_load(kv_lo);
  stage_write(0);
  __syncthreads();

  int cur = 0;
  for (int kv0 = kv_lo; kv0 < kv_max; kv0 += BKV) {
    const bool has_next = kv0 + BKV < kv_max;
    // T14 issue-early: next tile's global loads start now
    if (has_next) stage_load(kv0 + BKV);

    // ---------- QK^T: S[16 q x BKV] per wave ----------
    // kt OUTER / f INNER: NF independent accumulator chains (PMC: the
    // f-outer form spent 37% of wave cycles issue-stalled on the
    // dependent same-fragment MFMA chain)
    mfma_f4 s_frag[NF];
#pragma unroll
    for (int f = 0; f < NF; ++f) s_frag[f] = mfma_f4{0, 0, 0, 0};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kt = 0; kt < KT; ++kt) {
#pragma unroll
      for (int f = 0; f < NF; ++f) {
        const int krow = f * 16 + l16;
        mfma_bf8 bfrag = *reinterpret_cast<const mfma_bf8 *>(
            reinterpret_cast<char *>(&k_tile[cur][krow * D]) +
            kswz<D>(krow, (kt * 32 + lhi * 8) * 2));
        s_frag[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qfrag[kt], bfrag, s_frag[f], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // ---------- mask + online softmax (rows lhi*4+r per lane) ----------
    float p_vals[NF][4];
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = tile * BQ + wave * 16 + lhi * 4 + r;
      const int qpos = past + qrow;
      float sv[NF];
      float mx = -INFINITY;
#pragma unroll
      for (int f = 0; f < NF; ++f) {
        const int kvp = kv0 + f * 16 + l16;
        float s = s_frag[f][r] * scale;
        if (qrow >= q_len || kvp > qpos || kvp >= seq_len) s = -INFINITY;
        if (window > 0 && kvp <= qpos - window) s = -INFINITY;
        sv[f] = s;
        mx = fmaxf(mx, s);
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, 64));
      const float m_new = fmaxf(m_run[r], mx);
      float psum = 0.f;
#pragma unroll
      for (int f = 0; f < NF; ++f) {
        const float p = (sv[f] == -INFINITY || m_new == -INFINITY)
                            ? 0.f : __expf(sv[f] - m_new);
        p_vals[f][r] = p;
        psum += p;
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        psum += __shfl_xor(psum, off, 64);
      alpha[r] = (m_run[r] == -INFINITY || m_new == -INFINITY)
                     ? 0.f : __expf(m_run[r] - m_new);
      l_run[r] = l_run[r] * alpha[r] + psum;
      m_run[r] = m_new;
    }
#pragma unroll
    for (int nt = 0; nt < NT; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[nt][r] *= alpha[r];

    // ---------- P -> LDS -> A fragments ----------
    __hip_bfloat16 *pw = p_tile[wave];
#pragma unroll
    for (int f = 0; f < NF; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        pw[(lhi * 4 + r) * VT_STRIDE + f * 16 + l16] =
            __float2bfloat16(p_vals[f][r]);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mfma_bf8 pfrag[BKV / 32];
#pragma unroll
    for (int ks = 0; ks < BKV / 32; ++ks)
      pfrag[ks] = *reinterpret_cast<const mfma_bf8 *>(
          &pw[l16 * VT_STRIDE + ks * 32 + lhi * 8]);

    // ---------- PV (ks outer / nt inner: NT independent chains) ------
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < BKV / 32; ++ks) {
#pragma unroll
      for (int nt = 0; nt < NT; ++nt) {
        mfma_bf8 vfrag = *reinterpret_cast<const mfma_bf8 *>(
            &vt_tile[cur][(nt * 16 + l16) * VT_STRIDE + ks * 32 + lhi * 8]);
        o_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pfrag[ks], vfrag, o_acc[nt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // T14 write-late: land the next tile after everyone finished reading
    __syncthreads();
    if (has_next) {
      stage_write(cur ^ 1);
      cur ^= 1;
      __syncthreads();
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
