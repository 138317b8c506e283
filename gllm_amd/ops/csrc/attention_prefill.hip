// Varlen causal (chunked-)prefill attention over the paged KV cache,
// MFMA-based, for gfx950.
//
// v3 structure (v1/v2 ran 4 waves at 80 TF/s — PMC showed the time
// split across issue-stalls and VALU/LDS staging overhead, not MFMA):
// 512 threads = 8 waves at 2 waves/SIMD, BQ = 128 q rows per workgroup
// (16 per wave), K/V tiles of BKV=64 streamed through a 2-slot
// global_load_lds ring with counted s_waitcnt vmcnt (the proven
// skinny/MLA pipeline — no register staging instructions, loads fly
// under the previous tile's MFMAs). K staged XOR-swizzled via
// source-address swizzle; V staged linear and PV B-fragments gather
// straight from it (8 ds_read_b16 per fragment — replaces the v2
// transpose staging). Fully-visible interior tiles skip the causal
// mask lane work entirely.
//
// Capability parity: flash_attn_with_kvcache varlen semantics
// (reference layers/attention.py:77-141) — one kernel serves chunked
// prefill, mixed prefill+decode batches, prefix-cache hits and sliding
// windows.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 mfma_bf8;
typedef __attribute__((ext_vector_type(4))) float mfma_f4;

namespace {

constexpr int BLOCK = 512;   // 8 waves
constexpr int BQ = 128;      // q rows per workgroup (16 per wave)
constexpr int BKV = 64;      // kv tokens per ring slot
constexpr int RING = 2;
constexpr int NF = BKV / 16; // score fragments per wave (4)
constexpr int PT_STRIDE = BKV + 8;

template <int D>
DEV_INLINE int kswz(int row, int byte_off) {
  constexpr int MASK = (D * 2 / 16 - 1) & 15;
  return byte_off ^ ((row & MASK) << 4);
}

template <int D>
__global__ __launch_bounds__(BLOCK, 1) void paged_prefill_kernel(
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
  constexpr int KROW_B = D * 2;     // K/V row bytes
  constexpr int TILE_B = BKV * KROW_B;          // one K or V tile
  constexpr int GL_PER_WAVE = 2 * TILE_B / 1024 / 8;  // K+V chunks/wave
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;

  // ---- LDS ring: [K tile | V tile] per slot; per-wave P bounce ----
  __shared__ __attribute__((aligned(16))) char kv_ring[RING * 2 * TILE_B];
  __shared__ __hip_bfloat16 p_tile[8][16 * PT_STRIDE];

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

  // ---- glds staging: every wave stages GL_PER_WAVE 1-KiB chunks ----
  auto stage = [&](int kv0, int slot) {
    char *base = kv_ring + slot * 2 * TILE_B;
#pragma unroll
    for (int j = 0; j < GL_PER_WAVE; ++j) {
      const int p = (wave * GL_PER_WAVE + j) * 1024 + lane * 16;
      const bool is_k = p < TILE_B;
      const int pp = is_k ? p : p - TILE_B;
      const int row = pp / KROW_B;
      const int tok = min(kv0 + row, seq_len - 1);
      const long crow =
          (long)bt[tok / page_size] * page_size + tok % page_size;
      const __hip_bfloat16 *src = is_k ? k_cache : v_cache;
      const int col = is_k ? kswz<D>(row, pp % KROW_B) : pp % KROW_B;
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int *>(
              reinterpret_cast<const char *>(
                  src + (crow * num_kv_heads + kvh) * D) + col),
          reinterpret_cast<unsigned int *>(base + p), 16, 0, 0);
    }
  };

  const int nkt = (kv_max - kv_lo + BKV - 1) / BKV;
  const int pre = min(RING - 1, nkt);
  for (int t = 0; t < pre; ++t) stage(kv_lo + t * BKV, t % RING);

  for (int kt_i = 0; kt_i < nkt; ++kt_i) {
    const int kv0 = kv_lo + kt_i * BKV;
    const int slot = kt_i % RING;
    if (kt_i + RING - 1 < nkt)
      stage(kv_lo + (kt_i + RING - 1) * BKV, (kt_i + RING - 1) % RING);
    const int ahead = min(nkt - 1 - kt_i, RING - 1);
    if (ahead >= 1) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"n"(GL_PER_WAVE) : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __syncthreads();

    const char *ktile = kv_ring + slot * 2 * TILE_B;
    const __hip_bfloat16 *vtile = reinterpret_cast<const __hip_bfloat16 *>(
        ktile + TILE_B);

    // ---------- QK^T: S[16 q x BKV] (kt outer: NF independent) -------
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
            ktile + krow * KROW_B + kswz<D>(krow, (kt * 32 + lhi * 8) * 2));
        s_frag[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qfrag[kt], bfrag, s_frag[f], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // ---------- mask + online softmax (rows lhi*4+r per lane) --------
    // interior tiles (every key visible to every row of this wave, no
    // window cut) skip the per-element mask lanes entirely
    const int qpos_min = past + tile * BQ + wave * 16;
    const bool full_tile =
        qrow_valid && (tile * BQ + wave * 16 + 15 < q_len) &&
        (kv0 + BKV - 1 <= qpos_min) && (kv0 + BKV <= seq_len) &&
        (window <= 0 || kv0 > qpos_min + 15 - window);
    float p_vals[NF][4];
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = tile * BQ + wave * 16 + lhi * 4 + r;
      const int qpos = past + qrow;
      float sv[NF];
      float mx = -INFINITY;
      if (full_tile) {
#pragma unroll
        for (int f = 0; f < NF; ++f) {
          sv[f] = s_frag[f][r] * scale;
          mx = fmaxf(mx, sv[f]);
        }
      } else {
#pragma unroll
        for (int f = 0; f < NF; ++f) {
          const int kvp = kv0 + f * 16 + l16;
          float s = s_frag[f][r] * scale;
          if (qrow >= q_len || kvp > qpos || kvp >= seq_len)
            s = -INFINITY;
          if (window > 0 && kvp <= qpos - window) s = -INFINITY;
          sv[f] = s;
          mx = fmaxf(mx, s);
        }
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
        pw[(lhi * 4 + r) * PT_STRIDE + f * 16 + l16] =
            __float2bfloat16(p_vals[f][r]);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mfma_bf8 pfrag[BKV / 32];
#pragma unroll
    for (int ks = 0; ks < BKV / 32; ++ks)
      pfrag[ks] = *reinterpret_cast<const mfma_bf8 *>(
          &pw[l16 * PT_STRIDE + ks * 32 + lhi * 8]);

    // ---------- PV: B-fragments straight from the linear V tile ------
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < BKV / 32; ++ks) {
#pragma unroll
      for (int nt = 0; nt < NT; ++nt) {
        const int dim = nt * 16 + l16;
        mfma_bf8 vfrag;
        __hip_bfloat16 *ve = reinterpret_cast<__hip_bfloat16 *>(&vfrag);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          ve[j] = vtile[(ks * 32 + lhi * 8 + j) * D + dim];
        o_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pfrag[ks], vfrag, o_acc[nt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();  // everyone done with this ring slot
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
