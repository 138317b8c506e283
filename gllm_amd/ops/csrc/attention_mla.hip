// Absorbed-MLA paged attention for gfx950 (DeepSeek V2/V3/V3.2 family).
//
// Semantics parity: the reference's MLA decode backends + chunked-context
// prefill (reference layers/attention.py:366-446,653-925) collapse into
// ONE varlen kernel here. The absorbed latent cache holds a single
// shared row per token [c_kv (512) | k_pe (64)] = 576 dims; attention is
// MQA: every q head attends the same K row, and V is the first 512 dims
// of that row (zero-copy view — each KV byte is read exactly once for
// ALL heads, and the PV B-fragments read straight out of the K tile).
//
// MI355X-native design: a (token, head) pair is a GEMM row — q flattens
// to [T*H, 576] rows tiled 64 per workgroup; the causal mask depends
// only on the row's token, so one kernel serves decode (1 token x 128
// heads = 2 row tiles), chunked prefill, prefix hits and mixed batches.
// Decode split-KVs over blockIdx.z with an fp32 LSE merge.
//
// v2 execution structure (v1 ran 1 wave/SIMD and was latency-bound at
// ~300 GB/s KV): 512 threads = 8 waves at 2 waves/SIMD. Wave w owns
// row group w%4 (16 rows) and V half w/4 (256 dims) — the o
// accumulator halves to 64 VGPRs so two waves co-reside per SIMD; the
// QK^T S-tile is computed twice per row group (MFMA is not the bound).
// K tiles stream through a 3-slot LDS ring via global_load_lds issued
// by waves 0-3 with counted s_waitcnt vmcnt BEFORE the barrier (the
// proven skinny-GEMM pipeline): consumers see landed tiles at barrier
// release without ever waiting a vm counter of their own.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 mfma_bf8;
typedef __attribute__((ext_vector_type(4))) float mfma_f4;

namespace {

constexpr int BLOCK = 512;    // 8 waves
constexpr int BQ = 64;        // flattened (token, head) rows per workgroup
constexpr int BKV = 32;       // kv tokens per ring slot
constexpr int RING = 3;
constexpr int DK = 576;       // latent dims (kv_lora 512 + rope 64)
constexpr int DV = 512;
constexpr int KT = DK / 32;   // QK^T k-steps (18)
constexpr int NTW = 16;       // PV 16-wide n tiles per wave (256 dims)
constexpr int NF = BKV / 16;  // score fragments per wave (2)
constexpr int ROW_B = DK * 2; // K row bytes (1152)
constexpr int TILE_B = BKV * ROW_B;          // 36864
constexpr int GL_PER_WAVE = TILE_B / 1024 / 4;  // 9 (waves 0-3 stage)
constexpr int PT_STRIDE = BKV + 8;

DEV_INLINE int kswz(int row, int byte_off) {
  return byte_off ^ ((row & 7) << 4);
}

template <bool SPLIT>
__global__ __launch_bounds__(BLOCK, 1) void mla_attention_kernel(
    __hip_bfloat16 *__restrict__ out,      // [T, H, DV] (SPLIT: unused)
    float *__restrict__ partial_out,       // [S, TH, DV] (SPLIT only)
    float *__restrict__ partial_lse,       // [S, TH]
    const __hip_bfloat16 *__restrict__ q,  // [T, H, DK] contiguous
    const __hip_bfloat16 *__restrict__ kc, // [P, ps, 1, DK]
    const int *__restrict__ block_table,   // [B, max_pages]
    const int *__restrict__ seq_lens,      // [B]
    const int *__restrict__ qsl,           // [B+1]
    int max_pages, int page_size, int H, float scale, int num_splits,
    long th_total) {
  const int tile = blockIdx.x;
  const int b = blockIdx.y;
  const int split = blockIdx.z;

  const int q_start = qsl[b];
  const int q_len = qsl[b + 1] - q_start;
  const int rows = q_len * H;
  const int tid = threadIdx.x;
  if (tile * BQ >= rows) return;
  const int seq_len = seq_lens[b];
  const int past = seq_len - q_len;

  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;
  const int rg = wave & 3;        // row group (16 rows)
  const int vhalf = wave >> 2;    // V half (256 dims)

  // causal kv bound for this row tile, then the split's share of it
  const int last_tok = min(rows - 1, (tile + 1) * BQ - 1) / H;
  const int kv_hi_all = min(seq_len, past + last_tok + 1);
  const int split_len =
      ((kv_hi_all + BKV - 1) / BKV + num_splits - 1) / num_splits * BKV;
  const int kv_lo = split * split_len;
  const int kv_hi = min(kv_lo + split_len, kv_hi_all);

  const long row_base = (long)q_start * H + (long)tile * BQ;

  if (kv_lo >= kv_hi) {
    if (SPLIT) {
      for (int r = tid; r < BQ; r += BLOCK) {
        if (tile * BQ + r < rows)
          partial_lse[(long)split * th_total + row_base + r] = -INFINITY;
      }
    }
    return;
  }

  __shared__ __attribute__((aligned(16))) char k_ring[RING * TILE_B];
  __shared__ __hip_bfloat16 p_tile[8][16 * PT_STRIDE];

  // ---- Q fragments (A-operand): row = rg*16 + l16 ----
  const int frow = tile * BQ + rg * 16 + l16;
  const bool row_valid = frow < rows;
  mfma_bf8 qfrag[KT];
  {
    const __hip_bfloat16 *qp =
        q + ((long)q_start * H + frow) * DK + lhi * 8;
#pragma unroll
    for (int kt = 0; kt < KT; ++kt) {
      if (row_valid)
        qfrag[kt] = *reinterpret_cast<const mfma_bf8 *>(qp + kt * 32);
      else
        qfrag[kt] = mfma_bf8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -INFINITY; l_run[r] = 0.f; }
  mfma_f4 o_acc[NTW];
#pragma unroll
  for (int nt = 0; nt < NTW; ++nt) o_acc[nt] = mfma_f4{0, 0, 0, 0};

  const int *bt = block_table + (long)b * max_pages;

  // ---- glds staging (waves 0-3): 9 chunks of 1 KiB per wave ----
  auto stage = [&](int kt_idx, int slot) {
    const int kv0 = kv_lo + kt_idx * BKV;
    char *base = k_ring + slot * TILE_B;
#pragma unroll
    for (int j = 0; j < GL_PER_WAVE; ++j) {
      const int p = (wave * GL_PER_WAVE + j) * 1024 + lane * 16;
      const int row = p / ROW_B;
      const int col = kswz(row, p % ROW_B);
      const int tok = kv0 + row;
      // clamp to a valid cache row; masked later by the score mask
      const int ctok = min(tok, seq_len - 1);
      const long crow =
          (long)bt[ctok / page_size] * page_size + ctok % page_size;
      // aux=2 (nt): each latent row is streamed once per split —
      // non-temporal lands faster and leaves L2 to shared data
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int *>(
              reinterpret_cast<const char *>(kc) + crow * ROW_B + col),
          reinterpret_cast<unsigned int *>(
              base + (wave * GL_PER_WAVE + j) * 1024),
          16, 0, 2);
    }
  };

  const int nkt = (kv_hi - kv_lo + BKV - 1) / BKV;
  const int pre = min(RING - 1, nkt);
  if (wave < 4)
    for (int t = 0; t < pre; ++t) stage(t, t % RING);

  for (int kt_i = 0; kt_i < nkt; ++kt_i) {
    const int kv0 = kv_lo + kt_i * BKV;
    const int slot = kt_i % RING;
    if (wave < 4) {
      if (kt_i + RING - 1 < nkt)
        stage(kt_i + RING - 1, (kt_i + RING - 1) % RING);
      const int ahead = min(nkt - 1 - kt_i, RING - 1);
      if (ahead >= 2) {
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2 * GL_PER_WAVE)
                     : "memory");
      } else if (ahead == 1) {
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(1 * GL_PER_WAVE)
                     : "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
    }
    __syncthreads();  // stagers waited first -> tile kt_i is landed

    const char *ktile = k_ring + slot * TILE_B;

    // ---------- QK^T: S[16 rows x 32 tokens] ----------
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
            ktile + krow * ROW_B + kswz(krow, (kt * 32 + lhi * 8) * 2));
        s_frag[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qfrag[kt], bfrag, s_frag[f], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // ---------- mask + online softmax (rows lhi*4+r) ----------
    float p_vals[NF][4];
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int rr = tile * BQ + rg * 16 + lhi * 4 + r;
      const int qpos = past + rr / H;
      float sv[NF];
      float mx = -INFINITY;
#pragma unroll
      for (int f = 0; f < NF; ++f) {
        const int kvp = kv0 + f * 16 + l16;
        float s = s_frag[f][r] * scale;
        if (rr >= rows || kvp > qpos || kvp >= seq_len) s = -INFINITY;
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
    for (int nt = 0; nt < NTW; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[nt][r] *= alpha[r];

    // ---------- P -> LDS bounce -> A fragment ----------
    __hip_bfloat16 *pw = p_tile[wave];
#pragma unroll
    for (int f = 0; f < NF; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        pw[(lhi * 4 + r) * PT_STRIDE + f * 16 + l16] =
            __float2bfloat16(p_vals[f][r]);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mfma_bf8 pfrag = *reinterpret_cast<const mfma_bf8 *>(
        &pw[l16 * PT_STRIDE + lhi * 8]);

    // ---------- PV: o += P[16x32] x V[32 x 256-half] ----------
    // B-fragments gather from the K tile (V = first 512 dims of the
    // row): lane l reads tokens lhi*8+j at dim vhalf*256 + nt*16 + l16
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int nt = 0; nt < NTW; ++nt) {
      const int dim = vhalf * 256 + nt * 16 + l16;
      mfma_bf8 vfrag;
      __hip_bfloat16 *ve = reinterpret_cast<__hip_bfloat16 *>(&vfrag);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int vrow = lhi * 8 + j;
        ve[j] = *reinterpret_cast<const __hip_bfloat16 *>(
            ktile + vrow * ROW_B + kswz(vrow, dim * 2));
      }
      o_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          pfrag, vfrag, o_acc[nt], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();  // everyone done with this ring slot
  }

  // ---------- epilogue ----------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int rr = tile * BQ + rg * 16 + lhi * 4 + r;
    if (rr >= rows) continue;
    const float l = l_run[r];
    const float inv_l = (l > 0.f) ? 1.f / l : 0.f;
    const long orow = row_base - tile * BQ + rr;
    if (SPLIT) {
      float *op = partial_out +
          ((long)split * th_total + orow) * DV + vhalf * 256;
#pragma unroll
      for (int nt = 0; nt < NTW; ++nt)
        op[nt * 16 + l16] = o_acc[nt][r] * inv_l;
      if (l16 == 0 && vhalf == 0)
        partial_lse[(long)split * th_total + orow] =
            (l > 0.f) ? m_run[r] + __logf(l) : -INFINITY;
    } else {
      __hip_bfloat16 *op = out + orow * DV + vhalf * 256;
#pragma unroll
      for (int nt = 0; nt < NTW; ++nt)
        op[nt * 16 + l16] = __float2bfloat16(o_acc[nt][r] * inv_l);
    }
  }
}

// Merge split partials: out[row, :] = sum_s w_s * partial[s, row, :]
__global__ void mla_merge_kernel(__hip_bfloat16 *__restrict__ out,
                                 const float *__restrict__ partial_out,
                                 const float *__restrict__ partial_lse,
                                 int num_splits, long th_total) {
  const long row = blockIdx.x;
  const int d = threadIdx.x;  // 256 threads, 2 dims each
  float mx = -INFINITY;
  for (int s = 0; s < num_splits; ++s)
    mx = fmaxf(mx, partial_lse[(long)s * th_total + row]);
  float denom = 0.f, o0 = 0.f, o1 = 0.f;
  for (int s = 0; s < num_splits; ++s) {
    const float lse = partial_lse[(long)s * th_total + row];
    if (lse == -INFINITY) continue;
    const float w = __expf(lse - mx);
    denom += w;
    const float *p = partial_out + ((long)s * th_total + row) * DV;
    o0 += w * p[d];
    o1 += w * p[d + 256];
  }
  const float inv = (denom > 0.f) ? 1.f / denom : 0.f;
  out[row * DV + d] = __float2bfloat16(o0 * inv);
  out[row * DV + d + 256] = __float2bfloat16(o1 * inv);
}

// Scatter the per-token latent row [T, 1, DK] into the paged cache
// (reference concat_and_cache_mla, cache_kernels.py:161 — the caller
// has already concatenated [c_kv | k_pe]; the v cache is a view of the
// first 512 dims so one write covers both).
__global__ void cache_latent_kernel(const __hip_bfloat16 *__restrict__ k,
                                    __hip_bfloat16 *__restrict__ k_cache,
                                    const long *__restrict__ slot_mapping,
                                    int page_size, long k_stride) {
  const long t = blockIdx.x;
  const long slot = slot_mapping[t];
  const long row = (slot / page_size) * page_size + slot % page_size;
  const shortx8 *src = reinterpret_cast<const shortx8 *>(k + t * k_stride);
  shortx8 *dst = reinterpret_cast<shortx8 *>(k_cache + row * DK);
  for (int i = threadIdx.x; i < DK / 8; i += blockDim.x) dst[i] = src[i];
}

}  // namespace

void mla_paged_attention(torch::Tensor out, torch::Tensor q,
                         torch::Tensor k_cache, torch::Tensor block_table,
                         torch::Tensor seq_lens,
                         torch::Tensor query_start_loc, long max_query_len,
                         double scale, long max_seq_len) {
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "mla attn: bf16 only");
  TORCH_CHECK(q.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(q.size(2) == DK, "mla attn: latent dim must be ", DK);
  TORCH_CHECK(out.size(2) == DV);
  TORCH_CHECK(k_cache.size(2) == 1 && k_cache.size(3) == DK,
              "mla attn: cache must be the shared latent layout");
  TORCH_CHECK(block_table.scalar_type() == at::kInt);
  TORCH_CHECK(seq_lens.scalar_type() == at::kInt);
  const long T = q.size(0);
  if (T == 0) return;
  const int H = q.size(1);
  const int B = seq_lens.size(0);
  const int tiles = (int)((max_query_len * H + BQ - 1) / BQ);
  const long th_total = T * H;
  auto stream = at::cuda::getCurrentCUDAStream();

  // split-KV only for pure-decode batches (uniform kv ranges)
  int splits = 1;
  if (max_query_len == 1) {
    const long kv_tiles = (max_seq_len + BKV - 1) / BKV;
    while (splits < 16 && (long)B * tiles * splits < 640 &&
           splits * 2 <= kv_tiles)
      splits *= 2;
  }

  if (splits == 1) {
    hipLaunchKernelGGL((mla_attention_kernel<false>), dim3(tiles, B, 1),
                       dim3(BLOCK), 0, stream,
                       (__hip_bfloat16 *)out.data_ptr(), nullptr, nullptr,
                       (const __hip_bfloat16 *)q.data_ptr(),
                       (const __hip_bfloat16 *)k_cache.data_ptr(),
                       block_table.data_ptr<int>(), seq_lens.data_ptr<int>(),
                       query_start_loc.data_ptr<int>(),
                       (int)block_table.size(1), (int)k_cache.size(1), H,
                       (float)scale, 1, th_total);
    HIP_CHECK_KERNEL();
    return;
  }
  auto opts = q.options().dtype(at::kFloat);
  auto partial = torch::empty({splits, th_total, (long)DV}, opts);
  auto lse = torch::empty({splits, th_total}, opts);
  hipLaunchKernelGGL((mla_attention_kernel<true>), dim3(tiles, B, splits),
                     dim3(BLOCK), 0, stream, nullptr,
                     partial.data_ptr<float>(), lse.data_ptr<float>(),
                     (const __hip_bfloat16 *)q.data_ptr(),
                     (const __hip_bfloat16 *)k_cache.data_ptr(),
                     block_table.data_ptr<int>(), seq_lens.data_ptr<int>(),
                     query_start_loc.data_ptr<int>(),
                     (int)block_table.size(1), (int)k_cache.size(1), H,
                     (float)scale, splits, th_total);
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL((mla_merge_kernel), dim3((unsigned)th_total),
                     dim3(256), 0, stream,
                     (__hip_bfloat16 *)out.data_ptr(),
                     partial.data_ptr<float>(), lse.data_ptr<float>(),
                     splits, th_total);
  HIP_CHECK_KERNEL();
}

void cache_latent(torch::Tensor k, torch::Tensor k_cache,
                  torch::Tensor slot_mapping) {
  const long T = k.size(0);
  if (T == 0) return;
  TORCH_CHECK(k.scalar_type() == at::kBFloat16);
  TORCH_CHECK(k.stride(-1) == 1);
  TORCH_CHECK(k.numel() / T == DK, "cache_latent: row must be ", DK);
  TORCH_CHECK(k_cache.is_contiguous() && k_cache.size(3) == DK);
  TORCH_CHECK(slot_mapping.scalar_type() == at::kLong);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL((cache_latent_kernel), dim3(T), dim3(64), 0, stream,
                     (const __hip_bfloat16 *)k.data_ptr(),
                     (__hip_bfloat16 *)k_cache.data_ptr(),
                     slot_mapping.data_ptr<long>(), (int)k_cache.size(1),
                     k.stride(0));
  HIP_CHECK_KERNEL();
}
