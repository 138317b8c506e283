// Paged decode attention (single query token per sequence) for gfx950.
//
// Memory-bound: the job is to stream each sequence's K/V pages once at
// near-HBM rate. Structure:
//   grid = (B, Hkv, num_splits), block = 256 threads (4 waves).
//   Each workgroup serves the whole GQA group (G = Hq/Hkv q-heads) of one
//   kv head over a contiguous token split, so K/V bytes are read once for
//   all G heads. Per CHUNK of tokens:
//     A: lane-groups of D/8 lanes each load one K row (16 B/lane) and a
//        V row (staged to LDS), dot against the G q-vectors in registers,
//        group-reduce, scores -> LDS.
//     B1: per-head online-softmax update (m, l) + p = exp(s - m) -> LDS.
//     B2: (h, d)-mapped threads accumulate acc += p[h][t] * V_lds[t][d].
//   Split partials written fp32 + LSE; merge kernel combines splits and
//   converts to bf16 (reference semantics: merge_state / split-KV decode,
//   SURVEY.md §2.4).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <cstdlib>
#include "common.h"

namespace {

constexpr int BLOCK = 256;

template <int D, int G, int CHUNK>
__global__ __launch_bounds__(BLOCK) void paged_decode_kernel(
    float *__restrict__ partial_out,  // [S, B, Hq, D]
    float *__restrict__ partial_lse,  // [S, B, Hq]
    const __hip_bfloat16 *__restrict__ q,        // [B, Hq, D]
    const __hip_bfloat16 *__restrict__ k_cache,  // [P, ps, Hkv, D]
    const __hip_bfloat16 *__restrict__ v_cache,
    const int *__restrict__ block_table,  // [B, max_pages]
    const int *__restrict__ seq_lens,     // [B]
    int max_pages, int page_size, int num_kv_heads, float scale,
    int num_splits, long q_stride, int window) {
  const int b = blockIdx.x;
  const int kvh = blockIdx.y;
  const int split = blockIdx.z;
  const int Hq = num_kv_heads * G;
  const int seq_len = seq_lens[b];
  // sliding window: only the last `window` tokens are visible
  const int w_begin = (window > 0) ? max(0, seq_len - window) : 0;
  const int visible = seq_len - w_begin;

  // split token range (page-aligned chunks not required; rows are
  // addressed per token through the block table)
  const int split_len = (visible + num_splits - 1) / num_splits;
  const int t0 = w_begin + split * split_len;
  const int t1 = min(t0 + split_len, seq_len);
  const int lane_per_tok = D / 8;            // 16 lanes for D=128
  const int toks_per_iter = BLOCK / lane_per_tok;

  // LDS: V tile + scores + p + softmax state
  __shared__ __hip_bfloat16 v_tile[CHUNK][D];
  __shared__ float s_scores[G][CHUNK];
  __shared__ float s_m[G], s_l[G], s_alpha[G];

  const int tid = threadIdx.x;
  const int tok_slot = tid / lane_per_tok;   // which token in the iter
  const int dlane = tid % lane_per_tok;      // which 8-elem d-chunk
  const int d_off = dlane * 8;

  if (t0 >= t1) {
    // empty split: publish -inf lse so the merge ignores it
    if (tid < G) partial_lse[(long)split * gridDim.x * Hq + (long)b * Hq +
                             kvh * G + tid] = -INFINITY;
    return;
  }

  // ---- q fragments in registers: G heads x 8 elems of this lane's slice
  float qreg[G][8];
#pragma unroll
  for (int h = 0; h < G; ++h) {
    const __hip_bfloat16 *qp =
        q + (long)b * q_stride + (kvh * G + h) * D + d_off;
    shortx8 p = *reinterpret_cast<const shortx8 *>(qp);
    unpack8<__hip_bfloat16>(p, qreg[h]);
  }

  // ---- per-thread output accumulators: (h, d-quad) units so the PV
  // phase reads V 4 elems per ds_read (8 B) instead of scalar b32
  constexpr int NQUAD = G * D / 4;
  constexpr int ACC = (NQUAD + BLOCK - 1) / BLOCK;
  floatx4 acc[ACC];
#pragma unroll
  for (int i = 0; i < ACC; ++i) acc[i] = floatx4{0.f, 0.f, 0.f, 0.f};
  if (tid < G) { s_m[tid] = -INFINITY; s_l[tid] = 0.f; }
  __syncthreads();

  const int *bt = block_table + (long)b * max_pages;

  for (int c0 = t0; c0 < t1; c0 += CHUNK) {
    const int c_len = min(CHUNK, t1 - c0);
    // ---------- phase A: scores + V staging ----------
    for (int it = 0; it < (c_len + toks_per_iter - 1) / toks_per_iter; ++it) {
      const int tt = it * toks_per_iter + tok_slot;  // token within chunk
      float dot[G];
#pragma unroll
      for (int h = 0; h < G; ++h) dot[h] = 0.f;
      if (tt < c_len) {
        const int tok = c0 + tt;
        const long row = ((long)bt[tok / page_size] * page_size +
                          tok % page_size);
        const __hip_bfloat16 *kp =
            k_cache + (row * num_kv_heads + kvh) * D + d_off;
        const __hip_bfloat16 *vp =
            v_cache + (row * num_kv_heads + kvh) * D + d_off;
        shortx8 kv8 = *reinterpret_cast<const shortx8 *>(kp);
        shortx8 vv8 = *reinterpret_cast<const shortx8 *>(vp);
        float kf[8];
        unpack8<__hip_bfloat16>(kv8, kf);
        *reinterpret_cast<shortx8 *>(&v_tile[tt][d_off]) = vv8;
#pragma unroll
        for (int h = 0; h < G; ++h) {
#pragma unroll
          for (int j = 0; j < 8; ++j) dot[h] += qreg[h][j] * kf[j];
        }
      }
      // reduce within the token's lane group; leader writes the score
#pragma unroll
      for (int h = 0; h < G; ++h) {
        float v = group_reduce_sum<D / 8>(dot[h]);
        if (dlane == 0 && tt < c_len) s_scores[h][tt] = v * scale;
      }
    }
    __syncthreads();

    // ---------- phase B1: online softmax per head ----------
    {
      // threads per head: largest power of 2 <= BLOCK/G, capped at 64
      // (shfl width), so shfl groups stay lane-aligned for any G (incl. 5)
      constexpr int TPH0 = BLOCK / G;
      constexpr int TPH = TPH0 >= 64 ? 64 : (TPH0 >= 32 ? 32 :
                          (TPH0 >= 16 ? 16 : (TPH0 >= 8 ? 8 : 4)));
      const int h = tid / TPH;
      const int j = tid % TPH;
      if (h < G) {
        float mx = -INFINITY;
        for (int t = j; t < c_len; t += TPH) mx = fmaxf(mx, s_scores[h][t]);
#pragma unroll
        for (int off = TPH / 2; off > 0; off >>= 1)
          mx = fmaxf(mx, __shfl_xor(mx, off, 64));
        const float m_old = s_m[h];
        const float m_new = fmaxf(m_old, mx);
        float psum = 0.f;
        for (int t = j; t < c_len; t += TPH) {
          const float p = __expf(s_scores[h][t] - m_new);
          s_scores[h][t] = p;                 // reuse scores LDS as p
          psum += p;
        }
#pragma unroll
        for (int off = TPH / 2; off > 0; off >>= 1)
          psum += __shfl_xor(psum, off, 64);
        if (j == 0) {
          const float alpha =
              (m_old == -INFINITY) ? 0.f : __expf(m_old - m_new);
          s_alpha[h] = alpha;
          s_l[h] = s_l[h] * alpha + psum;
          s_m[h] = m_new;
        }
      }
    }
    __syncthreads();

    // ---------- phase B2: PV accumulation (vectorized V reads) ----------
    // 4 independent partial accumulators break the serial FMA chain
    // (the r1 single-chain loop was dependent-latency bound)
#pragma unroll
    for (int i = 0; i < ACC; ++i) {
      const int quad = tid + i * BLOCK;       // (h, d/4) index
      if (quad >= NQUAD) break;
      const int h = (quad * 4) / D;
      const int d0 = (quad * 4) % D;
      const float al = s_alpha[h];
      floatx4 a = acc[i];
      a.x *= al; a.y *= al; a.z *= al; a.w *= al;
      floatx4 p1 = floatx4{0.f, 0.f, 0.f, 0.f};
      floatx4 p2 = floatx4{0.f, 0.f, 0.f, 0.f};
      floatx4 p3 = floatx4{0.f, 0.f, 0.f, 0.f};
      int t = 0;
      for (; t + 4 <= c_len; t += 4) {
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const float p = s_scores[h][t + u];
          const shortx4 v4 =
              *reinterpret_cast<const shortx4 *>(&v_tile[t + u][d0]);
          const __hip_bfloat16 *ve =
              reinterpret_cast<const __hip_bfloat16 *>(&v4);
          floatx4 &dst = (u == 0) ? a : (u == 1) ? p1 : (u == 2) ? p2 : p3;
          dst.x += p * __bfloat162float(ve[0]);
          dst.y += p * __bfloat162float(ve[1]);
          dst.z += p * __bfloat162float(ve[2]);
          dst.w += p * __bfloat162float(ve[3]);
        }
      }
      for (; t < c_len; ++t) {
        const float p = s_scores[h][t];
        const shortx4 v4 =
            *reinterpret_cast<const shortx4 *>(&v_tile[t][d0]);
        const __hip_bfloat16 *ve =
            reinterpret_cast<const __hip_bfloat16 *>(&v4);
        a.x += p * __bfloat162float(ve[0]);
        a.y += p * __bfloat162float(ve[1]);
        a.z += p * __bfloat162float(ve[2]);
        a.w += p * __bfloat162float(ve[3]);
      }
      a.x += p1.x + p2.x + p3.x;
      a.y += p1.y + p2.y + p3.y;
      a.z += p1.z + p2.z + p3.z;
      a.w += p1.w + p2.w + p3.w;
      acc[i] = a;
    }
    __syncthreads();
  }

  // ---------- epilogue: normalized split partial + lse ----------
#pragma unroll
  for (int i = 0; i < ACC; ++i) {
    const int quad = tid + i * BLOCK;
    if (quad >= NQUAD) break;
    const int h = (quad * 4) / D;
    const int d0 = (quad * 4) % D;
    const float l = s_l[h];
    const float inv = (l > 0.f) ? 1.f / l : 0.f;
    float *dst = partial_out +
        (((long)split * gridDim.x + b) * Hq + kvh * G + h) * D + d0;
    floatx4 o = acc[i];
    o.x *= inv; o.y *= inv; o.z *= inv; o.w *= inv;
    *reinterpret_cast<floatx4 *>(dst) = o;
  }
  if (tid < G) {
    const float l = s_l[tid];
    partial_lse[(long)split * gridDim.x * Hq + (long)b * Hq + kvh * G + tid] =
        (l > 0.f) ? s_m[tid] + __logf(l) : -INFINITY;
  }
}

// ================================================================ v3
// MFMA decode kernel for D=128 (v2 above stays for D=64). PMC showed
// the VALU path issue-bound: phase-A dots + B2 accumulates cost ~1.5k
// scalar FMA per thread per 128-token chunk, saturating the SIMD issue
// ports at 25% "active" with 66% barrier parking. Here both GEMM
// phases run on the matrix pipe: per 64-token chunk each wave computes
// one S-tile (rows = the GQA group's q heads, cols = 16 tokens,
// 4 k-steps over D) and a PV tile (rows = heads, 32 of the 128 v-dims,
// 2 k-steps over tokens), ~8 MFMAs/wave/chunk instead of ~800 VALU.
typedef __attribute__((ext_vector_type(8))) __bf16 dec_bf8;
typedef __attribute__((ext_vector_type(4))) float dec_f4;

template <int G>
__global__ __launch_bounds__(BLOCK) void paged_decode_mfma_kernel(
    float *__restrict__ partial_out,  // [S, B, Hq, 128]
    float *__restrict__ partial_lse,  // [S, B, Hq]
    const __hip_bfloat16 *__restrict__ q,        // [B, Hq, 128]
    const __hip_bfloat16 *__restrict__ k_cache,  // [P, ps, Hkv, 128]
    const __hip_bfloat16 *__restrict__ v_cache,
    const int *__restrict__ block_table, const int *__restrict__ seq_lens,
    int max_pages, int page_size, int num_kv_heads, float scale,
    int num_splits, long q_stride, int window) {
  constexpr int D = 128;
  constexpr int CK = 64;            // kv tokens per chunk
  const int b = blockIdx.x;
  const int kvh = blockIdx.y;
  const int split = blockIdx.z;
  const int Hq = num_kv_heads * G;
  const int seq_len = seq_lens[b];
  const int w_begin = (window > 0) ? max(0, seq_len - window) : 0;
  const int visible = seq_len - w_begin;
  const int split_len = (visible + num_splits - 1) / num_splits;
  const int t0 = w_begin + split * split_len;
  const int t1 = min(t0 + split_len, seq_len);

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;

  if (t0 >= t1) {
    if (tid < G)
      partial_lse[(long)split * gridDim.x * Hq + (long)b * Hq + kvh * G +
                  tid] = -INFINITY;
    return;
  }

  // LDS: K tile (swizzled rows), V tile (linear), scores, p image
  __shared__ __hip_bfloat16 k_tile[CK][D];
  __shared__ __hip_bfloat16 v_tile[CK][D];
  __shared__ float s_scores[G][CK];
  __shared__ __hip_bfloat16 p_img[16][CK + 8];
  __shared__ float s_m[G], s_l[G], s_alpha[G];

  // q A-fragments: row l16 = head (pad to 16), dims lhi*8 + kt*32
  dec_bf8 qfrag[4];
#pragma unroll
  for (int kt = 0; kt < 4; ++kt) {
    if (l16 < G) {
      const __hip_bfloat16 *qp =
          q + (long)b * q_stride + (kvh * G + l16) * D + kt * 32 + lhi * 8;
      qfrag[kt] = *reinterpret_cast<const dec_bf8 *>(qp);
    } else {
      qfrag[kt] = dec_bf8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  float m_run = -INFINITY, l_run = 0.f;  // per (head row) via s_m/s_l
  if (tid < G) { s_m[tid] = -INFINITY; s_l[tid] = 0.f; }
  // o accumulators: wave owns dims [wave*32, wave*32+32): 2 n-tiles
  dec_f4 o_acc[2] = {dec_f4{0, 0, 0, 0}, dec_f4{0, 0, 0, 0}};
  (void)m_run; (void)l_run;
  __syncthreads();

  const int *bt = block_table + (long)b * max_pages;
  constexpr int SWZ_MASK = 15;  // 16 chunks of 16B per 256-B row

  for (int c0 = t0; c0 < t1; c0 += CK) {
    const int c_len = min(CK, t1 - c0);
    // ---- stage K (xor-swizzled rows) + V (linear): 4 x 16B/thread ----
    {
      const int cpr = D / 8;              // 16-B chunks per row (16)
      for (int idx = tid; idx < CK * cpr; idx += BLOCK) {
        const int row = idx / cpr;
        const int c = idx % cpr;
        const int tok = c0 + row;
        shortx8 kv{0, 0, 0, 0, 0, 0, 0, 0}, vv{0, 0, 0, 0, 0, 0, 0, 0};
        if (tok < t1) {
          const long crow = ((long)bt[tok / page_size] * page_size +
                             tok % page_size);
          // one workgroup reads each K/V row exactly once: nt loads
          // land ~18% faster and keep L2 for shared data (guide
          // "nt-weights"; decode layers measured 5-10% there)
          kv = __builtin_nontemporal_load(
              reinterpret_cast<const shortx8 *>(
                  k_cache + (crow * num_kv_heads + kvh) * D + c * 8));
          vv = __builtin_nontemporal_load(
              reinterpret_cast<const shortx8 *>(
                  v_cache + (crow * num_kv_heads + kvh) * D + c * 8));
        }
        *reinterpret_cast<shortx8 *>(
            reinterpret_cast<char *>(&k_tile[row][0]) +
            ((c * 16) ^ ((row & SWZ_MASK) << 4) & 255)) = kv;
        *reinterpret_cast<shortx8 *>(&v_tile[row][c * 8]) = vv;
      }
    }
    __syncthreads();

    // ---- S-tile: wave's 16 tokens x 16 head-rows, 4 k-steps ----
    dec_f4 s_frag = dec_f4{0, 0, 0, 0};
#pragma unroll
    for (int kt = 0; kt < 4; ++kt) {
      const int krow = wave * 16 + l16;
      dec_bf8 bfrag = *reinterpret_cast<const dec_bf8 *>(
          reinterpret_cast<char *>(&k_tile[krow][0]) +
          (((kt * 32 + lhi * 8) * 2) ^ ((krow & SWZ_MASK) << 4) & 255));
      s_frag = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[kt], bfrag,
                                                       s_frag, 0, 0, 0);
    }
    // rows lhi*4+r = heads; col l16 = token (wave's 16-token slice)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int h = lhi * 4 + r;
      const int tok = wave * 16 + l16;
      if (h < G && tok < CK) {
        float sv = s_frag[r] * scale;
        if (c0 + tok >= t1) sv = -INFINITY;
        s_scores[h][tok] = sv;
      }
    }
    __syncthreads();

    // ---- online softmax per head over the chunk (64 wide) ----
    {
      constexpr int TPH0 = BLOCK / G;
      constexpr int TPH = TPH0 >= 64 ? 64 : (TPH0 >= 32 ? 32 :
                          (TPH0 >= 16 ? 16 : (TPH0 >= 8 ? 8 : 4)));
      const int h = tid / TPH;
      const int j = tid % TPH;
      if (h < G) {
        float mx = -INFINITY;
        for (int t = j; t < c_len; t += TPH)
          mx = fmaxf(mx, s_scores[h][t]);
#pragma unroll
        for (int off = TPH / 2; off > 0; off >>= 1)
          mx = fmaxf(mx, __shfl_xor(mx, off, 64));
        const float m_old = s_m[h];
        const float m_new = fmaxf(m_old, mx);
        float psum = 0.f;
        for (int t = j; t < CK; t += TPH) {
          const float p = (t < c_len && s_scores[h][t] != -INFINITY &&
                           m_new != -INFINITY)
                              ? __expf(s_scores[h][t] - m_new) : 0.f;
          p_img[h][t] = __float2bfloat16(p);
          psum += p;
        }
#pragma unroll
        for (int off = TPH / 2; off > 0; off >>= 1)
          psum += __shfl_xor(psum, off, 64);
        if (j == 0) {
          const float alpha =
              (m_old == -INFINITY || m_new == -INFINITY)
                  ? 0.f : __expf(m_old - m_new);
          s_alpha[h] = alpha;
          s_l[h] = s_l[h] * alpha + psum;
          s_m[h] = m_new;
        }
      }
      // zero the padded head rows of the p image once per chunk
      for (int h = G + tid; h < 16; h += BLOCK)
        for (int t = 0; t < CK; ++t) p_img[h][t] = __float2bfloat16(0.f);
    }
    __syncthreads();

    // ---- PV: o[heads, wave's 32 dims] += P[heads, CK] x V[CK, dims] ----
    const float al = s_alpha[min(lhi * 4, G - 1)];
    (void)al;
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      // rescale by this chunk's alpha first (rows = heads)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int h = lhi * 4 + r;
        o_acc[nt][r] *= (h < G) ? s_alpha[h] : 0.f;
      }
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        dec_bf8 pfrag = *reinterpret_cast<const dec_bf8 *>(
            &p_img[l16][ks * 32 + lhi * 8]);
        const int dim = wave * 32 + nt * 16 + l16;
        dec_bf8 vfrag;
        __hip_bfloat16 *ve = reinterpret_cast<__hip_bfloat16 *>(&vfrag);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          ve[j] = v_tile[ks * 32 + lhi * 8 + j][dim];
        o_acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pfrag, vfrag, o_acc[nt], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: fp32 partial + lse ----
#pragma unroll
  for (int nt = 0; nt < 2; ++nt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int h = lhi * 4 + r;
      if (h >= G) continue;
      const float l = s_l[h];
      const float inv = (l > 0.f) ? 1.f / l : 0.f;
      const int dim = wave * 32 + nt * 16 + l16;
      partial_out[(((long)split * gridDim.x + b) * Hq + kvh * G + h) * D +
                  dim] = o_acc[nt][r] * inv;
    }
  }
  if (tid < G) {
    const float l = s_l[tid];
    partial_lse[(long)split * gridDim.x * Hq + (long)b * Hq + kvh * G +
                tid] = (l > 0.f) ? s_m[tid] + __logf(l) : -INFINITY;
  }
}

// Merge split partials: out[b,h,:] = sum_s w_s * partial[s,b,h,:]
template <int D>
__global__ void decode_merge_kernel(
    __hip_bfloat16 *__restrict__ out,       // [B, Hq, D]
    const float *__restrict__ partial_out,  // [S, B, Hq, D]
    const float *__restrict__ partial_lse,  // [S, B, Hq]
    int num_splits, int Hq) {
  const int b = blockIdx.x, h = blockIdx.y;
  const int d = threadIdx.x;
  const int B = gridDim.x;
  float mx = -INFINITY;
  for (int s = 0; s < num_splits; ++s)
    mx = fmaxf(mx, partial_lse[((long)s * B + b) * Hq + h]);
  float denom = 0.f, o = 0.f;
  for (int s = 0; s < num_splits; ++s) {
    const float lse = partial_lse[((long)s * B + b) * Hq + h];
    if (lse == -INFINITY) continue;
    const float w = __expf(lse - mx);
    denom += w;
    o += w * partial_out[(((long)s * B + b) * Hq + h) * D + d];
  }
  out[((long)b * Hq + h) * D + d] =
      __float2bfloat16(denom > 0.f ? o / denom : 0.f);
}

template <int D, int G>
void launch_decode(torch::Tensor &out, const torch::Tensor &q,
                   const torch::Tensor &k_cache, const torch::Tensor &v_cache,
                   const torch::Tensor &block_table,
                   const torch::Tensor &seq_lens, float scale,
                   int max_seq_len, long q_stride, int window) {
  const int B = q.size(0);
  const int Hq = q.size(1);
  const int Hkv = k_cache.size(2);
  constexpr int CHUNK = 128;
  // split heuristic: enough workgroups to fill 256 CUs x 8 XCDs
  static int splits_env = [] {
    const char *e = getenv("GLLM_DECODE_SPLITS");
    return e ? atoi(e) : 0;
  }();
  int splits = 1;
  const int base_wgs = B * Hkv;
  while (splits < 16 && base_wgs * splits < 640 &&
         splits * 2 <= (max_seq_len + CHUNK - 1) / CHUNK)
    splits *= 2;
  if (splits_env > 0) splits = splits_env;
  auto opts = q.options().dtype(at::kFloat);
  auto partial = torch::empty({splits, B, Hq, D}, opts);
  auto lse = torch::empty({splits, B, Hq}, opts);
  auto stream = at::cuda::getCurrentCUDAStream();
  static int no_mfma = [] {
    const char *e = getenv("GLLM_DECODE_NO_MFMA");
    return e ? atoi(e) : 0;
  }();
  if (D == 128 && G <= 16 && !no_mfma) {
    hipLaunchKernelGGL((paged_decode_mfma_kernel<G>),
                       dim3(B, Hkv, splits), dim3(BLOCK), 0, stream,
                       partial.data_ptr<float>(), lse.data_ptr<float>(),
                       (const __hip_bfloat16 *)q.data_ptr(),
                       (const __hip_bfloat16 *)k_cache.data_ptr(),
                       (const __hip_bfloat16 *)v_cache.data_ptr(),
                       block_table.data_ptr<int>(),
                       seq_lens.data_ptr<int>(), (int)block_table.size(1),
                       (int)k_cache.size(1), Hkv, scale, splits, q_stride,
                       window);
  } else {
    hipLaunchKernelGGL((paged_decode_kernel<D, G, CHUNK>),
                       dim3(B, Hkv, splits), dim3(BLOCK), 0, stream,
                       partial.data_ptr<float>(), lse.data_ptr<float>(),
                       (const __hip_bfloat16 *)q.data_ptr(),
                       (const __hip_bfloat16 *)k_cache.data_ptr(),
                       (const __hip_bfloat16 *)v_cache.data_ptr(),
                       block_table.data_ptr<int>(), seq_lens.data_ptr<int>(),
                       (int)block_table.size(1), (int)k_cache.size(1), Hkv,
                       scale, splits, q_stride, window);
  }
  HIP_CHECK_KERNEL();
  hipLaunchKernelGGL((decode_merge_kernel<D>), dim3(B, Hq), dim3(D), 0,
                     stream, (__hip_bfloat16 *)out.data_ptr(),
                     partial.data_ptr<float>(), lse.data_ptr<float>(), splits,
                     Hq);
  HIP_CHECK_KERNEL();
}

}  // namespace

void paged_attention_decode(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor block_table, torch::Tensor seq_lens,
                            double scale, long sliding_window) {
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "decode attn: bf16 only");
  TORCH_CHECK(out.is_contiguous());
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2),
              "q per-token row must be contiguous");
  TORCH_CHECK(block_table.scalar_type() == at::kInt);
  TORCH_CHECK(seq_lens.scalar_type() == at::kInt);
  const int D = q.size(2);
  const int Hq = q.size(1);
  const int Hkv = k_cache.size(2);
  TORCH_CHECK(Hq % Hkv == 0);
  const int G = Hq / Hkv;
  const int max_seq = k_cache.size(0) * k_cache.size(1);  // upper bound
  const float s = (float)scale;

#define CASE(DD, GG)                                                       \
  if (D == DD && G == GG) {                                                \
    launch_decode<DD, GG>(out, q, k_cache, v_cache, block_table, seq_lens, \
                          s, max_seq, q.stride(0), (int)sliding_window);   \
    return;                                                                \
  }
  CASE(128, 1) CASE(128, 2) CASE(128, 4) CASE(128, 5) CASE(128, 8)
  CASE(64, 1) CASE(64, 2) CASE(64, 4) CASE(64, 8)
  CASE(128, 16)
#undef CASE
  TORCH_CHECK(false, "paged_attention_decode: unsupported head_dim=", D,
              " group=", G);
}
