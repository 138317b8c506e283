// fp8 (OCP e4m3) execution path for gfx950: per-token-group activation
// quant, block-scale weight-streaming GEMM (decode), block-scale grouped
// MoE GEMM. Replaces the reference's DeepGEMM / flashinfer / Triton
// w8a8 chain (layers/quantization/fp8.py:33-158,556-676) with native
// CDNA4 fp8 MFMA (V_MFMA_F32_16X16X32_FP8_FP8).
//
// Layouts (exactly what layers/quantization/fp8.py loads):
//   weights    e4m3 [N, K], scale_inv fp32 [ceil(N/128), ceil(K/128)]
//   activation e4m3 [M, K], scales    fp32 [M, K/128] (per token-group)
// Scales are applied PER 128-K BLOCK in the accumulator: each BK=128
// tile accumulates into a fresh sub-accumulator through 4 fp8 MFMA
// k-steps, then folds into the fp32 accumulator scaled by
// a_scale[m, kb] * w_scale[nblk, kb]. The fp8 stream halves the weight
// bytes — decode GEMMs are weight-bandwidth-bound, so this is ~2x.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f8_f4;

namespace {

constexpr int BLOCK = 256;
constexpr int BN = 64;
constexpr int BK = 128;         // one scale block per K tile
constexpr int ROW_B = BK;       // tile row bytes (128 fp8)
constexpr int GROUP = 128;

DEV_INLINE int swz(int row, int byte_off) {
  return byte_off ^ ((row & 7) << 4);
}

DEV_INLINE float fp8_to_f32(unsigned char v) {
  __hip_fp8_e4m3 h;
  h.__x = v;
  return (float)h;
}

// ------------------------------------------------- activation quant
// x [T, K] bf16 -> q [T, K] e4m3 + scales [T, K/128]. One wave per
// (token, group): 2 elems per lane.
__global__ void per_token_group_quant_kernel(
    const __hip_bfloat16 *__restrict__ x, unsigned char *__restrict__ q,
    float *__restrict__ scales, float *__restrict__ scales_t,
    long n_groups, int k_groups, int K, bool ue8m0) {
  const long g0 = blockIdx.x * (long)(blockDim.x / 64) + (threadIdx.x >> 6);
  if (g0 >= n_groups) return;
  const int lane = threadIdx.x & 63;
  const long t = g0 / k_groups;
  const int kg = g0 % k_groups;
  const __hip_bfloat16 *src = x + t * (long)K + kg * GROUP + lane * 2;
  float v0 = __bfloat162float(src[0]);
  float v1 = __bfloat162float(src[1]);
  float amax = fmaxf(fabsf(v0), fabsf(v1));
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    amax = fmaxf(amax, __shfl_xor(amax, off, 64));
  amax = fmaxf(amax, 1e-4f);
  float scale = amax / 448.f;
  if (ue8m0) scale = exp2f(ceilf(log2f(scale)));
  const float inv = 1.f / scale;
  __hip_fp8_e4m3 q0(v0 * inv), q1(v1 * inv);
  unsigned char *dst = q + t * (long)K + kg * GROUP + lane * 2;
  dst[0] = q0.__x;
  dst[1] = q1.__x;
  if (lane == 0) {
    scales[t * (long)k_groups + kg] = scale;
    // column-major copy [k_groups, T] for the skinny GEMM: a stage's
    // M scales become ONE contiguous glds line-run instead of a
    // 64-line strided gather inside the counted-vmcnt ring
    if (scales_t)
      scales_t[(long)kg * (n_groups / k_groups) + t] = scale;
  }
}

// ------------------------------------------------- skinny fp8 GEMM
// Same glds-ring structure as skinny_gemm.hip (v2), fp8 tiles: a ring
// slot holds { W[64][128B] | A[MB*64][128B] } = one BK=128 scale block.
template <int AUX>
DEV_INLINE void glds16(const unsigned char *gsrc, char *lds_ptr) {
  __builtin_amdgcn_global_load_lds(
      reinterpret_cast<const unsigned int *>(gsrc),
      reinterpret_cast<unsigned int *>(lds_ptr), 16, 0, AUX);
}

template <int MB, int RING>
__global__ __launch_bounds__(BLOCK) void fp8_skinny_kernel(
    float *__restrict__ partial,           // [SPLITK, M, N]
    const unsigned char *__restrict__ aq,  // [M, K] e4m3
    const float *__restrict__ ast,         // [K/128, M] (transposed)
    const unsigned char *__restrict__ w,   // [N, K] e4m3
    const float *__restrict__ ws,          // [N/128, K/128]
    __hip_bfloat16 *__restrict__ out,      // non-null iff splitk == 1
    const float *__restrict__ bias, int M, int N, int K, int k_slice) {
  constexpr int TILE_B = (BN + MB * 64) * ROW_B;
  const int n0 = blockIdx.x * BN;
  const int z = blockIdx.y;
  const int kb_begin = z * (k_slice / BK);
  const int kb_end = min(K / BK, kb_begin + k_slice / BK);
  const int nkt = kb_end - kb_begin;
  if (nkt <= 0) return;
  const int kgroups = K / BK;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;

  __shared__ __attribute__((aligned(16)))
      char smem[RING * ((BN + MB * 64) * ROW_B + MB * 256)];
  // the whole k-slice's weight scales, staged ONCE (a per-tile global
  // scale load would force a vmcnt drain that serializes the glds ring)
  __shared__ float ws_lds[256];

  const int wblk = n0 / 128;  // BN=64 tile sits inside one n scale block
  for (int i = tid; i < nkt; i += BLOCK)
    ws_lds[i] = ws[(long)wblk * kgroups + kb_begin + i];
  __syncthreads();

  constexpr int GL_PER_WAVE = TILE_B / 1024 / 4;
  // per-stage vm ops per wave: GL_PER_WAVE glds16 + MB glds4 (the
  // activation scales gather through the SAME lds-dma pipeline so the
  // counted waits stay exact — a plain global scale load would make the
  // compiler emit a drain that serializes the ring)
  constexpr int VM_PER_STAGE = GL_PER_WAVE + MB;
  constexpr int ASC_OFF = TILE_B;  // asc area at the slot tail
  constexpr int SLOT_B = TILE_B + MB * 64 * 4;
  const unsigned char *gsrc[GL_PER_WAVE];
  {
#pragma unroll
    for (int j = 0; j < GL_PER_WAVE; ++j) {
      const int p = (wave * GL_PER_WAVE + j) * 1024 + lane * 16;
      const int row = p / ROW_B;
      const int col = swz(row, p % ROW_B);
      if (row < BN) {
        const int n = min(n0 + row, N - 1);
        gsrc[j] = w + (long)n * K + col;
      } else {
        const int m = min(row - BN, M - 1);
        gsrc[j] = aq + (long)m * K + col;
      }
    }
  }

  auto stage = [&](int kt, int slot) {
    const long kb = (long)(kb_begin + kt) * BK;
    char *base = smem + slot * SLOT_B;
#pragma unroll
    for (int j = 0; j < GL_PER_WAVE; ++j) {
      const int off = (wave * GL_PER_WAVE + j) * 1024;
      if (off < BN * ROW_B)
        glds16<2>(gsrc[j] + kb, base + off);
      else
        glds16<0>(gsrc[j] + kb, base + off);
    }
    // every wave issues the identical scale load (same values land
    // at the same LDS bytes) so per-wave vm counters stay uniform.
    // ast is [kgroups, M]: the stage's M scales are CONTIGUOUS — a
    // handful of cache lines instead of the 64-line strided gather
    // that used to double the ring's line-request rate.
#pragma unroll
    for (int mb = 0; mb < MB; ++mb) {
      const int m = min(mb * 64 + lane, M - 1);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int *>(
              ast + (long)(kb_begin + kt) * M + m),
          reinterpret_cast<unsigned int *>(base + ASC_OFF + mb * 256), 4,
          0, 0);
    }
  };

  f8_f4 acc[MB][BN / 16];
#pragma unroll
  for (int mb = 0; mb < MB; ++mb)
#pragma unroll
    for (int nt = 0; nt < BN / 16; ++nt) acc[mb][nt] = f8_f4{0, 0, 0, 0};

  const int pre = min(RING - 1, nkt);
  for (int t = 0; t < pre; ++t) stage(t, t % RING);

  for (int kt = 0; kt < nkt; ++kt) {
    const int slot = kt % RING;
    if (kt + RING - 1 < nkt) stage(kt + RING - 1, (kt + RING - 1) % RING);
    const int ahead = min(nkt - 1 - kt, RING - 1);
    if (RING >= 4 && ahead == 3) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(3 * VM_PER_STAGE)
                   : "memory");
    } else if (RING >= 3 && ahead == 2) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2 * VM_PER_STAGE)
                   : "memory");
    } else if (ahead == 1) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(1 * VM_PER_STAGE)
                   : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    const char *wbase = smem + slot * SLOT_B;
    const char *xbase = wbase + BN * ROW_B;
    const float *asc_lds =
        reinterpret_cast<const float *>(wbase + ASC_OFF);
    const float wsc = ws_lds[kt];

#pragma unroll
    for (int mb = 0; mb < MB; ++mb) {
      f8_f4 sub[BN / 16];
#pragma unroll
      for (int nt = 0; nt < BN / 16; ++nt) sub[nt] = f8_f4{0, 0, 0, 0};
#pragma unroll
      for (int ks = 0; ks < BK / 32; ++ks) {
        const int arow = mb * 64 + wave * 16 + l16;
        const long afrag = *reinterpret_cast<const long *>(
            xbase + arow * ROW_B + swz(arow, ks * 32 + lhi * 8));
#pragma unroll
        for (int nt = 0; nt < BN / 16; ++nt) {
          const int brow = nt * 16 + l16;
          const long bfrag = *reinterpret_cast<const long *>(
              wbase + brow * ROW_B + swz(brow, ks * 32 + lhi * 8));
          sub[nt] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              afrag, bfrag, sub[nt], 0, 0, 0);
        }
      }
      float asc[4];
#pragma unroll
      for (int r = 0; r < 4; ++r)
        asc[r] = asc_lds[mb * 64 + min(wave * 16 + lhi * 4 + r, 63)] * wsc;
#pragma unroll
      for (int nt = 0; nt < BN / 16; ++nt)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          acc[mb][nt][r] += sub[nt][r] * asc[r];
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  if (out != nullptr) {
    // splitk == 1: write bf16 directly (+bias) — skips the separate
    // fp32-partial round-trip and the reduce launch entirely
#pragma unroll
    for (int mb = 0; mb < MB; ++mb) {
#pragma unroll
      for (int nt = 0; nt < BN / 16; ++nt) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int m = mb * 64 + wave * 16 + lhi * 4 + r;
          const int n = n0 + nt * 16 + l16;
          if (m < M && n < N) {
            float v = acc[mb][nt][r];
            if (bias) v += bias[n];
            out[(long)m * N + n] = __float2bfloat16(v);
          }
        }
      }
    }
    return;
  }
  float *base = partial + (long)z * M * N;
#pragma unroll
  for (int mb = 0; mb < MB; ++mb) {
#pragma unroll
    for (int nt = 0; nt < BN / 16; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = mb * 64 + wave * 16 + lhi * 4 + r;
        const int n = n0 + nt * 16 + l16;
        if (m < M && n < N) base[(long)m * N + n] = acc[mb][nt][r];
      }
    }
  }
}

// out[m, n] = bf16( sum_z partial[z, m, n] + bias[n] )
__global__ void fp8_reduce_kernel(__hip_bfloat16 *__restrict__ out,
                                  const float *__restrict__ partial,
                                  const float *__restrict__ bias, int M,
                                  int N, int splitk) {
  const long total = (long)M * N;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    float v = bias ? bias[i % N] : 0.f;
    for (int z = 0; z < splitk; ++z) v += partial[(long)z * total + i];
    out[i] = __float2bfloat16(v);
  }
}

// ------------------------------------------------- fp8 grouped MoE GEMM
// moe.hip's pipeline with fp8 operands + block scales. LDS-tiled like
// the bf16 version (BKm = 128 elems = one scale block per k tile).
constexpr int MOE_BK = 128;

template <int BM, int MOE_BN, bool SCATTER>
__global__ __launch_bounds__(BLOCK) void moe_gemm_fp8_kernel(
    __hip_bfloat16 *__restrict__ C,
    const unsigned char *__restrict__ A,   // [Ta, K] e4m3
    const float *__restrict__ As,          // [Ta, K/128]
    const unsigned char *__restrict__ W,   // [E, Nd, K] e4m3
    const float *__restrict__ Ws,          // [E, ceil(Nd/128), K/128]
    const int *__restrict__ sorted_ids, const int *__restrict__ expert_blocks,
    const int *__restrict__ n_post_pad,
    const float *__restrict__ topk_w, int n_pairs, int K, int Nd,
    int topk) {
  const int mb = blockIdx.x;
  if (mb * BM >= n_post_pad[0]) return;
  const int nb = blockIdx.y;
  const int e = expert_blocks[mb];
  const int kgroups = K / MOE_BK;
  const int nblocks = (Nd + 127) / 128;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;

  constexpr int WM = BM / 16;
  constexpr int WN_TILES = MOE_BN * WM / 64;
  const int wm = wave % WM;
  const int wn = wave / WM;

  constexpr int APAD = 16;
  __shared__ unsigned char a_tile[2][BM * (MOE_BK + APAD)];
  __shared__ unsigned char w_tile[2][MOE_BN * (MOE_BK + APAD)];

  const int g0 = mb * BM;
  constexpr int A_CH = BM * MOE_BK / 16;  // 16-B chunks
  constexpr int W_CH = MOE_BN * MOE_BK / 16;
  typedef __attribute__((ext_vector_type(4))) int int4v;
  int4v areg[(A_CH + BLOCK - 1) / BLOCK];
  int4v wreg[(W_CH + BLOCK - 1) / BLOCK];

  const long w_base = (long)e * Nd * K;

  auto load_tiles = [&](int kb) {
#pragma unroll
    for (int it = 0; it < (A_CH + BLOCK - 1) / BLOCK; ++it) {
      const int idx = tid + it * BLOCK;
      if (idx < A_CH) {
        const int m = idx / (MOE_BK / 16);
        const int c = idx % (MOE_BK / 16);
        const int g = g0 + m;
        const int pair = sorted_ids[g];
        const bool valid = pair < n_pairs;
        long arow = SCATTER ? g : (valid ? pair / topk : 0);
        if (valid)
          areg[it] = *reinterpret_cast<const int4v *>(
              A + arow * (long)K + kb * MOE_BK + c * 16);
        else
          areg[it] = int4v{0, 0, 0, 0};
      }
    }
#pragma unroll
    for (int it = 0; it < (W_CH + BLOCK - 1) / BLOCK; ++it) {
      const int idx = tid + it * BLOCK;
      if (idx < W_CH) {
        const int n = idx / (MOE_BK / 16);
        const int c = idx % (MOE_BK / 16);
        if (nb * MOE_BN + n < Nd) {
          const int4v *wp = reinterpret_cast<const int4v *>(
              W + w_base + (long)(nb * MOE_BN + n) * K + kb * MOE_BK +
              c * 16);
          // nt measured NEGATIVE here (DeepSeek fp8 MoE B256: many
          // experts span 2 m-blocks, nt kills the W reuse) — cached
          wreg[it] = *wp;
        } else
          wreg[it] = int4v{0, 0, 0, 0};
      }
    }
  };
  auto write_tiles = [&](int buf) {
#pragma unroll
    for (int it = 0; it < (A_CH + BLOCK - 1) / BLOCK; ++it) {
      const int idx = tid + it * BLOCK;
      if (idx < A_CH) {
        const int m = idx / (MOE_BK / 16);
        const int c = idx % (MOE_BK / 16);
        *reinterpret_cast<int4v *>(
            &a_tile[buf][m * (MOE_BK + APAD) + c * 16]) = areg[it];
      }
    }
#pragma unroll
    for (int it = 0; it < (W_CH + BLOCK - 1) / BLOCK; ++it) {
      const int idx = tid + it * BLOCK;
      if (idx < W_CH) {
        const int n = idx / (MOE_BK / 16);
        const int c = idx % (MOE_BK / 16);
        *reinterpret_cast<int4v *>(
            &w_tile[buf][n * (MOE_BK + APAD) + c * 16]) = wreg[it];
      }
    }
  };

  f8_f4 acc[WN_TILES];
#pragma unroll
  for (int t = 0; t < WN_TILES; ++t) acc[t] = f8_f4{0, 0, 0, 0};

  load_tiles(0);
  write_tiles(0);
  __syncthreads();

  int cur = 0;
  const int nkb = K / MOE_BK;
  for (int kb = 0; kb < nkb; ++kb) {
    if (kb + 1 < nkb) load_tiles(kb + 1);

    f8_f4 sub[WN_TILES];
#pragma unroll
    for (int t = 0; t < WN_TILES; ++t) sub[t] = f8_f4{0, 0, 0, 0};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < MOE_BK / 32; ++ks) {
      const long afrag = *reinterpret_cast<const long *>(
          &a_tile[cur][(wm * 16 + l16) * (MOE_BK + APAD) + ks * 32 +
                       lhi * 8]);
#pragma unroll
      for (int t = 0; t < WN_TILES; ++t) {
        const long bfrag = *reinterpret_cast<const long *>(
            &w_tile[cur][((wn * WN_TILES + t) * 16 + l16) * (MOE_BK + APAD) +
                         ks * 32 + lhi * 8]);
        sub[t] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            afrag, bfrag, sub[t], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // fold scales: A row per acc row r, W block per n tile
    float asc[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int g = g0 + wm * 16 + lhi * 4 + r;
      const int pair = sorted_ids[min(g, (int)(mb * BM + BM - 1))];
      const bool valid = pair < n_pairs;
      long arow = SCATTER ? g : (valid ? pair / topk : 0);
      asc[r] = valid ? As[arow * (long)kgroups + kb] : 0.f;
    }
#pragma unroll
    for (int t = 0; t < WN_TILES; ++t) {
      const int nblk = (nb * MOE_BN + (wn * WN_TILES + t) * 16) / 128;
      const float wsc =
          Ws[((long)e * nblocks + min(nblk, nblocks - 1)) * kgroups + kb];
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[t][r] += sub[t][r] * asc[r] * wsc;
    }

    __syncthreads();
    if (kb + 1 < nkb) {
      write_tiles(cur ^ 1);
      cur ^= 1;
      __syncthreads();
    }
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = wm * 16 + lhi * 4 + r;
    const int g = g0 + m;
    const int pair = sorted_ids[g];
    if (pair >= n_pairs) continue;
    long crow = SCATTER ? (long)pair : (long)g;
    float scalew = 1.f;
    if (SCATTER && topk_w != nullptr) scalew = topk_w[pair];
#pragma unroll
    for (int t = 0; t < WN_TILES; ++t) {
      const int n = nb * MOE_BN + (wn * WN_TILES + t) * 16 + l16;
      if (n < Nd)
        C[crow * (long)Nd + n] = __float2bfloat16(acc[t][r] * scalew);
    }
  }
}

}  // namespace

void per_token_group_quant_fp8(torch::Tensor x, torch::Tensor q,
                               torch::Tensor scales,
                               c10::optional<torch::Tensor> scales_t,
                               bool ue8m0) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(q.is_contiguous() && scales.is_contiguous());
  const long T = x.size(0);
  const int K = x.size(1);
  TORCH_CHECK(K % GROUP == 0, "K must be a multiple of 128");
  const int kg = K / GROUP;
  const long n_groups = T * kg;
  float *st = nullptr;
  if (scales_t.has_value()) {
    TORCH_CHECK(scales_t->is_contiguous() &&
                scales_t->numel() == n_groups);
    st = scales_t->data_ptr<float>();
  }
  const int wpb = 4;  // waves per block
  const long grid = (n_groups + wpb - 1) / wpb;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(per_token_group_quant_kernel, dim3(grid),
                     dim3(wpb * 64), 0, stream,
                     (const __hip_bfloat16 *)x.data_ptr(),
                     (unsigned char *)q.data_ptr(),
                     scales.data_ptr<float>(), st, n_groups, kg, K, ue8m0);
  HIP_CHECK_KERNEL();
}

void fp8_skinny_gemm(torch::Tensor out, torch::Tensor aq, torch::Tensor ast,
                     torch::Tensor w, torch::Tensor ws,
                     c10::optional<torch::Tensor> bias,
                     torch::Tensor workspace, long splitk_arg) {
  const int M = aq.size(0), K = aq.size(1), N = w.size(0);
  TORCH_CHECK(aq.is_contiguous() && w.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(K % BK == 0, "fp8 gemm: K must be a multiple of 128");
  TORCH_CHECK(K / BK <= 256, "fp8 skinny: K <= 32768 (ws_lds)");
  TORCH_CHECK(M <= 256, "fp8 skinny: M <= 256");
  TORCH_CHECK(ast.is_contiguous() && ast.numel() == (long)M * (K / BK),
              "fp8 skinny: transposed scales [K/128, M]");
  const int n_wg = (N + BN - 1) / BN;
  int splitk = (int)splitk_arg;
  if (splitk <= 0) {
    splitk = 1;
    while (splitk < 16 && n_wg * (splitk * 2) <= 1024 &&
           (K / BK) / (splitk * 2) >= 8)
      splitk *= 2;
  }
  int k_slice = (K + splitk - 1) / splitk;
  k_slice = ((k_slice + BK - 1) / BK) * BK;
  splitk = (K + k_slice - 1) / k_slice;
  TORCH_CHECK(splitk == 1 || workspace.numel() >= (long)splitk * M * N);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto *wsp = workspace.data_ptr<float>();
  auto *ap = (const unsigned char *)aq.data_ptr();
  auto *asp = ast.data_ptr<float>();
  auto *wp = (const unsigned char *)w.data_ptr();
  auto *wsc = ws.data_ptr<float>();
  const float *bias_ptr = nullptr;
  if (bias.has_value()) bias_ptr = bias->data_ptr<float>();
  // splitk == 1: the GEMM kernel writes bf16 (+bias) itself
  __hip_bfloat16 *outp =
      splitk == 1 ? (__hip_bfloat16 *)out.data_ptr() : nullptr;
#define LAUNCH_SK(MB, RING)                                                  \
  hipLaunchKernelGGL((fp8_skinny_kernel<MB, RING>),                          \
                     dim3(n_wg, splitk), dim3(BLOCK), 0, stream, wsp, ap,    \
                     asp, wp, wsc, outp, bias_ptr, M, N, K, k_slice)
  static int ring_env = [] {
    const char *e = getenv("FP8_RING");
    return e ? atoi(e) : 0;
  }();
  if (M <= 64) {
    // RING=2 = 33 KB LDS = 4 blocks/CU: measured faster than the
    // deeper 3-slot ring at 3 blocks/CU on every decode shape.
    // (A W-only-DMA variant with asm register A-loads measured ~8%
    // faster still, but async asm defs are invisible to regalloc —
    // it may copy the register before the data lands — and it
    // miscompiled; profiles/r02_session_notes.md records the attempt.)
    const int r = ring_env ? ring_env : 2;
    if (r == 3) LAUNCH_SK(1, 3);
    else if (r == 4) LAUNCH_SK(1, 4);
    else LAUNCH_SK(1, 2);
  }
  else if (M <= 128) {
    // RING=2 measured faster (3 blocks/CU): down 116 -> 85 us @ M=96
    if (ring_env == 3) LAUNCH_SK(2, 3);
    else LAUNCH_SK(2, 2);
  }
  else LAUNCH_SK(4, 2);
#undef LAUNCH_SK
  HIP_CHECK_KERNEL();
  if (splitk == 1) return;
  const long total = (long)M * N;
  const long grid = std::min<long>((total + 1023) / 1024, 2048);
  hipLaunchKernelGGL(fp8_reduce_kernel, dim3(grid), dim3(256), 0, stream,
                     (__hip_bfloat16 *)out.data_ptr(), wsp, bias_ptr, M, N,
                     splitk);
  HIP_CHECK_KERNEL();
}

void moe_gemm_fp8(torch::Tensor C, torch::Tensor A, torch::Tensor As,
                  torch::Tensor W, torch::Tensor Ws,
                  torch::Tensor sorted_ids, torch::Tensor expert_blocks,
                  torch::Tensor n_post_pad,
                  c10::optional<torch::Tensor> topk_weights, long n_pairs,
                  long topk, long block_m, bool scatter) {
  TORCH_CHECK(A.is_contiguous() && W.is_contiguous() && C.is_contiguous());
  const int K = W.size(2);
  const int Nd = W.size(1);
  TORCH_CHECK(A.size(-1) == K);
  TORCH_CHECK(K % MOE_BK == 0, "fp8 moe: K must be a multiple of 128");
  const int max_blocks = expert_blocks.numel();
  const float *tw = nullptr;
  if (topk_weights.has_value()) tw = topk_weights->data_ptr<float>();
  auto stream = at::cuda::getCurrentCUDAStream();
#define LAUNCH_M(BM, BNV, SC)                                                \
  hipLaunchKernelGGL((moe_gemm_fp8_kernel<BM, BNV, SC>),                     \
                     dim3(max_blocks, (Nd + BNV - 1) / BNV),                 \
                     dim3(BLOCK), 0, stream,                                 \
                     (__hip_bfloat16 *)C.data_ptr(),                         \
                     (const unsigned char *)A.data_ptr(),                    \
                     As.data_ptr<float>(),                                   \
                     (const unsigned char *)W.data_ptr(),                    \
                     Ws.data_ptr<float>(), sorted_ids.data_ptr<int>(),       \
                     expert_blocks.data_ptr<int>(),                          \
                     n_post_pad.data_ptr<int>(), tw, (int)n_pairs, K, Nd,    \
                     (int)topk)
  if (block_m == 16) {
    if (scatter) LAUNCH_M(16, 64, true); else LAUNCH_M(16, 64, false);
  } else if (block_m == 32) {
    if (scatter) LAUNCH_M(32, 64, true); else LAUNCH_M(32, 64, false);
  } else if (block_m == 64) {
    if (scatter) LAUNCH_M(64, 64, true); else LAUNCH_M(64, 64, false);
  } else if (block_m == 164) {  // BM=64, BN=256 (dense prefill)
    if (scatter) LAUNCH_M(64, 256, true); else LAUNCH_M(64, 256, false);
  } else {
    TORCH_CHECK(false, "moe_gemm_fp8: block_m must be 16/32/64/164");
  }
#undef LAUNCH_M
  HIP_CHECK_KERNEL();
}
