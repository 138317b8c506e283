// int4 (AWQ/GPTQ) fused-dequant weight-streaming GEMM for gfx950.
//
// Reference role: gptq_marlin_repack + moe_wna16_marlin_gemm
// (_custom_ops.py:499-572) — int4 weights execute WITHOUT a bf16
// dequant materialization. MI355X shape: the skinny glds-ring GEMM
// streaming PACKED nibbles (0.5 B/elem — 4x less weight traffic than
// bf16) with in-fragment dequant: each MFMA B-fragment unpacks 8
// nibbles from one dword and applies the group's (scale, bias) with
// bias = -zero*scale, gathered through the same counted-vmcnt ring as
// the fp8 kernel's scales.
//
// Canonical layout (produced once at load by
// layers/quantization/int4.py::repack_canonical):
//   wq4    uint8 [N, K/2]   (low nibble = even k, high = odd k)
//   sb     fp32  [N, K/group, 2]  ({scale, -zero*scale} per group)
// Tile: BK = 256 elems = 128 B packed per W row; X tile bf16 512 B/row.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 i4_bf8;
typedef __attribute__((ext_vector_type(4))) float i4_f4;

namespace {

constexpr int BLOCK = 256;
constexpr int BN = 64;
constexpr int BK = 256;          // elements per K tile (2 quant groups)
constexpr int WROW_B = BK / 2;   // 128 B packed W row
constexpr int XROW_B = BK * 2;   // 512 B bf16 X row

DEV_INLINE int swz(int row, int byte_off) {
  return byte_off ^ ((row & 7) << 4);
}

template <int AUX>
DEV_INLINE void glds16(const unsigned char *gsrc, char *lds_ptr) {
  __builtin_amdgcn_global_load_lds(
      reinterpret_cast<const unsigned int *>(gsrc),
      reinterpret_cast<unsigned int *>(lds_ptr), 16, 0, AUX);
}

// 8 nibbles of one packed dword -> bf16[8] as q*s + bz. Byte-wise
// v_cvt_f32_ubyte on the masked even/odd nibble planes: ~2 AND + 8 cvt
// + 8 fma + 8 cvt instead of 8x (bfe + cvt + fma + cvt) — the unpack
// is the VALU hot loop of the whole kernel (PMC: 91% of issue).
DEV_INLINE void unpack8(unsigned int w4, float s, float bz,
                        __hip_bfloat16 *be) {
  const unsigned int lo = w4 & 0x0F0F0F0Fu;         // elems 0,2,4,6
  const unsigned int hi = (w4 >> 4) & 0x0F0F0F0Fu;  // elems 1,3,5,7
  // uitofp(byte-extract) pattern-matches to v_cvt_f32_ubyte0..3
  be[0] = __float2bfloat16((float)(lo & 0xffu) * s + bz);
  be[2] = __float2bfloat16((float)((lo >> 8) & 0xffu) * s + bz);
  be[4] = __float2bfloat16((float)((lo >> 16) & 0xffu) * s + bz);
  be[6] = __float2bfloat16((float)(lo >> 24) * s + bz);
  be[1] = __float2bfloat16((float)(hi & 0xffu) * s + bz);
  be[3] = __float2bfloat16((float)((hi >> 8) & 0xffu) * s + bz);
  be[5] = __float2bfloat16((float)((hi >> 16) & 0xffu) * s + bz);
  be[7] = __float2bfloat16((float)(hi >> 24) * s + bz);
}

template <int MB, int RING>
__global__ __launch_bounds__(BLOCK) void int4_skinny_kernel(
    float *__restrict__ partial,            // [SPLITK, M, N]
    const __hip_bfloat16 *__restrict__ x,   // [M, K]
    const unsigned char *__restrict__ wq,   // [N, K/2]
    const float *__restrict__ sbt,          // [K/group, 2, N] transposed
    __hip_bfloat16 *__restrict__ out,       // non-null iff splitk == 1
    const float *__restrict__ bias,
    int M, int N, int K, int k_slice, int group) {
  constexpr int TILE_B = BN * WROW_B + MB * 64 * XROW_B;
  const int n0 = blockIdx.x * BN;
  const int z = blockIdx.y;
  const int kt_begin = z * (k_slice / BK);
  const int kt_end = min(K / BK, kt_begin + k_slice / BK);
  const int nkt = kt_end - kt_begin;
  if (nkt <= 0) return;
  const int kgroups = K / group;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;

  // slot layout: [W 8 KB][X MB*32 KB][sb gpt*512 B]
  constexpr int SB_OFF = TILE_B;
  constexpr int SLOT_B = TILE_B + 4 * 512;  // up to 4 groups per tile
  __shared__ __attribute__((aligned(16))) char smem[RING * SLOT_B];

  constexpr int GL_PER_WAVE = TILE_B / 1024 / 4;
  const unsigned char *gsrc[GL_PER_WAVE];
  {
#pragma unroll
    for (int j = 0; j < GL_PER_WAVE; ++j) {
      const int p = (wave * GL_PER_WAVE + j) * 1024 + lane * 16;
      if (p < BN * WROW_B) {
        const int row = p / WROW_B;
        const int col = swz(row, p % WROW_B);
        const int n = min(n0 + row, N - 1);
        gsrc[j] = wq + (long)n * (K / 2) + col;
      } else {
        const int px = p - BN * WROW_B;
        const int m = min(px / XROW_B, M - 1);
        const int col = px % XROW_B;  // bytes within the bf16 row
        gsrc[j] = reinterpret_cast<const unsigned char *>(x) +
                  ((long)m * K) * 2 + col;
      }
    }
  }

  // per-stage vm ops per wave: GL_PER_WAVE glds16 + 2 groups x 2 glds4
  // (scale and bias gathers, every wave issues identical copies so the
  // counted waits stay uniform)
  auto stage = [&](int kt, int slot) {
    char *base = smem + slot * SLOT_B;
    const long koff_w = (long)(kt_begin + kt) * WROW_B;   // packed bytes
    const long koff_x = (long)(kt_begin + kt) * XROW_B;   // bf16 bytes
#pragma unroll
    for (int j = 0; j < GL_PER_WAVE; ++j) {
      const int p = (wave * GL_PER_WAVE + j) * 1024;
      if (p < BN * WROW_B)
        glds16<2>(gsrc[j] + koff_w, base + p);
      else
        glds16<0>(gsrc[j] + koff_x, base + p);
    }
    const int g0 = (kt_begin + kt) * 2;  // group 128, BK 256 -> gpt = 2
    const int n = min(n0 + lane, N - 1);
    // sbt is [K/group, 2, N]: each (group, scale|bias) row is lane-
    // contiguous — a few cache lines, not a 64-line strided gather
#pragma unroll
    for (int g = 0; g < 2; ++g) {
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int *>(
              sbt + ((long)(g0 + g) * 2 + 0) * N + n),
          reinterpret_cast<unsigned int *>(base + SB_OFF + g * 512), 4,
          0, 0);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int *>(
              sbt + ((long)(g0 + g) * 2 + 1) * N + n),
          reinterpret_cast<unsigned int *>(base + SB_OFF + g * 512 + 256),
          4, 0, 0);
    }
  };

  i4_f4 acc[MB][BN / 16];
#pragma unroll
  for (int mb = 0; mb < MB; ++mb)
#pragma unroll
    for (int nt = 0; nt < BN / 16; ++nt) acc[mb][nt] = i4_f4{0, 0, 0, 0};

  const int pre = min(RING - 1, nkt);
  for (int t = 0; t < pre; ++t) stage(t, t % RING);
  constexpr int VMS = GL_PER_WAVE + 4;  // glds16s + 2x2 scale gathers

  for (int kt = 0; kt < nkt; ++kt) {
    const int slot = kt % RING;
    if (kt + RING - 1 < nkt) stage(kt + RING - 1, (kt + RING - 1) % RING);
    const int ahead = min(nkt - 1 - kt, RING - 1);
    if (RING >= 3 && ahead >= 2) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"n"(2 * VMS) : "memory");
    } else if (ahead == 1) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"n"(1 * VMS) : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    const char *wbase = smem + slot * SLOT_B;
    const char *xbase = wbase + BN * WROW_B;
    const float *sbb = reinterpret_cast<const float *>(wbase + SB_OFF);

#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      const int g = (ks * 32) / 128;  // quant group within the tile
#pragma unroll
      for (int mb = 0; mb < MB; ++mb) {
        const int arow = mb * 64 + wave * 16 + l16;
        i4_bf8 afrag = *reinterpret_cast<const i4_bf8 *>(
            xbase + arow * XROW_B + (ks * 32 + lhi * 8) * 2);
#pragma unroll
        for (int nt = 0; nt < BN / 16; ++nt) {
          const int brow = nt * 16 + l16;
          // unpack 8 nibbles from one dword of the packed row
          const unsigned int w4 = *reinterpret_cast<const unsigned int *>(
              wbase + brow * WROW_B +
              swz(brow, (ks * 32 + lhi * 8) / 2));
          const float s = sbb[g * 128 + brow];
          const float bz = sbb[g * 128 + 64 + brow];
          i4_bf8 bfrag;
          unpack8(w4, s, bz,
                  reinterpret_cast<__hip_bfloat16 *>(&bfrag));
          acc[mb][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag, bfrag, acc[mb][nt], 0, 0, 0);
        }
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  if (out != nullptr) {  // splitk == 1: direct bf16 (+bias), no reduce
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

// --------------------------------------------- MB=1 decode variant
// Role-swapped tile for M <= 64. PMC on the generic tile showed 91% of
// issue cycles were VALU: with X rows as the MFMA A-operand every wave
// unpacks the ENTIRE 64-row W tile (4x duplicated). Here each wave owns
// 16 W rows as the A-operand (unpacks each packed dword exactly once)
// and the X tokens are the shared B-operand staged in LDS. SBK=128
// (one quant group per stage): slot = 4 KB W + 16 KB X + 512 B
// (scale,bias) = 20.5 KB, RING=3 -> 2 blocks/CU.
template <int RING>
__global__ __launch_bounds__(BLOCK) void int4_skinny_mb1_kernel(
    float *__restrict__ partial,            // [SPLITK, M, N]
    const __hip_bfloat16 *__restrict__ x,   // [M, K]
    const unsigned char *__restrict__ wq,   // [N, K/2]
    const float *__restrict__ sbt,          // [K/group, 2, N]
    __hip_bfloat16 *__restrict__ out,       // non-null iff splitk == 1
    const float *__restrict__ bias,
    int M, int N, int K, int k_slice) {
  constexpr int SBK = 128;       // elems per stage = 1 quant group
  constexpr int WR_B = SBK / 2;  // 64 B packed W row
  constexpr int XR_B = SBK * 2;  // 256 B bf16 X row
  const int n0 = blockIdx.x * BN;
  const int z = blockIdx.y;
  const int kt_begin = z * (k_slice / SBK);
  const int kt_end = min(K / SBK, kt_begin + k_slice / SBK);
  const int nkt = kt_end - kt_begin;
  if (nkt <= 0) return;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;

  constexpr int W_B = BN * WR_B;       // 4 KB
  constexpr int X_OFF = W_B;
  constexpr int X_B = 64 * XR_B;       // 16 KB
  constexpr int SB_OFF = W_B + X_B;
  constexpr int SLOT_B = SB_OFF + 512;
  __shared__ __attribute__((aligned(16))) char smem[RING * SLOT_B];

  // 64-B W rows: swizzle must stay inside the row (bits 4-5)
  auto swzw = [](int row, int off) { return off ^ ((row & 3) << 4); };
  // 256-B X rows: bits 4-6
  auto swzx = [](int row, int off) { return off ^ ((row & 7) << 4); };

  constexpr int GL_W = W_B / 1024 / 4;   // 1 glds16 per wave
  constexpr int GL_X = X_B / 1024 / 4;   // 4 glds16 per wave
  const unsigned char *gw[GL_W];
  const unsigned char *gx[GL_X];
  {
#pragma unroll
    for (int j = 0; j < GL_W; ++j) {
      const int p = (wave * GL_W + j) * 1024 + lane * 16;
      const int row = p / WR_B;
      const int n = min(n0 + row, N - 1);
      gw[j] = wq + (long)n * (K / 2) + swzw(row, p % WR_B);
    }
#pragma unroll
    for (int j = 0; j < GL_X; ++j) {
      const int p = (wave * GL_X + j) * 1024 + lane * 16;
      const int row = p / XR_B;
      const int m = min(row, M - 1);
      gx[j] = reinterpret_cast<const unsigned char *>(x) +
              (long)m * K * 2 + swzx(row, p % XR_B);
    }
  }

  auto stage = [&](int kt, int slot) {
    char *base = smem + slot * SLOT_B;
    const long kw = (long)(kt_begin + kt) * WR_B;
    const long kx = (long)(kt_begin + kt) * XR_B;
#pragma unroll
    for (int j = 0; j < GL_W; ++j)
      glds16<2>(gw[j] + kw, base + (wave * GL_W + j) * 1024);
#pragma unroll
    for (int j = 0; j < GL_X; ++j)
      glds16<0>(gx[j] + kx, base + X_OFF + (wave * GL_X + j) * 1024);
    const int g = kt_begin + kt;
    const int n = min(n0 + lane, N - 1);
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const unsigned int *>(sbt + ((long)g * 2) * N + n),
        reinterpret_cast<unsigned int *>(base + SB_OFF), 4, 0, 0);
    __builtin_amdgcn_global_load_lds(
        reinterpret_cast<const unsigned int *>(
            sbt + ((long)g * 2 + 1) * N + n),
        reinterpret_cast<unsigned int *>(base + SB_OFF + 256), 4, 0, 0);
  };
  constexpr int VMS = GL_W + GL_X + 2;

  i4_f4 acc[4];  // 16 own-N-rows x 64 tokens
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) acc[mt] = i4_f4{0, 0, 0, 0};

  const int wr = wave * 16 + l16;  // this lane's W row within the tile
  const int pre = min(RING - 1, nkt);
  for (int t = 0; t < pre; ++t) stage(t, t % RING);

  for (int kt = 0; kt < nkt; ++kt) {
    const int slot = kt % RING;
    if (kt + RING - 1 < nkt) stage(kt + RING - 1, (kt + RING - 1) % RING);
    const int ahead = min(nkt - 1 - kt, RING - 1);
    if (RING >= 3 && ahead >= 2) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"n"(2 * VMS) : "memory");
    } else if (ahead == 1) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"n"(1 * VMS) : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    const char *wbase = smem + slot * SLOT_B;
    const char *xbase = wbase + X_OFF;
    const float *sbl = reinterpret_cast<const float *>(wbase + SB_OFF);
    const float s = sbl[wr];
    const float bz = sbl[64 + wr];

#pragma unroll
    for (int ks = 0; ks < SBK / 32; ++ks) {
      const unsigned int w4 = *reinterpret_cast<const unsigned int *>(
          wbase + wr * WR_B + swzw(wr, (ks * 32 + lhi * 8) / 2));
      i4_bf8 afrag;
      unpack8(w4, s, bz, reinterpret_cast<__hip_bfloat16 *>(&afrag));
#pragma unroll
      for (int mt = 0; mt < 4; ++mt) {
        const int m = mt * 16 + l16;
        const i4_bf8 bfrag = *reinterpret_cast<const i4_bf8 *>(
            xbase + m * XR_B + swzx(m, (ks * 32 + lhi * 8) * 2));
        acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[mt], 0, 0, 0);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  const int n = n0 + wave * 16 + lhi * 4;
  if (out != nullptr) {  // splitk == 1: direct bf16 (+bias)
#pragma unroll
    for (int mt = 0; mt < 4; ++mt) {
      const int m = mt * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        if (m < M && n + r < N) {
          float v = acc[mt][r];
          if (bias) v += bias[n + r];
          out[(long)m * N + n + r] = __float2bfloat16(v);
        }
      }
    }
    return;
  }
  float *base = partial + (long)z * M * N;
#pragma unroll
  for (int mt = 0; mt < 4; ++mt) {
    const int m = mt * 16 + l16;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      if (m < M && n + r < N) base[(long)m * N + n + r] = acc[mt][r];
    }
  }
}

// ------------------------------------------------- int4 grouped MoE
// moe.hip's grouped-GEMM pipeline with packed-nibble weights (w4a16 —
// activations stay bf16, matching the reference's moe_wna16 kernels).
// BK = 128 elems = one quant group per k tile (group size 128).
constexpr int MOE_BN = 64;
constexpr int MOE_BK = 128;

template <int BM, bool SCATTER>
__global__ __launch_bounds__(BLOCK) void moe_gemm_int4_kernel(
    __hip_bfloat16 *__restrict__ C,
    const __hip_bfloat16 *__restrict__ A,   // [Ta, K] bf16
    const unsigned char *__restrict__ W,    // [E, Nd, K/2] packed
    const float *__restrict__ SB,           // [E, Nd, K/128, 2]
    const int *__restrict__ sorted_ids, const int *__restrict__ expert_blocks,
    const int *__restrict__ n_post_pad,
    const float *__restrict__ topk_w, int n_pairs, int K, int Nd,
    int topk) {
  const int mb = blockIdx.x;
  if (mb * BM >= n_post_pad[0]) return;
  const int nb = blockIdx.y;
  const int e = expert_blocks[mb];
  const int kgroups = K / MOE_BK;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;

  constexpr int WM = BM / 16;
  constexpr int WN_TILES = WM;
  const int wm = wave % WM;
  const int wn = wave / WM;

  constexpr int APAD = 8;
  __shared__ __hip_bfloat16 a_tile[2][BM * (MOE_BK + APAD)];
  __shared__ unsigned char w_tile[2][MOE_BN * (MOE_BK / 2 + 8)];
  __shared__ float sb_tile[2][MOE_BN * 2];

  const int g0 = mb * BM;
  constexpr int A_CH = BM * MOE_BK * 2 / 16;       // 16-B chunks (bf16)
  constexpr int W_CH = MOE_BN * (MOE_BK / 2) / 16; // 16-B chunks (nibbles)
  typedef __attribute__((ext_vector_type(4))) int int4v;
  int4v areg[(A_CH + BLOCK - 1) / BLOCK];
  int4v wreg[(W_CH + BLOCK - 1) / BLOCK];
  float sreg[2];  // (s, b) staged by the first MOE_BN*2 threads

  const long w_base = (long)e * Nd * (K / 2);
  const long sb_base = (long)e * Nd * kgroups * 2;

  auto load_tiles = [&](int kb) {
#pragma unroll
    for (int it = 0; it < (A_CH + BLOCK - 1) / BLOCK; ++it) {
      const int idx = tid + it * BLOCK;
      if (idx < A_CH) {
        const int m = idx / (MOE_BK * 2 / 16);
        const int c = idx % (MOE_BK * 2 / 16);
        const int g = g0 + m;
        const int pair = sorted_ids[g];
        const bool valid = pair < n_pairs;
        long arow = SCATTER ? g : (valid ? pair / topk : 0);
        if (valid)
          areg[it] = *reinterpret_cast<const int4v *>(
              reinterpret_cast<const char *>(A + arow * (long)K +
                                             kb * MOE_BK) + c * 16);
        else
          areg[it] = int4v{0, 0, 0, 0};
      }
    }
#pragma unroll
    for (int it = 0; it < (W_CH + BLOCK - 1) / BLOCK; ++it) {
      const int idx = tid + it * BLOCK;
      if (idx < W_CH) {
        const int n = idx / (MOE_BK / 2 / 16);
        const int c = idx % (MOE_BK / 2 / 16);
        if (nb * MOE_BN + n < Nd) {
          const int4v *wp = reinterpret_cast<const int4v *>(
              W + w_base + (long)(nb * MOE_BN + n) * (K / 2) +
              kb * (MOE_BK / 2) + c * 16);
          // nt measured NEGATIVE here (DeepSeek fp8 MoE B256: many
          // experts span 2 m-blocks, nt kills the W reuse) — cached
          wreg[it] = *wp;
        } else
          wreg[it] = int4v{0, 0, 0, 0};
      }
    }
    if (tid < MOE_BN * 2) {
      const int n = tid >> 1;
      const int which = tid & 1;
      const int nn = min(nb * MOE_BN + n, Nd - 1);
      sreg[0] = SB[sb_base + ((long)nn * kgroups + kb) * 2 + which];
    }
  };
  auto write_tiles = [&](int buf) {
#pragma unroll
    for (int it = 0; it < (A_CH + BLOCK - 1) / BLOCK; ++it) {
      const int idx = tid + it * BLOCK;
      if (idx < A_CH) {
        const int m = idx / (MOE_BK * 2 / 16);
        const int c = idx % (MOE_BK * 2 / 16);
        *reinterpret_cast<int4v *>(
            reinterpret_cast<char *>(&a_tile[buf][m * (MOE_BK + APAD)]) +
            c * 16) = areg[it];
      }
    }
#pragma unroll
    for (int it = 0; it < (W_CH + BLOCK - 1) / BLOCK; ++it) {
      const int idx = tid + it * BLOCK;
      if (idx < W_CH) {
        const int n = idx / (MOE_BK / 2 / 16);
        const int c = idx % (MOE_BK / 2 / 16);
        *reinterpret_cast<int4v *>(
            &w_tile[buf][n * (MOE_BK / 2 + 8) + c * 16]) = wreg[it];
      }
    }
    if (tid < MOE_BN * 2) sb_tile[buf][tid] = sreg[0];
  };

  i4_f4 acc[WN_TILES];
#pragma unroll
  for (int t = 0; t < WN_TILES; ++t) acc[t] = i4_f4{0, 0, 0, 0};

  load_tiles(0);
  write_tiles(0);
  __syncthreads();

  int cur = 0;
  const int nkb = K / MOE_BK;
  for (int kb = 0; kb < nkb; ++kb) {
    if (kb + 1 < nkb) load_tiles(kb + 1);

    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < MOE_BK / 32; ++ks) {
      i4_bf8 afrag = *reinterpret_cast<const i4_bf8 *>(
          &a_tile[cur][(wm * 16 + l16) * (MOE_BK + APAD) + ks * 32 +
                       lhi * 8]);
#pragma unroll
      for (int t = 0; t < WN_TILES; ++t) {
        const int brow = (wn * WN_TILES + t) * 16 + l16;
        const unsigned int w4 = *reinterpret_cast<const unsigned int *>(
            &w_tile[cur][brow * (MOE_BK / 2 + 8) + (ks * 32 + lhi * 8) / 2]);
        const float s = sb_tile[cur][brow * 2];
        const float bz = sb_tile[cur][brow * 2 + 1];
        i4_bf8 bfrag;
        unpack8(w4, s, bz,
                reinterpret_cast<__hip_bfloat16 *>(&bfrag));
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[t], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

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

__global__ void int4_reduce_kernel(__hip_bfloat16 *__restrict__ out,
                                   const float *__restrict__ partial,
                                   const float *__restrict__ bias, int M,
                                   int N, int splitk) {
  const long total = (long)M * N;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    float v = bias ? bias[i % N] : 0.f;
    for (int zz = 0; zz < splitk; ++zz)
      v += partial[(long)zz * total + i];
    out[i] = __float2bfloat16(v);
  }
}

}  // namespace

void moe_gemm_int4(torch::Tensor C, torch::Tensor A, torch::Tensor W,
                   torch::Tensor SB, torch::Tensor sorted_ids,
                   torch::Tensor expert_blocks, torch::Tensor n_post_pad,
                   c10::optional<torch::Tensor> topk_weights, long n_pairs,
                   long topk, long block_m, bool scatter) {
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 && A.is_contiguous());
  TORCH_CHECK(W.scalar_type() == at::kByte && W.is_contiguous());
  TORCH_CHECK(SB.scalar_type() == at::kFloat && SB.is_contiguous());
  const int Nd = W.size(1);
  const int K = W.size(2) * 2;
  TORCH_CHECK(A.size(-1) == K);
  TORCH_CHECK(K % MOE_BK == 0, "int4 moe: K must be a multiple of 128");
  const int max_blocks = expert_blocks.numel();
  const int n_tiles = (Nd + MOE_BN - 1) / MOE_BN;
  const float *tw = nullptr;
  if (topk_weights.has_value()) tw = topk_weights->data_ptr<float>();
  auto stream = at::cuda::getCurrentCUDAStream();
#define LAUNCH_MI4(BM, SC)                                                  \
  hipLaunchKernelGGL((moe_gemm_int4_kernel<BM, SC>),                        \
                     dim3(max_blocks, n_tiles), dim3(BLOCK), 0, stream,     \
                     (__hip_bfloat16 *)C.data_ptr(),                        \
                     (const __hip_bfloat16 *)A.data_ptr(),                  \
                     W.data_ptr<unsigned char>(), SB.data_ptr<float>(),     \
                     sorted_ids.data_ptr<int>(),                            \
                     expert_blocks.data_ptr<int>(),                         \
                     n_post_pad.data_ptr<int>(), tw, (int)n_pairs, K, Nd,   \
                     (int)topk)
  if (block_m == 16) {
    if (scatter) LAUNCH_MI4(16, true); else LAUNCH_MI4(16, false);
  } else if (block_m == 32) {
    if (scatter) LAUNCH_MI4(32, true); else LAUNCH_MI4(32, false);
  } else if (block_m == 64) {
    if (scatter) LAUNCH_MI4(64, true); else LAUNCH_MI4(64, false);
  } else {
    TORCH_CHECK(false, "moe_gemm_int4: block_m must be 16/32/64");
  }
#undef LAUNCH_MI4
  HIP_CHECK_KERNEL();
}

void int4_skinny_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor wq,
                      torch::Tensor sbt, c10::optional<torch::Tensor> bias,
                      torch::Tensor workspace, long group) {
  const int M = x.size(0), K = x.size(1), N = wq.size(0);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(wq.scalar_type() == at::kByte && wq.is_contiguous());
  TORCH_CHECK(sbt.scalar_type() == at::kFloat && sbt.is_contiguous());
  TORCH_CHECK(sbt.numel() == (long)(K / group) * 2 * N,
              "int4 skinny: transposed (scale,bias) [K/group, 2, N]");
  TORCH_CHECK(K % BK == 0, "int4 gemm: K must be a multiple of 256");
  TORCH_CHECK(group == 128, "int4 gemm: group size 128");
  TORCH_CHECK(M <= 256);
  const int n_wg = (N + BN - 1) / BN;
  int splitk = 1;
  while (splitk < 16 && n_wg * (splitk * 2) <= 1024 &&
         (K / BK) / (splitk * 2) >= 8)
    splitk *= 2;
  int k_slice = (K + splitk - 1) / splitk;
  k_slice = ((k_slice + BK - 1) / BK) * BK;
  splitk = (K + k_slice - 1) / k_slice;
  TORCH_CHECK(splitk == 1 || workspace.numel() >= (long)splitk * M * N);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto *ws = workspace.data_ptr<float>();
  const float *bias_ptr = nullptr;
  if (bias.has_value()) bias_ptr = bias->data_ptr<float>();
  __hip_bfloat16 *outp =
      splitk == 1 ? (__hip_bfloat16 *)out.data_ptr() : nullptr;
#define LAUNCH_I4(MB, RING)                                                 \
  hipLaunchKernelGGL((int4_skinny_kernel<MB, RING>),                        \
                     dim3(n_wg, splitk), dim3(BLOCK), 0, stream, ws,        \
                     (const __hip_bfloat16 *)x.data_ptr(),                  \
                     wq.data_ptr<unsigned char>(), sbt.data_ptr<float>(),   \
                     outp, bias_ptr, M, N, K, k_slice, (int)group)
  // LDS: slot = 8K (W) + MB*32K (X); MB=4 fits only single-buffered.
  // M<=64 uses the register-X variant (10 KB slots, ~4 blocks/CU).
  if (M <= 64)
    hipLaunchKernelGGL((int4_skinny_mb1_kernel<3>),
                       dim3(n_wg, splitk), dim3(BLOCK), 0, stream, ws,
                       (const __hip_bfloat16 *)x.data_ptr(),
                       wq.data_ptr<unsigned char>(), sbt.data_ptr<float>(),
                       outp, bias_ptr, M, N, K, k_slice);
  else if (M <= 128) LAUNCH_I4(2, 2);
  else LAUNCH_I4(4, 1);
#undef LAUNCH_I4
  HIP_CHECK_KERNEL();
  if (splitk == 1) return;
  const long total = (long)M * N;
  const long grid = std::min<long>((total + 1023) / 1024, 2048);
  hipLaunchKernelGGL(int4_reduce_kernel, dim3(grid), dim3(256), 0, stream,
                     (__hip_bfloat16 *)out.data_ptr(), ws, bias_ptr, M, N,
                     splitk);
  HIP_CHECK_KERNEL();
}
