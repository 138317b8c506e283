// Decode-regime ("skinny") GEMM for gfx950: out[M,N] = x[M,K] @ W[N,K]^T,
// bf16 in / fp32 accumulate / bf16 out, M <= 256.
//
// Why it exists: at decode batch sizes hipBLASLt reaches only 1-3.8 TB/s
// of weight streaming on Qwen-32B projection shapes while the chip
// sustains ~6.3 TB/s (scripts/gemm_sweep.py, profiles/). The GEMM is a
// pure weight stream; the kernel's job is to read W once at full rate.
//
// v2 structure (guide T3+T4: glds ring with counted vmcnt):
//   grid = (N/BN, SPLITK), block = 256 (4 waves). One workgroup owns a
//   BN=64-column tile for ALL M rows (MB 64-row blocks) and a contiguous
//   K-slice, so W is streamed exactly once regardless of M.
//   Per K-step, a tile { W[64][64] | x[MB*64][64] } is DMA'd straight
//   into an LDS ring via __builtin_amdgcn_global_load_lds (16 B/lane),
//   with the T2 XOR swizzle applied on the per-lane SOURCE address
//   (glds writes lane-linear; rule 21). The ring keeps 2 tiles in
//   flight across raw s_barriers with counted s_waitcnt vmcnt(N) — the
//   +40..83% pattern of cdna_hip_programming.md §5 'Pipelining across
//   barriers'. MFMA 16x16x32: wave w computes rows of every M-block
//   (4 C-frags per block), B-fragments straight from the swizzled W
//   image.
//   Split-K partials fp32 [SPLITK, M, N]; reduce kernel sums + bias.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <cstdlib>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 sg_bf8;
typedef __attribute__((ext_vector_type(4))) float sg_f4;

namespace {

constexpr int BLOCK = 256;
constexpr int BN = 64;
constexpr int BK = 64;
constexpr int ROW_B = BK * 2;           // tile row bytes (128)

DEV_INLINE int swz(int row, int byte_off) {
  // 8 16-B slots per 128-B row; spread the 16-lane b128 column read
  return byte_off ^ ((row & 7) << 4);
}

// One glds instruction moves 64 lanes x 16 B = 1 KiB, lane-linear at
// lds_base + lane*16. The tile image is linear rows of 128 B; the
// source address carries the inverse swizzle.
template <int AUX>
DEV_INLINE void glds16(const __hip_bfloat16 *gsrc, char *lds_ptr) {
  __builtin_amdgcn_global_load_lds(
      reinterpret_cast<const unsigned int *>(gsrc),
      reinterpret_cast<unsigned int *>(lds_ptr), 16, 0, AUX);
}

template <int MB, int RING>
__global__ __launch_bounds__(BLOCK) void skinny_gemm_kernel(
    float *__restrict__ partial,          // [SPLITK, M, N]
    const __hip_bfloat16 *__restrict__ x, // [M, K]
    const __hip_bfloat16 *__restrict__ w, // [N, K]
    int M, int N, int K, int k_slice) {
  constexpr int TILE_B = (BN + MB * 64) * ROW_B;   // bytes per ring slot
  const int n0 = blockIdx.x * BN;
  const int z = blockIdx.y;
  const int k_begin = z * k_slice;
  const int k_end = min(K, k_begin + k_slice);
  const int nkt = (k_end - k_begin + BK - 1) / BK;  // K divisible by BK

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;

  // static LDS: the r1 replay fault ("write to read-only page") on
  // MB>=2 under hipGraph replay is suspected dynamic-LDS + graph
  // interaction; sizes are template constants, so declare statically
  __shared__ __attribute__((aligned(16))) char smem[RING * TILE_B];

  // ---- glds source mapping -------------------------------------------
  // Tile image: W rows [0,64) then x rows [0, MB*64), 128 B each, row
  // r's bytes XOR-swizzled. glds j of wave w covers image bytes
  // [(w*G + j)*1024, ...+1024), lane l -> byte p = base + l*16:
  // row = p/128, col = (p%128) ^ swz(row).
  constexpr int GL_PER_WAVE = TILE_B / 1024 / 4;   // glds per wave/tile
  const __hip_bfloat16 *gsrc[GL_PER_WAVE];
  {
    const long wrow_stride = K;  // elements
#pragma unroll
    for (int j = 0; j < GL_PER_WAVE; ++j) {
      const int p = (wave * GL_PER_WAVE + j) * 1024 + lane * 16;
      const int row = p / ROW_B;
      const int col = swz(row, p % ROW_B);         // involution
      if (row < BN) {
        const int n = min(n0 + row, N - 1);
        gsrc[j] = w + (long)n * wrow_stride + col / 2;
      } else {
        const int m = min(row - BN, M - 1);
        gsrc[j] = x + (long)m * K + col / 2;
      }
    }
  }

  // W rows are streamed ONCE per CU -> non-temporal (aux=2) lifts the
  // per-CU LDS-DMA landing cadence (guide row nt-weights: -18% landed
  // latency, chip 6.5-6.8 TB/s). x is re-read by every N-workgroup
  // through L2 -> default policy.
  auto stage = [&](int kt, int slot) {
    const int kb = k_begin + kt * BK;
    char *base = smem + slot * TILE_B;
#pragma unroll
    for (int j = 0; j < GL_PER_WAVE; ++j) {
      const int off = (wave * GL_PER_WAVE + j) * 1024;
      if (off < BN * ROW_B)
        glds16<2>(gsrc[j] + kb, base + off);
      else
        glds16<0>(gsrc[j] + kb, base + off);
    }
  };

  // ---- accumulators ---------------------------------------------------
  sg_f4 acc[MB][BN / 16];
#pragma unroll
  for (int mb = 0; mb < MB; ++mb)
#pragma unroll
    for (int nt = 0; nt < BN / 16; ++nt) acc[mb][nt] = sg_f4{0, 0, 0, 0};

  // ---- prologue: fill the ring ---------------------------------------
  const int pre = min(RING - 1, nkt);
  for (int t = 0; t < pre; ++t) stage(t, t % RING);

  for (int kt = 0; kt < nkt; ++kt) {
    const int slot = kt % RING;
    if (kt + RING - 1 < nkt) stage(kt + RING - 1, (kt + RING - 1) % RING);
    // Wait until tile kt's glds landed. vmcnt must equal the loads still
    // allowed in flight = GL_PER_WAVE x (tiles staged BEYOND kt). Near
    // the slice tail fewer tiles are ahead, so the count shrinks — a
    // fixed steady-state count would let the wait pass while tile kt's
    // own DMA is still flying (stale LDS reads).
    const int ahead = min(nkt - 1 - kt, RING - 1);
    if (RING >= 4 && ahead == 3) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(3 * GL_PER_WAVE)
                   : "memory");
    } else if (RING >= 3 && ahead == 2) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2 * GL_PER_WAVE)
                   : "memory");
    } else if (ahead == 1) {
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(1 * GL_PER_WAVE)
                   : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    const char *wbase = smem + slot * TILE_B;
    const char *xbase = wbase + BN * ROW_B;
#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      // B fragments: lane holds W[n = nt*16 + l16][k = ks*32 + lhi*8..]
      sg_bf8 bfrag[BN / 16];
#pragma unroll
      for (int nt = 0; nt < BN / 16; ++nt) {
        const int brow = nt * 16 + l16;
        bfrag[nt] = *reinterpret_cast<const sg_bf8 *>(
            wbase + brow * ROW_B + swz(brow, (ks * 32 + lhi * 8) * 2));
      }
#pragma unroll
      for (int mb = 0; mb < MB; ++mb) {
        const int arow = mb * 64 + wave * 16 + l16;
        sg_bf8 afrag = *reinterpret_cast<const sg_bf8 *>(
            xbase + arow * ROW_B + swz(arow, (ks * 32 + lhi * 8) * 2));
#pragma unroll
        for (int nt = 0; nt < BN / 16; ++nt)
          acc[mb][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag, bfrag[nt], acc[mb][nt], 0, 0, 0);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // ---- epilogue -------------------------------------------------------
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
__global__ void skinny_reduce_kernel(
    __hip_bfloat16 *__restrict__ out, const float *__restrict__ partial,
    const float *__restrict__ bias, int M, int N, int splitk) {
  const long total = (long)M * N;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    float v = bias ? bias[i % N] : 0.f;
    for (int z = 0; z < splitk; ++z) v += partial[(long)z * total + i];
    out[i] = __float2bfloat16(v);
  }
}

template <int MB, int RING>
void launch_skinny(float *partial, const __hip_bfloat16 *x,
                   const __hip_bfloat16 *w, int M, int N, int K, int k_slice,
                   int splitk, hipStream_t stream) {
  hipLaunchKernelGGL((skinny_gemm_kernel<MB, RING>),
                     dim3((N + BN - 1) / BN, splitk), dim3(BLOCK), 0,
                     stream, partial, x, w, M, N, K, k_slice);
}

}  // namespace

void skinny_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                 c10::optional<torch::Tensor> bias,
                 torch::Tensor workspace, long splitk_arg) {
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(K % BK == 0, "skinny_gemm: K must be a multiple of 64");
  TORCH_CHECK(M <= 256, "skinny_gemm: M <= 256");
  const int n_wg = (N + BN - 1) / BN;
  int splitk = (int)splitk_arg;
  if (splitk <= 0) {
    splitk = 1;
    while (splitk < 16 && n_wg * splitk < 512 && (K / (splitk * 2)) >= BK)
      splitk *= 2;
  }
  int k_slice = (K + splitk - 1) / splitk;
  k_slice = ((k_slice + BK - 1) / BK) * BK;
  splitk = (K + k_slice - 1) / k_slice;

  TORCH_CHECK(workspace.numel() >= (long)splitk * M * N,
              "skinny_gemm workspace too small");
  auto stream = at::cuda::getCurrentCUDAStream();
  auto *ws = workspace.data_ptr<float>();
  auto *xp = (const __hip_bfloat16 *)x.data_ptr();
  auto *wp = (const __hip_bfloat16 *)w.data_ptr();
  static int ring_env = [] {
    const char *e = getenv("SK_RING");
    return e ? atoi(e) : 0;
  }();
  if (M <= 64) {
    const int r = ring_env ? ring_env : 3;
    if (r == 2) launch_skinny<1, 2>(ws, xp, wp, M, N, K, k_slice, splitk,
                                    stream);
    else if (r == 4) launch_skinny<1, 4>(ws, xp, wp, M, N, K, k_slice,
                                         splitk, stream);
    else launch_skinny<1, 3>(ws, xp, wp, M, N, K, k_slice, splitk, stream);
  } else if (M <= 128) {
    const int r = ring_env ? ring_env : 3;
    if (r == 2) launch_skinny<2, 2>(ws, xp, wp, M, N, K, k_slice, splitk,
                                    stream);
    else launch_skinny<2, 3>(ws, xp, wp, M, N, K, k_slice, splitk, stream);
  } else {
    const int r = ring_env ? ring_env : 2;
    if (r >= 3) launch_skinny<4, 3>(ws, xp, wp, M, N, K, k_slice, splitk,
                                    stream);
    else launch_skinny<4, 2>(ws, xp, wp, M, N, K, k_slice, splitk, stream);
  }
  HIP_CHECK_KERNEL();

  const float *bias_ptr = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->scalar_type() == at::kFloat);
    bias_ptr = bias->data_ptr<float>();
  }
  const long total = (long)M * N;
  const int block = 256;
  const long grid = std::min<long>((total + block * 4 - 1) / (block * 4),
                                   2048);
  hipLaunchKernelGGL(skinny_reduce_kernel, dim3(grid), dim3(block), 0,
                     stream, (__hip_bfloat16 *)out.data_ptr(), ws, bias_ptr,
                     M, N, splitk);
  HIP_CHECK_KERNEL();
}
