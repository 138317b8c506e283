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

template <int MB, int RING>
__global__ __launch_bounds__(BLOCK) void int4_skinny_kernel(
    float *__restrict__ partial,            // [SPLITK, M, N]
    const __hip_bfloat16 *__restrict__ x,   // [M, K]
    const unsigned char *__restrict__ wq,   // [N, K/2]
    const float *__restrict__ sb,           // [N, K/group, 2]
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
#pragma unroll
    for (int g = 0; g < 2; ++g) {
      // scales (4 B/lane), then biases (4 B/lane)
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int *>(
              sb + ((long)n * kgroups + g0 + g) * 2),
          reinterpret_cast<unsigned int *>(base + SB_OFF + g * 512), 4,
          0, 0);
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int *>(
              sb + ((long)n * kgroups + g0 + g) * 2 + 1),
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
          __hip_bfloat16 *be = reinterpret_cast<__hip_bfloat16 *>(&bfrag);
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const float q = (float)((w4 >> (4 * e)) & 0xF);
            be[e] = __float2bfloat16(q * s + bz);
          }
          acc[mb][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag, bfrag, acc[mb][nt], 0, 0, 0);
        }
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
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

void int4_skinny_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor wq,
                      torch::Tensor sb, c10::optional<torch::Tensor> bias,
                      torch::Tensor workspace, long group) {
  const int M = x.size(0), K = x.size(1), N = wq.size(0);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(wq.scalar_type() == at::kByte && wq.is_contiguous());
  TORCH_CHECK(sb.scalar_type() == at::kFloat && sb.is_contiguous());
  TORCH_CHECK(K % BK == 0, "int4 gemm: K must be a multiple of 256");
  TORCH_CHECK(group == 128, "int4 gemm: group size 128");
  TORCH_CHECK(M <= 256);
  const int n_wg = (N + BN - 1) / BN;
  int splitk = 1;
  while (splitk < 16 && n_wg * splitk < 512 && (K / (splitk * 2)) >= BK)
    splitk *= 2;
  int k_slice = (K + splitk - 1) / splitk;
  k_slice = ((k_slice + BK - 1) / BK) * BK;
  splitk = (K + k_slice - 1) / k_slice;
  TORCH_CHECK(workspace.numel() >= (long)splitk * M * N);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto *ws = workspace.data_ptr<float>();
#define LAUNCH_I4(MB, RING)                                                 \
  hipLaunchKernelGGL((int4_skinny_kernel<MB, RING>),                        \
                     dim3(n_wg, splitk), dim3(BLOCK), 0, stream, ws,        \
                     (const __hip_bfloat16 *)x.data_ptr(),                  \
                     wq.data_ptr<unsigned char>(), sb.data_ptr<float>(),    \
                     M, N, K, k_slice, (int)group)
  // LDS: slot = 8K (W) + MB*32K (X); MB=4 fits only single-buffered
  if (M <= 64) LAUNCH_I4(1, 3);
  else if (M <= 128) LAUNCH_I4(2, 2);
  else LAUNCH_I4(4, 1);
#undef LAUNCH_I4
  HIP_CHECK_KERNEL();
  const float *bias_ptr = nullptr;
  if (bias.has_value()) bias_ptr = bias->data_ptr<float>();
  const long total = (long)M * N;
  const long grid = std::min<long>((total + 1023) / 1024, 2048);
  hipLaunchKernelGGL(int4_reduce_kernel, dim3(grid), dim3(256), 0, stream,
                     (__hip_bfloat16 *)out.data_ptr(), ws, bias_ptr, M, N,
                     splitk);
  HIP_CHECK_KERNEL();
}
