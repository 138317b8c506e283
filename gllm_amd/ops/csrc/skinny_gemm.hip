// Decode-regime ("skinny") GEMM for gfx950: out[M,N] = x[M,K] @ W[N,K]^T,
// bf16 in / fp32 accumulate / bf16 out, M <= 256.
//
// Why it exists: at decode batch sizes hipBLASLt reaches only 1-3.8 TB/s
// of weight streaming on Qwen-32B projection shapes (profiles/
// r01_qwen32b_bench_kernels.md; scripts/gemm_sweep.py) while the chip
// sustains ~6.3 TB/s. In this regime the GEMM is a pure weight stream:
// the kernel's only job is to read W once at full bandwidth.
//
// Structure: grid = (N/BN, ceil(M/BM), SPLITK), 256 threads (4 waves).
//   Each workgroup owns a BN=64-column x BM=64-row output tile and a
//   contiguous K-slice. W tile [BN][BK] and x tile [BM][BK] are staged in
//   LDS (XOR-swizzled rows for conflict-free ds_read_b128), MFMA
//   16x16x32 per wave (wave = one 16-row M-tile x all 64 N-cols).
//   Register-staged T14 pipeline: next tile's global loads issue before
//   the compute phase, LDS writes land after the barrier.
//   Split-K partials are fp32 [SPLITK, M, N]; a reduce kernel sums them
//   (+bias) to bf16 — the slab round trip costs <<10% of W traffic at
//   these shapes.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 sg_bf8;
typedef __attribute__((ext_vector_type(4))) float sg_f4;

namespace {

constexpr int BLOCK = 256;
constexpr int BM = 64;
constexpr int BN = 64;
constexpr int BK = 64;

DEV_INLINE int swz(int row, int byte_off) {
  return byte_off ^ ((row & 7) << 4);  // BK*2=128-B rows: 8-slot spread
}

__global__ __launch_bounds__(BLOCK) void skinny_gemm_kernel(
    float *__restrict__ partial,          // [SPLITK, M, N]
    const __hip_bfloat16 *__restrict__ x, // [M, K]
    const __hip_bfloat16 *__restrict__ w, // [N, K]
    int M, int N, int K, int k_slice) {
  const int n0 = blockIdx.x * BN;
  const int m0 = blockIdx.y * BM;
  const int z = blockIdx.z;
  const int k_begin = z * k_slice;
  const int k_end = min(K, k_begin + k_slice);

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;

  __shared__ __hip_bfloat16 w_tile[BN * BK];
  __shared__ __hip_bfloat16 x_tile[BM * BK];

  // staging assignment: 256 threads x 32 B = 8 KB per tile.
  // thread covers tile row r = tid/8 (x2 iters of 32 rows... BN=64 rows,
  // 8 chunks of 16B per 128-B row): r = tid/8 + it*32, chunk c = tid%8.
  const int srow = tid >> 3;          // 0..31
  const int schunk = tid & 7;         // 16-B chunk within the row

  sg_f4 acc[BN / 16];                 // 4 C fragments (16 rows x 64 cols)
#pragma unroll
  for (int i = 0; i < BN / 16; ++i) acc[i] = sg_f4{0, 0, 0, 0};

  const int nkt = (k_end - k_begin + BK - 1) / BK;
  shortx8 wreg[2], xreg[2];

  auto load_tile = [&](int kt, shortx8 *wr, shortx8 *xr) {
    const int kb = k_begin + kt * BK;
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int row = srow + it * 32;
      const int kk = kb + schunk * 8;
      // W row n0+row; pad rows read row 0 (results discarded via N guard)
      const int wn = min(n0 + row, N - 1);
      wr[it] = (kk + 8 <= k_end)
                   ? *reinterpret_cast<const shortx8 *>(
                         w + (long)wn * K + kk)
                   : shortx8{0, 0, 0, 0, 0, 0, 0, 0};
      const int xm = m0 + row;
      xr[it] = (xm < M && kk + 8 <= k_end)
                   ? *reinterpret_cast<const shortx8 *>(
                         x + (long)xm * K + kk)
                   : shortx8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  };

  auto store_tile = [&](const shortx8 *wr, const shortx8 *xr) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int row = srow + it * 32;
      *reinterpret_cast<shortx8 *>(
          reinterpret_cast<char *>(&w_tile[row * BK]) +
          swz(row, schunk * 16)) = wr[it];
      *reinterpret_cast<shortx8 *>(
          reinterpret_cast<char *>(&x_tile[row * BK]) +
          swz(row, schunk * 16)) = xr[it];
    }
  };

  // prologue: tile 0
  load_tile(0, wreg, xreg);
  store_tile(wreg, xreg);
  __syncthreads();

  for (int kt = 0; kt < nkt; ++kt) {
    // issue next tile's loads before computing (T14 issue-early)
    if (kt + 1 < nkt) load_tile(kt + 1, wreg, xreg);

    // compute from LDS tile kt
#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      // A fragment: x rows m = wave*16 + l16, k = ks*32 + lhi*8
      const int arow = wave * 16 + l16;
      sg_bf8 afrag = *reinterpret_cast<const sg_bf8 *>(
          reinterpret_cast<const char *>(&x_tile[arow * BK]) +
          swz(arow, (ks * 32 + lhi * 8) * 2));
#pragma unroll
      for (int nt = 0; nt < BN / 16; ++nt) {
        const int brow = nt * 16 + l16;   // W row (output col)
        sg_bf8 bfrag = *reinterpret_cast<const sg_bf8 *>(
            reinterpret_cast<const char *>(&w_tile[brow * BK]) +
            swz(brow, (ks * 32 + lhi * 8) * 2));
        acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[nt], 0, 0, 0);
      }
    }
    __syncthreads();
    if (kt + 1 < nkt) {
      store_tile(wreg, xreg);
      __syncthreads();
    }
  }

  // epilogue: fp32 partial
  float *base = partial + (long)z * M * N;
#pragma unroll
  for (int nt = 0; nt < BN / 16; ++nt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = m0 + wave * 16 + lhi * 4 + r;
      const int n = n0 + nt * 16 + l16;
      if (m < M && n < N) base[(long)m * N + n] = acc[nt][r];
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

}  // namespace

void skinny_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                 c10::optional<torch::Tensor> bias,
                 torch::Tensor workspace, long splitk_arg) {
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(K % 8 == 0);
  const int n_wg = ((N + BN - 1) / BN) * ((M + BM - 1) / BM);
  int splitk = (int)splitk_arg;
  if (splitk <= 0) {
    splitk = 1;
    while (splitk < 16 && n_wg * splitk < 512 &&
           (K / (splitk * 2)) >= BK)
      splitk *= 2;
  }
  int k_slice = (K + splitk - 1) / splitk;
  k_slice = ((k_slice + BK - 1) / BK) * BK;
  splitk = (K + k_slice - 1) / k_slice;

  TORCH_CHECK(workspace.numel() >= (long)splitk * M * N,
              "skinny_gemm workspace too small");
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(skinny_gemm_kernel,
                     dim3((N + BN - 1) / BN, (M + BM - 1) / BM, splitk),
                     dim3(BLOCK), 0, stream,
                     workspace.data_ptr<float>(),
                     (const __hip_bfloat16 *)x.data_ptr(),
                     (const __hip_bfloat16 *)w.data_ptr(), M, N, K, k_slice);
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
                     stream, (__hip_bfloat16 *)out.data_ptr(),
                     workspace.data_ptr<float>(), bias_ptr, M, N, splitk);
  HIP_CHECK_KERNEL();
}
