// Fused-MoE kernel family for gfx950: device-side token alignment,
// grouped MFMA GEMM over sorted (token, expert) pairs, weighted combine.
//
// Reference semantics: moe_align_block_size (_custom_ops.py:424-451),
// fused_moe_kernel (fused_moe_triton/fused_moe.py:210), moe_sum
// (_custom_ops.py:393-421). MI355X-native: everything stays on device
// (no per-layer host sync), so MoE decode steps are hipGraph-capturable;
// tile sizes picked per batch regime by the host (BLOCK_M 16 for sparse
// decode routing, 64 for dense prefill blocks), MFMA 16x16x32 bf16 with
// LDS-tiled A/W operands.
//
// Pipeline per MoE layer:
//   moe_align(topk_ids)           -> sorted_ids, expert_blocks, n_post[1]
//   moe_gemm mode 0 (gather x)    -> inter1 [rows_pad, 2I]   (sorted)
//   silu_and_mul                  -> inter_act [rows_pad, I]
//   moe_gemm mode 1 (scatter y*w) -> pair_out [T*K, H]
//   moe_sum                       -> out [T, H]

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 mfma_bf8;
typedef __attribute__((ext_vector_type(4))) float mfma_f4;

namespace {

// ---------------------------------------------------------------- align
// One workgroup: histogram -> padded cumsum -> scatter. E_local <= 1024.
constexpr int ALIGN_BLOCK = 1024;
constexpr int MAX_E = 1024;

__global__ __launch_bounds__(ALIGN_BLOCK) void moe_align_kernel(
    const int *__restrict__ ids,  // [N] global expert id per (token,k) pair
    int N, int E_local, int expert_start, int BM,
    int *__restrict__ sorted_ids,     // [cap] pair indices, N = pad sentinel
    int *__restrict__ expert_blocks,  // [max_blocks] local expert per block
    int *__restrict__ n_post_pad) {   // [1]
  __shared__ int cnt[MAX_E];
  __shared__ int off[MAX_E];
  __shared__ int fill[MAX_E];
  const int tid = threadIdx.x;
  for (int e = tid; e < E_local; e += ALIGN_BLOCK) cnt[e] = 0;
  __syncthreads();
  for (int i = tid; i < N; i += ALIGN_BLOCK) {
    const int e = ids[i] - expert_start;
    if (0 <= e && e < E_local) atomicAdd(&cnt[e], 1);
  }
  __syncthreads();
  if (tid == 0) {
    int total_blocks = 0;
    for (int e = 0; e < E_local; ++e) {
      off[e] = total_blocks * BM;
      fill[e] = 0;
      const int nb = (cnt[e] + BM - 1) / BM;
      for (int b = 0; b < nb; ++b) expert_blocks[total_blocks + b] = e;
      total_blocks += nb;
    }
    n_post_pad[0] = total_blocks * BM;
  }
  __syncthreads();
  const int total = n_post_pad[0];
  for (int i = tid; i < total; i += ALIGN_BLOCK) sorted_ids[i] = N;
  __syncthreads();
  for (int i = tid; i < N; i += ALIGN_BLOCK) {
    const int e = ids[i] - expert_start;
    if (0 <= e && e < E_local) {
      const int pos = off[e] + atomicAdd(&fill[e], 1);
      sorted_ids[pos] = i;
    }
  }
}

// ---------------------------------------------------------------- gemm
// C[m, n] = sum_k A[row(m), k] * W[e, n, k]  (W row-major over k).
// GATHER (gemm1): row(m) = sorted_ids[g]/topk into x; C row = g (sorted).
// SCATTER (gemm2): row(m) = g (sorted inter buffer); C row =
//   sorted_ids[g] (pair index), scaled by topk_w[pair].
constexpr int BK = 64;
constexpr int GEMM_BLOCK = 256;

// BN is a template parameter: 64 for sparse decode routing, 256 for
// dense prefill blocks (the wide tile quarters the A-tile re-reads
// that dominate prefill: A is re-staged once per n-tile of the grid)
template <int BM, int BN, bool SCATTER>
__global__ __launch_bounds__(GEMM_BLOCK) void moe_gemm_kernel(
    __hip_bfloat16 *__restrict__ C,
    const __hip_bfloat16 *__restrict__ A,
    const __hip_bfloat16 *__restrict__ W,  // [E_local, Nd, K]
    const int *__restrict__ sorted_ids, const int *__restrict__ expert_blocks,
    const int *__restrict__ n_post_pad,
    const float *__restrict__ topk_w,  // [n_pairs] (SCATTER only, may be 0)
    int n_pairs, int K, int Nd, int topk) {
  const int mb = blockIdx.x;
  if (mb * BM >= n_post_pad[0]) return;
  const int nb = blockIdx.y;
  const int e = expert_blocks[mb];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l16 = lane & 15;
  const int lhi = lane >> 4;

  // wave tiling over the BM x BN block: (BM/16) x (BN/16) tiles over
  // 4 waves; wave owns one m sub-tile row and BN*WM/64 n tiles
  constexpr int WM = BM / 16;            // m sub-tiles (1, 2 or 4)
  constexpr int WN_TILES = BN * WM / 64; // 16-wide n tiles per wave
  const int wm = wave % WM;
  const int wn = wave / WM;              // n-tile group (4/WM groups)

  constexpr int APAD = 8;
  __shared__ __hip_bfloat16 a_tile[2][BM * (BK + APAD)];
  __shared__ __hip_bfloat16 w_tile[2][BN * (BK + APAD)];

  // per-row source/destination ids
  const int g0 = mb * BM;

  // ---- staging lambdas: 16-B chunks across the block ----
  // A tile: BM rows x BK cols; W tile: BN rows x BK cols
  constexpr int A_CH = BM * BK / 8;         // 16-B chunks
  constexpr int W_CH = BN * BK / 8;
  shortx8 areg[(A_CH + GEMM_BLOCK - 1) / GEMM_BLOCK];
  shortx8 wreg[W_CH / GEMM_BLOCK];

  const long w_base = (long)e * Nd * K;

  auto load_tiles = [&](int k0) {
#pragma unroll
    for (int it = 0; it < (A_CH + GEMM_BLOCK - 1) / GEMM_BLOCK; ++it) {
      const int idx = tid + it * GEMM_BLOCK;
      if (idx < A_CH) {
        const int m = idx / (BK / 8);
        const int c = idx % (BK / 8);
        const int g = g0 + m;
        const int pair = sorted_ids[g];
        long arow;
        bool valid = pair < n_pairs;
        if (SCATTER) arow = g;           // sorted-order inter buffer
        else arow = valid ? pair / topk : 0;
        if (valid && k0 + c * 8 < K)
          areg[it] = *reinterpret_cast<const shortx8 *>(
              A + arow * (long)K + k0 + c * 8);
        else
          areg[it] = shortx8{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
#pragma unroll
    for (int it = 0; it < W_CH / GEMM_BLOCK; ++it) {
      const int idx = tid + it * GEMM_BLOCK;
      const int n = idx / (BK / 8);
      const int c = idx % (BK / 8);
      if (nb * BN + n < Nd && k0 + c * 8 < K) {
        const shortx8 *wp = reinterpret_cast<const shortx8 *>(
            W + w_base + (long)(nb * BN + n) * K + k0 + c * 8);
        // decode tiles (BM<=32, ~1 m-block per expert) stream each W
        // row exactly once: non-temporal keeps the small A tile L2-
        // resident (guide "nt-weights"); prefill re-reads W -> cached
        if constexpr (BM <= 32)
          wreg[it] = __builtin_nontemporal_load(wp);
        else
          wreg[it] = *wp;
      } else
        wreg[it] = shortx8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  };
  auto write_tiles = [&](int buf) {
#pragma unroll
    for (int it = 0; it < (A_CH + GEMM_BLOCK - 1) / GEMM_BLOCK; ++it) {
      const int idx = tid + it * GEMM_BLOCK;
      if (idx < A_CH) {
        const int m = idx / (BK / 8);
        const int c = idx % (BK / 8);
        *reinterpret_cast<shortx8 *>(&a_tile[buf][m * (BK + APAD) + c * 8]) =
            areg[it];
      }
    }
#pragma unroll
    for (int it = 0; it < W_CH / GEMM_BLOCK; ++it) {
      const int idx = tid + it * GEMM_BLOCK;
      const int n = idx / (BK / 8);
      const int c = idx % (BK / 8);
      *reinterpret_cast<shortx8 *>(&w_tile[buf][n * (BK + APAD) + c * 8]) =
          wreg[it];
    }
  };

  mfma_f4 acc[WN_TILES];
#pragma unroll
  for (int t = 0; t < WN_TILES; ++t) acc[t] = mfma_f4{0, 0, 0, 0};

  load_tiles(0);
  write_tiles(0);
  __syncthreads();

  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += BK) {
    if (k0 + BK < K) load_tiles(k0 + BK);

    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      mfma_bf8 afrag = *reinterpret_cast<const mfma_bf8 *>(
          &a_tile[cur][(wm * 16 + l16) * (BK + APAD) + ks * 32 + lhi * 8]);
#pragma unroll
      for (int t = 0; t < WN_TILES; ++t) {
        mfma_bf8 wfrag = *reinterpret_cast<const mfma_bf8 *>(
            &w_tile[cur][((wn * WN_TILES + t) * 16 + l16) * (BK + APAD) +
                         ks * 32 + lhi * 8]);
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, wfrag, acc[t], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    __syncthreads();
    if (k0 + BK < K) {
      write_tiles(cur ^ 1);
      cur ^= 1;
      __syncthreads();
    }
  }

  // ---- epilogue: C rows lhi*4+r of the wave's m tile ----
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
      const int n = nb * BN + (wn * WN_TILES + t) * 16 + l16;
      if (n < Nd)
        C[crow * (long)Nd + n] = __float2bfloat16(acc[t][r] * scalew);
    }
  }
}

// ---------------------------------------------------------------- sum
// out[t, :] = sum_k pair_out[t*K + k, :]
template <typename T>
__global__ void moe_sum_kernel(T *__restrict__ out,
                               const T *__restrict__ pair_out, long TH,
                               int topk, int H) {
  const int nvec = H / 8;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x;
       idx < TH * nvec; idx += (long)gridDim.x * blockDim.x) {
    const long t = idx / nvec;
    const int c = idx % nvec;
    float accv[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int k = 0; k < topk; ++k) {
      shortx8 v = reinterpret_cast<const shortx8 *>(
          pair_out + (t * topk + k) * (long)H)[c];
      float f[8];
      unpack8<T>(v, f);
#pragma unroll
      for (int j = 0; j < 8; ++j) accv[j] += f[j];
    }
    reinterpret_cast<shortx8 *>(out + t * (long)H)[c] = pack8<T>(accv);
  }
}

}  // namespace

void moe_align(torch::Tensor topk_ids, long E_local, long expert_start,
               long block_m, torch::Tensor sorted_ids,
               torch::Tensor expert_blocks, torch::Tensor n_post_pad) {
  const int N = topk_ids.numel();
  TORCH_CHECK(topk_ids.scalar_type() == at::kInt);
  TORCH_CHECK(topk_ids.is_contiguous());
  TORCH_CHECK(E_local <= MAX_E);
  TORCH_CHECK(sorted_ids.numel() >=
              N + E_local * (block_m - 1) + 1);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(moe_align_kernel, dim3(1), dim3(ALIGN_BLOCK), 0, stream,
                     topk_ids.data_ptr<int>(), N, (int)E_local,
                     (int)expert_start, (int)block_m,
                     sorted_ids.data_ptr<int>(),
                     expert_blocks.data_ptr<int>(),
                     n_post_pad.data_ptr<int>());
  HIP_CHECK_KERNEL();
}

void moe_gemm(torch::Tensor C, torch::Tensor A, torch::Tensor W,
              torch::Tensor sorted_ids, torch::Tensor expert_blocks,
              torch::Tensor n_post_pad,
              c10::optional<torch::Tensor> topk_weights, long n_pairs,
              long topk, long block_m, bool scatter) {
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 &&
              W.scalar_type() == at::kBFloat16);
  TORCH_CHECK(A.is_contiguous() && W.is_contiguous() && C.is_contiguous());
  const int K = W.size(2);
  const int Nd = W.size(1);
  TORCH_CHECK(A.size(-1) == K);
  TORCH_CHECK(K % 8 == 0, "moe_gemm: K must be a multiple of 8");
  const int max_blocks = expert_blocks.numel();
  const float *tw = nullptr;
  if (topk_weights.has_value()) {
    TORCH_CHECK(topk_weights->scalar_type() == at::kFloat);
    tw = topk_weights->data_ptr<float>();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
#define LAUNCH(BM, BNV, SC)                                                 \
  hipLaunchKernelGGL((moe_gemm_kernel<BM, BNV, SC>),                        \
                     dim3(max_blocks, (Nd + BNV - 1) / BNV),                \
                     dim3(GEMM_BLOCK), 0, stream,                           \
                     (__hip_bfloat16 *)C.data_ptr(),                        \
                     (const __hip_bfloat16 *)A.data_ptr(),                  \
                     (const __hip_bfloat16 *)W.data_ptr(),                  \
                     sorted_ids.data_ptr<int>(),                            \
                     expert_blocks.data_ptr<int>(),                         \
                     n_post_pad.data_ptr<int>(), tw, (int)n_pairs, K, Nd,   \
                     (int)topk)
  if (block_m == 16) {
    if (scatter) LAUNCH(16, 64, true); else LAUNCH(16, 64, false);
  } else if (block_m == 32) {
    if (scatter) LAUNCH(32, 64, true); else LAUNCH(32, 64, false);
  } else if (block_m == 64) {
    if (scatter) LAUNCH(64, 64, true); else LAUNCH(64, 64, false);
  } else if (block_m == 164) {  // BM=64, BN=256 (dense prefill)
    if (scatter) LAUNCH(64, 256, true); else LAUNCH(64, 256, false);
  } else {
    TORCH_CHECK(false, "moe_gemm: block_m must be 16/32/64/164");
  }
#undef LAUNCH
  HIP_CHECK_KERNEL();
}

void moe_sum(torch::Tensor out, torch::Tensor pair_out, long topk) {
  TORCH_CHECK(out.scalar_type() == at::kBFloat16);
  TORCH_CHECK(out.is_contiguous() && pair_out.is_contiguous());
  const long T = out.size(0);
  const int H = out.size(-1);
  TORCH_CHECK(H % 8 == 0);
  const long total = T * (H / 8);
  const int block = 256;
  const long grid = std::min<long>((total + block - 1) / block, 4096);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL((moe_sum_kernel<__hip_bfloat16>), dim3(grid),
                     dim3(block), 0, stream,
                     (__hip_bfloat16 *)out.data_ptr(),
                     (const __hip_bfloat16 *)pair_out.data_ptr(), T,
                     (int)topk, H);
  HIP_CHECK_KERNEL();
}
