// Sampling-path kernels. Reference: layers/repetition_penalty.py
// (_scaling_penalty_kernel Triton) — per-row scaling penalty against a
// persistent uint8 seen-token mask pool (core/penalty.py). Memory
// bound: [B, V] fp32 logits read+write plus [B, V] uint8 mask reads;
// a single pass fused with the slot gather.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

namespace {

__global__ void rep_penalty_kernel(float* __restrict__ logits,
                                   const unsigned char* __restrict__ pool,
                                   long pool_stride,
                                   const long* __restrict__ slots,
                                   const float* __restrict__ penalties,
                                   int V) {
  int b = blockIdx.y;
  long slot = slots[b];
  float p = penalties[b];
  if (slot < 0 || p == 1.f) return;
  const unsigned char* mask = pool + slot * pool_stride;
  float* row = logits + (long)b * V;
  float inv = 1.f / p;
  int i = (blockIdx.x * blockDim.x + threadIdx.x) * 4;
  int stride = gridDim.x * blockDim.x * 4;
  for (; i < V; i += stride) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int k = i + j;
      if (k < V && mask[k]) {
        float v = row[k];
        row[k] = v > 0.f ? v * inv : v * p;
      }
    }
  }
}

}  // namespace

// ------------------------------------------------------------------
// Sorting-free fused top-k / top-p / min-p filter (reference:
// sgl-kernel top_k_top_p_sampling_from_probs, layers/sampler.py:8-19).
//
// Per row: a 3-level radix select over the fp32 probability bits (10
// bits per level, count + mass histograms in LDS) finds the EXACT
// combined threshold — the scan walks bins from the top and stops when
// the kept count reaches top_k or the kept mass reaches top_p,
// reproducing the sequential "top-k then top-p on the k-kept prefix"
// semantics of the torch path. Boundary ties are all kept (the torch
// sort keeps the lowest-index ties first — a measure-zero difference
// on real logits). min_p folds in as max(threshold, min_p * row_max).
// 4 streaming passes of [V] replace the full [B, V] descending sort.
// Graph-safe: no host reads.

namespace {

constexpr int TKP_BINS = 1024;
constexpr int TKP_BLOCK = 256;

__global__ __launch_bounds__(TKP_BLOCK) void topk_topp_filter_kernel(
    float *__restrict__ probs, const int *__restrict__ top_ks,
    const float *__restrict__ top_ps, const float *__restrict__ min_ps,
    int V) {
  const int row = blockIdx.x;
  int k = top_ks[row];
  float P = top_ps[row];
  const float mp = min_ps ? min_ps[row] : 0.f;
  const bool need_k = (k > 0 && k < V);
  const bool need_p = (P < 1.0f);
  if (!need_k && !need_p && mp <= 0.f) return;
  if (!need_k) k = V + 1;
  if (!need_p) P = INFINITY;

  float *p = probs + (long)row * V;
  const int tid = threadIdx.x;

  __shared__ int cnt[TKP_BINS];
  __shared__ float mass[TKP_BINS];
  __shared__ float s_red[TKP_BLOCK / 64];
  __shared__ float s_max;
  __shared__ unsigned s_prefix;
  __shared__ int s_cnt_above;
  __shared__ float s_mass_above, s_kept_mass;
  __shared__ int s_done;  // 1 = no bin triggered (keep everything)

  if (tid == 0) {
    s_prefix = 0;
    s_cnt_above = 0;
    s_mass_above = 0.f;
    s_done = 0;
  }

  float rmax = 0.f;
  // ---- 3 radix levels: bits [20,30) [10,20) [0,10) ----
#pragma unroll
  for (int lv = 0; lv < 3; ++lv) {
    const int shift = 20 - 10 * lv;
    for (int i = tid; i < TKP_BINS; i += TKP_BLOCK) {
      cnt[i] = 0;
      mass[i] = 0.f;
    }
    __syncthreads();
    const unsigned prefix = s_prefix;
    if (s_done) break;
    for (int i = tid; i < V; i += TKP_BLOCK) {
      const float v = p[i];
      if (lv == 0) rmax = fmaxf(rmax, v);
      const unsigned key = __float_as_uint(v);
      if (lv == 0 || (key >> (shift + 10)) == prefix) {
        const unsigned bin = (key >> shift) & (TKP_BINS - 1);
        atomicAdd(&cnt[bin], 1);
        atomicAdd(&mass[bin], v);
      }
    }
    if (lv == 0) {
      // block-reduce the row max
      for (int off = 32; off > 0; off >>= 1)
        rmax = fmaxf(rmax, __shfl_xor(rmax, off, 64));
      if ((tid & 63) == 0) s_red[tid >> 6] = rmax;
    }
    __syncthreads();
    if (tid == 0) {
      if (lv == 0) {
        float m = s_red[0];
        for (int w = 1; w < TKP_BLOCK / 64; ++w) m = fmaxf(m, s_red[w]);
        s_max = m;
      }
      int ca = s_cnt_above;
      float ma = s_mass_above;
      int chosen = -1;
      for (int b = TKP_BINS - 1; b >= 0; --b) {
        const int c2 = ca + cnt[b];
        const float m2 = ma + mass[b];
        if (c2 >= k || m2 >= P) {
          chosen = b;
          if (lv == 2) s_kept_mass = m2;  // ties all kept
          break;
        }
        ca = c2;
        ma = m2;
      }
      if (chosen < 0) {
        s_done = 1;           // neither bound binds: keep everything
        s_kept_mass = ma;
      } else {
        s_cnt_above = ca;
        s_mass_above = ma;
        s_prefix = (s_prefix << 10) | (unsigned)chosen;
      }
    }
    __syncthreads();
  }

  float thresh = s_done ? 0.f : __uint_as_float(s_prefix);
  float kept_mass = s_kept_mass;

  // ---- min_p: threshold on the renormalized row max ----
  if (mp > 0.f) {
    const float t2 = mp * s_max;
    if (t2 > thresh) {
      thresh = t2;
      // recompute kept mass above the raised threshold
      float acc = 0.f;
      for (int i = tid; i < V; i += TKP_BLOCK) {
        const float v = p[i];
        if (v >= t2) acc += v;
      }
      for (int off = 32; off > 0; off >>= 1)
        acc += __shfl_xor(acc, off, 64);
      if ((tid & 63) == 0) s_red[tid >> 6] = acc;
      __syncthreads();
      if (tid == 0) {
        float m = 0.f;
        for (int w = 0; w < TKP_BLOCK / 64; ++w) m += s_red[w];
        s_kept_mass = m;
      }
      __syncthreads();
      kept_mass = s_kept_mass;
    }
  }

  // ---- filter + renormalize ----
  const float inv = (kept_mass > 0.f) ? 1.f / kept_mass : 0.f;
  for (int i = tid; i < V; i += TKP_BLOCK) {
    const float v = p[i];
    p[i] = (v >= thresh) ? v * inv : 0.f;
  }
}

}  // namespace

void topk_topp_filter(torch::Tensor probs, torch::Tensor top_ks,
                      torch::Tensor top_ps,
                      c10::optional<torch::Tensor> min_ps) {
  TORCH_CHECK(probs.is_cuda() && probs.dtype() == at::kFloat &&
              probs.is_contiguous());
  TORCH_CHECK(top_ks.dtype() == at::kInt && top_ps.dtype() == at::kFloat);
  const int B = probs.size(0), V = probs.size(1);
  const float *mp = nullptr;
  if (min_ps.has_value()) {
    TORCH_CHECK(min_ps->dtype() == at::kFloat);
    mp = min_ps->data_ptr<float>();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  topk_topp_filter_kernel<<<dim3(B), dim3(TKP_BLOCK), 0, stream>>>(
      probs.data_ptr<float>(), top_ks.data_ptr<int>(),
      top_ps.data_ptr<float>(), mp, V);
}

void apply_repetition_penalty(torch::Tensor logits, torch::Tensor pool,
                              torch::Tensor slots,
                              torch::Tensor penalties) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == at::kFloat &&
              logits.is_contiguous());
  TORCH_CHECK(pool.dtype() == at::kByte && pool.is_contiguous());
  TORCH_CHECK(slots.dtype() == at::kLong);
  int B = logits.size(0), V = logits.size(1);
  TORCH_CHECK(pool.size(1) >= V, "mask pool narrower than vocab");
  auto stream = at::cuda::getCurrentCUDAStream();
  dim3 grid((V + 4 * 256 - 1) / (4 * 256), B);
  rep_penalty_kernel<<<grid, 256, 0, stream>>>(
      logits.data_ptr<float>(), pool.data_ptr<unsigned char>(),
      pool.stride(0), slots.data_ptr<long>(),
      penalties.data_ptr<float>(), V);
}
