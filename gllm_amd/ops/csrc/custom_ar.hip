// Custom all-reduce over hipIpc-mapped peer buffers (xGMI direct
// loads). Reference role: sgl_kernel.allreduce two-shot NVLink AR
// (custom_all_reduce.py) — the MI355X replacement
// reads peers straight over point-to-point xGMI links instead of a
// per-link-bound RCCL ring, worth it for the small decode-size TP
// messages (<= a few MB).
//
// Round-1 shape: ONE-SHOT — every rank copies its input into its own
// shared buffer, a flag barrier makes all buffers globally visible,
// then each rank reduces ALL peers' full buffers locally (for world<=8
// and decode-size payloads the redundant reads are cheaper than a
// second synchronization round). Two-shot (reduce-scatter + gather)
// and hipGraph-captured registration are the round-2 upgrades
// (ROADMAP.md). Off by default; enable with GLLM_CUSTOM_AR=1
// (parallel/custom_all_reduce.py gates eligibility).
//
// Buffer layout per rank: [signal: 64 x uint32 (256 B)] [data bytes].
// The barrier uses a monotonically increasing epoch (host-tracked), so
// buffers never need resetting between calls.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <vector>

#define CAR_MAX_WORLD 8
#define CAR_SIGNAL_BYTES 256

namespace {

struct PeerPtrs {
  void* p[CAR_MAX_WORLD];
};

using bf16 = __hip_bfloat16;

__global__ void car_barrier_kernel(PeerPtrs sig, int rank, int world,
                                   unsigned int epoch) {
  // thread t posts my arrival into peer t's signal slot [rank], then
  // waits for every peer's arrival in MY slot [t].
  int t = threadIdx.x;
  __threadfence_system();  // make the copy-in globally visible first
  if (t < world) {
    volatile unsigned int* peer_sig =
        reinterpret_cast<volatile unsigned int*>(sig.p[t]);
    __atomic_store_n(const_cast<unsigned int*>(&peer_sig[rank]), epoch,
                     __ATOMIC_RELEASE);
  }
  if (t < world) {
    volatile unsigned int* my_sig =
        reinterpret_cast<volatile unsigned int*>(sig.p[rank]);
    while (__atomic_load_n(const_cast<unsigned int*>(&my_sig[t]),
                           __ATOMIC_ACQUIRE) < epoch) {
    }
  }
  __threadfence_system();
}

template <typename T>
__global__ void car_copy_in_kernel(const T* __restrict__ src,
                                   T* __restrict__ dst, size_t n) {
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = src[i];
}

// out[i] = sum over ranks of data_r[i]; fp32 accumulation for bf16.
template <typename T>
__global__ void car_reduce_kernel(PeerPtrs data, T* __restrict__ out,
                                  size_t n, int world) {
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float acc = 0.f;
#pragma unroll
    for (int r = 0; r < CAR_MAX_WORLD; ++r) {
      if (r < world) {
        acc += (float)reinterpret_cast<const T*>(data.p[r])[i];
      }
    }
    out[i] = (T)acc;
  }
}

}  // namespace

// ---------------------------------------------------------------- API
// Allocate the shared buffer; returns (device_ptr, ipc_handle bytes).
std::pair<int64_t, py::bytes> car_alloc(int64_t data_bytes) {
  void* ptr = nullptr;
  size_t total = CAR_SIGNAL_BYTES + (size_t)data_bytes;
  C10_CUDA_CHECK(hipMalloc(&ptr, total));
  C10_CUDA_CHECK(hipMemset(ptr, 0, total));
  hipIpcMemHandle_t handle;
  C10_CUDA_CHECK(hipIpcGetMemHandle(&handle, ptr));
  return {reinterpret_cast<int64_t>(ptr),
          py::bytes(reinterpret_cast<const char*>(&handle),
                    sizeof(handle))};
}

// v2 layout: [signal 256][epoch cell 256][data max_bytes][out slices]
std::pair<int64_t, py::bytes> car_alloc_v2(int64_t max_bytes) {
  void* ptr = nullptr;
  size_t total = 2 * CAR_SIGNAL_BYTES + (size_t)max_bytes +
                 (size_t)max_bytes / 2 + 256;
  C10_CUDA_CHECK(hipMalloc(&ptr, total));
  C10_CUDA_CHECK(hipMemset(ptr, 0, total));
  hipIpcMemHandle_t handle;
  C10_CUDA_CHECK(hipIpcGetMemHandle(&handle, ptr));
  return {reinterpret_cast<int64_t>(ptr),
          py::bytes(reinterpret_cast<const char*>(&handle),
                    sizeof(handle))};
}

int64_t car_open(py::bytes handle_bytes) {
  std::string s = handle_bytes;
  TORCH_CHECK(s.size() == sizeof(hipIpcMemHandle_t), "bad handle size");
  hipIpcMemHandle_t handle;
  memcpy(&handle, s.data(), sizeof(handle));
  void* ptr = nullptr;
  C10_CUDA_CHECK(hipIpcOpenMemHandle(&ptr, handle,
                                     hipIpcMemLazyEnablePeerAccess));
  return reinterpret_cast<int64_t>(ptr);
}

void car_close(int64_t ptr) { (void)hipIpcCloseMemHandle((void*)ptr); }

// Non-owning bf16 tensor view over raw device memory (the disagg
// GPU-direct slot pool reads/writes through this).
torch::Tensor car_view_tensor(int64_t ptr, int64_t numel) {
  auto opts = torch::TensorOptions()
                  .dtype(at::kBFloat16)
                  .device(at::kCUDA, at::cuda::current_device());
  return torch::from_blob(reinterpret_cast<void*>(ptr), {numel}, opts);
}
void car_free(int64_t ptr) { (void)hipFree((void*)ptr); }

void car_all_reduce(torch::Tensor inout, std::vector<int64_t> ptrs,
                    int64_t rank, int64_t world, int64_t epoch) {
  TORCH_CHECK(inout.is_cuda() && inout.is_contiguous());
  TORCH_CHECK(world >= 2 && world <= CAR_MAX_WORLD);
  TORCH_CHECK((int64_t)ptrs.size() == world);
  size_t n = inout.numel();
  auto stream = at::cuda::getCurrentCUDAStream();

  PeerPtrs sig, data;
  for (int r = 0; r < world; ++r) {
    char* base = reinterpret_cast<char*>(ptrs[r]);
    sig.p[r] = base;
    data.p[r] = base + CAR_SIGNAL_BYTES;
  }
  int threads = 256;
  int blocks = (int)std::min<size_t>(512, (n + threads - 1) / threads);

  if (inout.scalar_type() == at::kBFloat16) {
    car_copy_in_kernel<bf16><<<blocks, threads, 0, stream>>>(
        reinterpret_cast<const bf16*>(inout.data_ptr()),
        reinterpret_cast<bf16*>(data.p[rank]), n);
    car_barrier_kernel<<<1, CAR_MAX_WORLD, 0, stream>>>(
        sig, (int)rank, (int)world, (unsigned int)epoch);
    car_reduce_kernel<bf16><<<blocks, threads, 0, stream>>>(
        data, reinterpret_cast<bf16*>(inout.data_ptr()), n, (int)world);
  } else if (inout.scalar_type() == at::kFloat) {
    car_copy_in_kernel<float><<<blocks, threads, 0, stream>>>(
        inout.data_ptr<float>(), reinterpret_cast<float*>(data.p[rank]),
        n);
    car_barrier_kernel<<<1, CAR_MAX_WORLD, 0, stream>>>(
        sig, (int)rank, (int)world, (unsigned int)epoch);
    car_reduce_kernel<float><<<blocks, threads, 0, stream>>>(
        data, inout.data_ptr<float>(), n, (int)world);
  } else {
    TORCH_CHECK(false, "custom AR supports bf16/fp32");
  }
  // a second barrier before the NEXT call's copy-in is unnecessary:
  // each rank only rewrites its OWN data region, and the next call's
  // barrier (greater epoch) orders it against peers' reads because
  // reads happen between the two barriers on every rank's stream.
  car_barrier_kernel<<<1, CAR_MAX_WORLD, 0, stream>>>(
      sig, (int)rank, (int)world, (unsigned int)(epoch + 1));
}

// ================================================================ v2
// Device-epoch + two-shot. The epoch lives in a device cell so the
// whole AR is hipGraph-capturable: every rank calls the collective the
// same number of times, so the cells stay lock-step without any host
// value baked into the capture (reference custom_all_reduce.py:266-391
// solves this with a graph-buffer registration handshake; a device
// counter needs none).
//
// Buffer layout per rank: [signal 256 B][epoch cell 256 B]
//                         [data max_bytes][out max_bytes/world region]
// One-shot (small n): copy-in -> barrier -> full local reduce ->
//   tail barrier (2 epochs/call).
// Two-shot (large n): copy-in -> barrier -> reduce OWN slice across
//   peers into own out region -> barrier -> gather peers' out slices
//   -> tail barrier (3 epochs/call). xGMI traffic per rank drops from
//   (world-1)*n to ~2n/world*(world-1) reads.

namespace {

__global__ void car_barrier_dev_kernel(PeerPtrs sig, int rank, int world,
                                       unsigned int* epoch_cell,
                                       int bump) {
  const unsigned int target = *epoch_cell + (unsigned int)bump;
  int t = threadIdx.x;
  __threadfence_system();
  if (t < world) {
    volatile unsigned int* peer_sig =
        reinterpret_cast<volatile unsigned int*>(sig.p[t]);
    __atomic_store_n(const_cast<unsigned int*>(&peer_sig[rank]), target,
                     __ATOMIC_RELEASE);
    volatile unsigned int* my_sig =
        reinterpret_cast<volatile unsigned int*>(sig.p[rank]);
    while (__atomic_load_n(const_cast<unsigned int*>(&my_sig[t]),
                           __ATOMIC_ACQUIRE) < target) {
    }
  }
  __threadfence_system();
}

__global__ void car_epoch_bump_kernel(unsigned int* epoch_cell, int by) {
  if (threadIdx.x == 0) *epoch_cell += (unsigned int)by;
}

// rank's slice reduced across peers into its own out region
template <typename T>
__global__ void car_rs_kernel(PeerPtrs data, T* __restrict__ own_out,
                              size_t slice_lo, size_t slice_n,
                              int world) {
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < slice_n; i += stride) {
    float acc = 0.f;
#pragma unroll
    for (int r = 0; r < CAR_MAX_WORLD; ++r) {
      if (r < world)
        acc += (float)reinterpret_cast<const T*>(data.p[r])[slice_lo + i];
    }
    own_out[i] = (T)acc;
  }
}

// gather every rank's out slice into the result tensor
template <typename T>
__global__ void car_ag_kernel(PeerPtrs outs, T* __restrict__ res,
                              size_t n, size_t slice, int world) {
  size_t i = blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int r = (int)(i / slice);
    res[i] = reinterpret_cast<const T*>(outs.p[r])[i - r * slice];
  }
}

}  // namespace

// ptrs: per-rank base pointers; max_bytes: the data region size used at
// allocation time (out region starts at signal+epoch+max_bytes).
void car_all_reduce_v2(torch::Tensor inout, std::vector<int64_t> ptrs,
                       int64_t rank, int64_t world, int64_t max_bytes) {
  TORCH_CHECK(inout.is_cuda() && inout.is_contiguous());
  TORCH_CHECK(world >= 2 && world <= CAR_MAX_WORLD);
  TORCH_CHECK((int64_t)ptrs.size() == world);
  const size_t n = inout.numel();
  const size_t esize = inout.element_size();
  TORCH_CHECK((int64_t)(n * esize) <= max_bytes, "AR payload > buffer");
  auto stream = at::cuda::getCurrentCUDAStream();

  PeerPtrs sig, data, outs;
  for (int r = 0; r < world; ++r) {
    char* base = reinterpret_cast<char*>(ptrs[r]);
    sig.p[r] = base;
    data.p[r] = base + 2 * CAR_SIGNAL_BYTES;
    outs.p[r] = base + 2 * CAR_SIGNAL_BYTES + max_bytes;
  }
  unsigned int* cell = reinterpret_cast<unsigned int*>(
      reinterpret_cast<char*>(ptrs[rank]) + CAR_SIGNAL_BYTES);
  const int threads = 256;
  const int blocks = (int)std::min<size_t>(512, (n + threads - 1) / threads);
  // two-shot pays off once the redundant one-shot reads dominate the
  // extra barrier: ~256 KB crossover on xGMI
  const bool twoshot = n * esize >= (size_t)256 * 1024 && n % world == 0;

#define CAR_DISPATCH(T, TPTR)                                               \
  do {                                                                      \
    car_copy_in_kernel<T><<<blocks, threads, 0, stream>>>(                  \
        reinterpret_cast<const T*>(TPTR), reinterpret_cast<T*>(             \
            data.p[rank]), n);                                              \
    car_barrier_dev_kernel<<<1, CAR_MAX_WORLD, 0, stream>>>(                \
        sig, (int)rank, (int)world, cell, 1);                               \
    if (twoshot) {                                                          \
      const size_t slice = n / world;                                       \
      car_rs_kernel<T><<<blocks, threads, 0, stream>>>(                     \
          data, reinterpret_cast<T*>(outs.p[rank]), rank * slice, slice,    \
          (int)world);                                                      \
      car_barrier_dev_kernel<<<1, CAR_MAX_WORLD, 0, stream>>>(              \
          sig, (int)rank, (int)world, cell, 2);                             \
      car_ag_kernel<T><<<blocks, threads, 0, stream>>>(                     \
          outs, reinterpret_cast<T*>(TPTR), n, slice, (int)world);          \
      car_barrier_dev_kernel<<<1, CAR_MAX_WORLD, 0, stream>>>(              \
          sig, (int)rank, (int)world, cell, 3);                             \
      car_epoch_bump_kernel<<<1, 64, 0, stream>>>(cell, 3);                 \
    } else {                                                                \
      car_reduce_kernel<T><<<blocks, threads, 0, stream>>>(                 \
          data, reinterpret_cast<T*>(TPTR), n, (int)world);                 \
      car_barrier_dev_kernel<<<1, CAR_MAX_WORLD, 0, stream>>>(              \
          sig, (int)rank, (int)world, cell, 2);                             \
      car_epoch_bump_kernel<<<1, 64, 0, stream>>>(cell, 2);                 \
    }                                                                       \
  } while (0)

  if (inout.scalar_type() == at::kBFloat16) {
    CAR_DISPATCH(bf16, inout.data_ptr());
  } else if (inout.scalar_type() == at::kFloat) {
    CAR_DISPATCH(float, inout.data_ptr());
  } else {
    TORCH_CHECK(false, "custom AR supports bf16/fp32");
  }
#undef CAR_DISPATCH
}
