// gpuprobe — hand-written CDNA4 (gfx950) bandwidth kernels.
//
// Used by the probe subsystem for the k=1 degenerate point of the
// BASELINE.md curve (single-GPU "all-reduce" has no interconnect: the
// meaningful sanity number is HBM3E streaming bandwidth, spec 8 TB/s,
// ≈6.3 TB/s achievable per MI355X_MICROARCH.md) and by GPU numerics
// tests.
//
// Kernel notes (per /opt/skills/guides/cdna_hip_programming.md):
//  * 256-thread blocks = 4 wave64s; 16 B/lane vectorized access
//    (uint4) -> 4 KiB per block per instruction.
//  * grid-stride with >> 256 workgroups so all 8 XCDs / 256 CUs fill.
//  * copy is the float4-copy pattern that measures ~79% of HBM peak.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#define WAVE 64
#define BLOCK 256
// Measured on MI355X (gpurun copy sweep, 1 GiB buffers): 1024 blocks
// (4 per CU) + nontemporal beats larger grids — 5816 vs 4533 GB/s at
// 4096 blocks; NT avoids LLC pollution on pure streams.
#define DEFAULT_COPY_BLOCKS 1024
// Pure write streams saturate earlier: >≈900 workgroups the write-drain
// path contends and bandwidth collapses (measured: 640 wg 6,229 GB/s,
// 1024 wg 4,300 GB/s at 1 GiB — profiles/write_bw_sweep_mi355x.json).
#define DEFAULT_WRITE_BLOCKS 640

__global__ void copy_kernel_v4(const uint4* __restrict__ src,
                               uint4* __restrict__ dst, size_t n4) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) dst[i] = src[i];
}

// Non-temporal variant: bypasses L2/LLC allocation on both sides, which
// matters for pure streaming (the 256 MiB Infinity Cache otherwise
// absorbs part of the stream and skews small-buffer numbers).
typedef unsigned int uint4v __attribute__((ext_vector_type(4)));

__global__ void copy_kernel_v4_nt(const uint4v* __restrict__ src,
                                  uint4v* __restrict__ dst, size_t n4) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) {
    uint4v v = __builtin_nontemporal_load(&src[i]);
    __builtin_nontemporal_store(v, &dst[i]);
  }
}

// 4x unrolled NT variant: four independent uint4 transfers in flight
// per thread per iteration (more memory-level parallelism per wave).
__global__ void copy_kernel_v4_nt_u4(const uint4v* __restrict__ src,
                                     uint4v* __restrict__ dst, size_t n4) {
  size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  size_t i = tid;
  size_t bound = n4 >= 3 * stride ? n4 - 3 * stride : 0;
  for (; i < bound; i += 4 * stride) {
    uint4v a = __builtin_nontemporal_load(&src[i]);
    uint4v b = __builtin_nontemporal_load(&src[i + stride]);
    uint4v c = __builtin_nontemporal_load(&src[i + 2 * stride]);
    uint4v d = __builtin_nontemporal_load(&src[i + 3 * stride]);
    __builtin_nontemporal_store(a, &dst[i]);
    __builtin_nontemporal_store(b, &dst[i + stride]);
    __builtin_nontemporal_store(c, &dst[i + 2 * stride]);
    __builtin_nontemporal_store(d, &dst[i + 3 * stride]);
  }
  for (; i < n4; i += stride) {
    uint4v v = __builtin_nontemporal_load(&src[i]);
    __builtin_nontemporal_store(v, &dst[i]);
  }
}

// Contiguous-chunk partitioning: block b owns one contiguous region
// (DRAM-page-friendly; each wave still issues coalesced 1 KiB lines).
__global__ void copy_kernel_v4_nt_chunk(const uint4v* __restrict__ src,
                                        uint4v* __restrict__ dst, size_t n4) {
  size_t per_block = (n4 + gridDim.x - 1) / gridDim.x;
  size_t begin = (size_t)blockIdx.x * per_block;
  size_t end = begin + per_block < n4 ? begin + per_block : n4;
  for (size_t i = begin + threadIdx.x; i < end; i += blockDim.x) {
    uint4v v = __builtin_nontemporal_load(&src[i]);
    __builtin_nontemporal_store(v, &dst[i]);
  }
}

// Mixed variant: regular (cache-allocating) loads + nontemporal stores.
// Answers whether letting the read stream allocate in the 256 MiB LLC
// helps or hurts a pure 1 GiB stream (no reuse => expected ~parity,
// measured to settle it).
__global__ void copy_kernel_v4_mixed(const uint4v* __restrict__ src,
                                     uint4v* __restrict__ dst, size_t n4) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) {
    uint4v v = src[i];
    __builtin_nontemporal_store(v, &dst[i]);
  }
}

__global__ void copy_kernel_b(const unsigned char* __restrict__ src,
                              unsigned char* __restrict__ dst, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = src[i];
}

// Read-only streaming: block-sum of uint4 lanes into one u64 per block
// (measures pure HBM read bandwidth; the result write is negligible).
__global__ void read_sum_kernel(const uint4* __restrict__ src, size_t n4,
                                unsigned long long* __restrict__ out) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  unsigned long long acc = 0;
  for (; i < n4; i += stride) {
    uint4 v = src[i];
    acc += (unsigned long long)v.x + v.y + v.z + v.w;
  }
  __shared__ unsigned long long sh[BLOCK / WAVE];
  // wave-level reduce via shuffles (64-wide waves on CDNA)
  for (int off = WAVE / 2; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, WAVE);
  int lane = threadIdx.x % WAVE;
  int wid = threadIdx.x / WAVE;
  if (lane == 0) sh[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned long long total = 0;
    for (int w = 0; w < BLOCK / WAVE; ++w) total += sh[w];
    atomicAdd(out, total);
  }
}

// Write-only streaming: NT fill (completes the read/write/copy triple).
__global__ void fill_kernel_v4_nt(uint4v* __restrict__ dst, size_t n4,
                                  unsigned int word) {
  uint4v v = {word, word, word, word};
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) __builtin_nontemporal_store(v, &dst[i]);
}

static void check_pair(const at::Tensor& dst, const at::Tensor& src) {
  TORCH_CHECK(src.is_cuda() && dst.is_cuda(), "gpuprobe: tensors must be on GPU");
  TORCH_CHECK(src.is_contiguous() && dst.is_contiguous(),
              "gpuprobe: tensors must be contiguous");
  TORCH_CHECK(src.nbytes() == dst.nbytes(), "gpuprobe: size mismatch");
}

void copy(at::Tensor dst, at::Tensor src) {
  check_pair(dst, src);
  size_t nbytes = src.nbytes();
  auto stream = at::hip::getCurrentHIPStream();
  if (nbytes % 16 == 0) {
    size_t n4 = nbytes / 16;
    int blocks = (int)std::min<size_t>((n4 + BLOCK - 1) / BLOCK, DEFAULT_COPY_BLOCKS);
    hipLaunchKernelGGL(copy_kernel_v4_nt, dim3(blocks), dim3(BLOCK), 0, stream,
                       (const uint4v*)src.data_ptr(), (uint4v*)dst.data_ptr(), n4);
  } else {
    int blocks = (int)std::min<size_t>((nbytes + BLOCK - 1) / BLOCK, 4096);
    hipLaunchKernelGGL(copy_kernel_b, dim3(blocks), dim3(BLOCK), 0, stream,
                       (const unsigned char*)src.data_ptr(),
                       (unsigned char*)dst.data_ptr(), nbytes);
  }
  C10_HIP_KERNEL_LAUNCH_CHECK();
}

// Timed device-to-device streaming copy; returns achieved GB/s
// (bytes read + bytes written over wall time, hipEvent-timed).
// blocks=0 picks the default; nontemporal selects the NT variant.
double copy_bw_gbps(int64_t nbytes, int64_t iters, int64_t blocks_arg,
                    bool nontemporal, int64_t variant, int64_t threads_arg) {
  TORCH_CHECK(nbytes > 0 && nbytes % 16 == 0, "nbytes must be positive, 16-aligned");
  auto opts = at::TensorOptions().dtype(at::kByte).device(at::kCUDA);
  at::Tensor src = at::empty({nbytes}, opts);
  at::Tensor dst = at::empty({nbytes}, opts);
  src.fill_(1);
  auto stream = at::hip::getCurrentHIPStream();
  size_t n4 = (size_t)nbytes / 16;
  // workgroup SIZE is a runtime launch dim (the kernels read blockDim),
  // so the wave-granularity axis can be swept without new kernels
  int threads = threads_arg > 0 ? (int)threads_arg : BLOCK;
  TORCH_CHECK(threads % WAVE == 0 && threads <= 1024, "threads: multiple of 64, <=1024");
  int blocks = blocks_arg > 0
                   ? (int)blocks_arg
                   : (int)std::min<size_t>((n4 + threads - 1) / threads, DEFAULT_COPY_BLOCKS);
  auto launch = [&]() {
    if (variant == 1)
      hipLaunchKernelGGL(copy_kernel_v4_nt_u4, dim3(blocks), dim3(threads), 0, stream,
                         (const uint4v*)src.data_ptr(), (uint4v*)dst.data_ptr(), n4);
    else if (variant == 2)
      hipLaunchKernelGGL(copy_kernel_v4_nt_chunk, dim3(blocks), dim3(threads), 0,
                         stream, (const uint4v*)src.data_ptr(),
                         (uint4v*)dst.data_ptr(), n4);
    else if (variant == 3)
      hipLaunchKernelGGL(copy_kernel_v4_mixed, dim3(blocks), dim3(threads), 0,
                         stream, (const uint4v*)src.data_ptr(),
                         (uint4v*)dst.data_ptr(), n4);
    else if (nontemporal)
      hipLaunchKernelGGL(copy_kernel_v4_nt, dim3(blocks), dim3(threads), 0, stream,
                         (const uint4v*)src.data_ptr(), (uint4v*)dst.data_ptr(), n4);
    else
      hipLaunchKernelGGL(copy_kernel_v4, dim3(blocks), dim3(threads), 0, stream,
                         (const uint4*)src.data_ptr(), (uint4*)dst.data_ptr(), n4);
  };
  for (int w = 0; w < 3; ++w) launch();
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, stream);
  for (int64_t i = 0; i < iters; ++i) launch();
  (void)hipEventRecord(t1, stream);
  (void)hipEventSynchronize(t1);
  float ms = 0.f;
  (void)hipEventElapsedTime(&ms, t0, t1);
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  C10_HIP_KERNEL_LAUNCH_CHECK();
  TORCH_CHECK(at::equal(dst, src), "copy_bw_gbps: variant ", variant,
              " produced wrong output");
  double sec = ms / 1e3;
  return (double)nbytes * 2.0 * iters / sec / 1e9;
}

double write_bw_gbps(int64_t nbytes, int64_t iters, int64_t blocks_arg) {
  TORCH_CHECK(nbytes > 0 && nbytes % 16 == 0, "nbytes must be positive, 16-aligned");
  auto opts = at::TensorOptions().dtype(at::kByte).device(at::kCUDA);
  at::Tensor dst = at::empty({nbytes}, opts);
  auto stream = at::hip::getCurrentHIPStream();
  size_t n4 = (size_t)nbytes / 16;
  int blocks = blocks_arg > 0
                   ? (int)blocks_arg
                   : (int)std::min<size_t>((n4 + BLOCK - 1) / BLOCK, DEFAULT_WRITE_BLOCKS);
  auto launch = [&]() {
    hipLaunchKernelGGL(fill_kernel_v4_nt, dim3(blocks), dim3(BLOCK), 0, stream,
                       (uint4v*)dst.data_ptr(), n4, 0x01010101u);
  };
  for (int w = 0; w < 3; ++w) launch();
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, stream);
  for (int64_t i = 0; i < iters; ++i) launch();
  (void)hipEventRecord(t1, stream);
  (void)hipEventSynchronize(t1);
  float ms = 0.f;
  (void)hipEventElapsedTime(&ms, t0, t1);
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  C10_HIP_KERNEL_LAUNCH_CHECK();
  TORCH_CHECK((int64_t)dst.sum(at::kLong).item<int64_t>() == nbytes,
              "write_bw_gbps: fill produced wrong output");
  return (double)nbytes * iters / (ms / 1e3) / 1e9;
}

double read_bw_gbps(int64_t nbytes, int64_t iters, int64_t blocks_arg) {
  TORCH_CHECK(nbytes > 0 && nbytes % 16 == 0, "nbytes must be positive, 16-aligned");
  auto opts = at::TensorOptions().dtype(at::kByte).device(at::kCUDA);
  at::Tensor src = at::empty({nbytes}, opts);
  src.fill_(1);
  at::Tensor out = at::zeros({1}, at::TensorOptions().dtype(at::kLong).device(at::kCUDA));
  auto stream = at::hip::getCurrentHIPStream();
  size_t n4 = (size_t)nbytes / 16;
  // Reads peak at 1024 wg (4 waves/CU): 6,115 GB/s vs 5,768 at 4096
  // (profiles/read_bw_sweep_mi355x.json)
  int blocks = blocks_arg > 0
                   ? (int)blocks_arg
                   : (int)std::min<size_t>((n4 + BLOCK - 1) / BLOCK, DEFAULT_COPY_BLOCKS);
  for (int w = 0; w < 3; ++w)
    hipLaunchKernelGGL(read_sum_kernel, dim3(blocks), dim3(BLOCK), 0, stream,
                       (const uint4*)src.data_ptr(), n4,
                       (unsigned long long*)out.data_ptr());
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, stream);
  for (int64_t i = 0; i < iters; ++i)
    hipLaunchKernelGGL(read_sum_kernel, dim3(blocks), dim3(BLOCK), 0, stream,
                       (const uint4*)src.data_ptr(), n4,
                       (unsigned long long*)out.data_ptr());
  (void)hipEventRecord(t1, stream);
  (void)hipEventSynchronize(t1);
  float ms = 0.f;
  (void)hipEventElapsedTime(&ms, t0, t1);
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  C10_HIP_KERNEL_LAUNCH_CHECK();
  double sec = ms / 1e3;
  return (double)nbytes * iters / sec / 1e9;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  // The timed probes release the GIL: a multi-second hipEvent-timed
  // block must not starve other Python threads (the node agent serves
  // gRPC from the same process — measured: with the GIL held, every
  // concurrent RPC hit DEADLINE_EXCEEDED during a probe block).
  m.def("copy", &copy, "streaming uint4 copy kernel (dst, src)",
        py::call_guard<py::gil_scoped_release>());
  m.def("copy_bw_gbps", &copy_bw_gbps, "timed d2d copy bandwidth",
        py::call_guard<py::gil_scoped_release>(),
        py::arg("nbytes"), py::arg("iters") = 20, py::arg("blocks") = 0,
        py::arg("nontemporal") = true, py::arg("variant") = 0,
        py::arg("threads") = 0);
  m.def("read_bw_gbps", &read_bw_gbps, "timed read bandwidth",
        py::call_guard<py::gil_scoped_release>(),
        py::arg("nbytes"), py::arg("iters") = 20, py::arg("blocks") = 0);
  m.def("write_bw_gbps", &write_bw_gbps, "timed write-only (NT fill) bandwidth",
        py::call_guard<py::gil_scoped_release>(),
        py::arg("nbytes"), py::arg("iters") = 20, py::arg("blocks") = 0);
}
