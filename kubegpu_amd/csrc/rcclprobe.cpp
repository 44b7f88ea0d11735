// rcclprobe — in-pod RCCL-over-xGMI all-reduce bandwidth probe.
//
// The verification half of the MI355X scheduler (SURVEY.md §2.2, §5
// "Distributed communication backend"): the scheduler places k-GPU pods
// on the subset with the best xGMI ring; this probe ring-all-reduces
// over the GPUs actually visible in the pod (after ROCR_VISIBLE_DEVICES
// + /dev/dri injection) and reports achieved GB/s, closing the loop
// between the topology model and measured bandwidth.  The reference has
// no analog — its topology tree is an unverified proxy ("same gpugrp0
// => fast", nvidia_gpu_manager.go:177-180).
//
// Single-process multi-GPU (ncclCommInitAll): the pod's GPU set is a
// single node's xGMI hive, which is exactly the regime where one process
// driving k devices over RCCL is the cheapest correct harness.
//
// Build: hipcc --offload-arch=gfx950 -O2 rcclprobe.cpp -lrccl -o bin/rcclprobe
//
// Output (one JSON line):
//   {"ndev": 8, "bytes": 268435456, "iters": 20, "warmup": 5,
//    "time_ms_per_iter": 3.1, "algbw_gbps": 86.6, "busbw_gbps": 151.6,
//    "dtype": "bf16"}
// busbw = algbw * 2*(n-1)/n  (ring all-reduce moves 2*(n-1)/n * bytes
// per rank over its slowest link).

#include <hip/hip_runtime.h>
#include <hip/hip_bfloat16.h>
#include <rccl/rccl.h>

#include <hip/hip_fp16.h>

#include <chrono>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#define HIPCHECK(cmd)                                                          \
  do {                                                                         \
    hipError_t e = (cmd);                                                      \
    if (e != hipSuccess) {                                                     \
      fprintf(stderr, "rcclprobe: HIP error %s at %s:%d\n",                    \
              hipGetErrorString(e), __FILE__, __LINE__);                       \
      exit(1);                                                                 \
    }                                                                          \
  } while (0)

#define NCCLCHECK(cmd)                                                         \
  do {                                                                         \
    ncclResult_t r = (cmd);                                                    \
    if (r != ncclSuccess) {                                                    \
      fprintf(stderr, "rcclprobe: RCCL error %s at %s:%d\n",                   \
              ncclGetErrorString(r), __FILE__, __LINE__);                      \
      exit(1);                                                                 \
    }                                                                          \
  } while (0)

__global__ void fill_bf16(unsigned short* p, size_t n, unsigned short bits) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) p[i] = bits;
}

static float bf16_to_float(unsigned short b) {
  unsigned int u = (unsigned int)b << 16;
  float f;
  memcpy(&f, &u, sizeof(f));
  return f;
}

int main(int argc, char** argv) {
  long long bytes = 256ll << 20;  // 256 MiB per rank
  int iters = 20, warmup = 5, ndev = -1;
  bool check = true;
  std::string devlist;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&]() -> const char* {
      if (i + 1 >= argc) {
        fprintf(stderr, "rcclprobe: missing value for %s\n", a.c_str());
        exit(2);
      }
      return argv[++i];
    };
    if (a == "--bytes") bytes = atoll(next());
    else if (a == "--iters") iters = atoi(next());
    else if (a == "--warmup") warmup = atoi(next());
    else if (a == "--ndev") ndev = atoi(next());
    else if (a == "--devices") devlist = next();
    else if (a == "--no-check") check = false;
    else {
      fprintf(stderr,
              "usage: rcclprobe [--ndev N | --devices 0,1,..] [--bytes B] "
              "[--iters I] [--warmup W]\n");
      return 2;
    }
  }

  int avail = 0;
  HIPCHECK(hipGetDeviceCount(&avail));
  std::vector<int> devs;
  if (!devlist.empty()) {
    char* s = strdup(devlist.c_str());
    for (char* tok = strtok(s, ","); tok; tok = strtok(nullptr, ","))
      devs.push_back(atoi(tok));
    free(s);
  } else {
    if (ndev < 0) ndev = avail;
    for (int i = 0; i < ndev; ++i) devs.push_back(i);
  }
  int n = (int)devs.size();
  if (n == 0 || n > avail) {
    fprintf(stderr, "rcclprobe: %d devices requested, %d available\n", n, avail);
    return 1;
  }

  size_t count = (size_t)(bytes / 2);  // bf16 elements
  std::vector<void*> send(n), recv(n);
  std::vector<hipStream_t> streams(n);
  for (int i = 0; i < n; ++i) {
    HIPCHECK(hipSetDevice(devs[i]));
    HIPCHECK(hipMalloc(&send[i], count * 2));
    HIPCHECK(hipMalloc(&recv[i], count * 2));
    // bf16 1.0 everywhere: after all-reduce(sum) every element == n,
    // exactly representable -> bitwise-checkable numerics.
    hipLaunchKernelGGL(fill_bf16, dim3(1024), dim3(256), 0, nullptr,
                       (unsigned short*)send[i], count, (unsigned short)0x3F80);
    HIPCHECK(hipDeviceSynchronize());
    HIPCHECK(hipStreamCreate(&streams[i]));
  }

  std::vector<ncclComm_t> comms(n);
  NCCLCHECK(ncclCommInitAll(comms.data(), n, devs.data()));

  auto run_iter = [&]() {
    NCCLCHECK(ncclGroupStart());
    for (int i = 0; i < n; ++i)
      NCCLCHECK(ncclAllReduce(send[i], recv[i], count, ncclBfloat16, ncclSum,
                              comms[i], streams[i]));
    NCCLCHECK(ncclGroupEnd());
  };
  auto sync_all = [&]() {
    for (int i = 0; i < n; ++i) {
      HIPCHECK(hipSetDevice(devs[i]));
      HIPCHECK(hipStreamSynchronize(streams[i]));
    }
  };

  for (int w = 0; w < warmup; ++w) run_iter();
  sync_all();

  auto t0 = std::chrono::steady_clock::now();
  for (int it = 0; it < iters; ++it) run_iter();
  sync_all();
  auto t1 = std::chrono::steady_clock::now();

  // numerics check: every element of every rank's recv == (float)n
  // (one all-reduce of ones; iters>1 reduce the same send buffer, so the
  // result is n after every iteration)
  int bad = 0;
  if (check) {
    std::vector<unsigned short> host(256);
    for (int i = 0; i < n && bad == 0; ++i) {
      HIPCHECK(hipSetDevice(devs[i]));
      HIPCHECK(hipMemcpy(host.data(), recv[i], host.size() * 2,
                         hipMemcpyDeviceToHost));
      for (unsigned short v : host)
        if (std::fabs(bf16_to_float(v) - (float)n) > 1e-3f * n) ++bad;
    }
    if (bad) {
      fprintf(stderr, "rcclprobe: NUMERICS CHECK FAILED (%d bad elements)\n", bad);
      return 3;
    }
  }

  double sec = std::chrono::duration<double>(t1 - t0).count();
  double per_iter = sec / iters;
  double algbw = (double)(count * 2) / per_iter / 1e9;  // GB/s per rank payload
  double factor = n > 1 ? 2.0 * (n - 1) / n : 1.0;
  double busbw = algbw * factor;

  printf(
      "{\"ndev\": %d, \"bytes\": %lld, \"iters\": %d, \"warmup\": %d, "
      "\"time_ms_per_iter\": %.3f, \"algbw_gbps\": %.2f, \"busbw_gbps\": %.2f, "
      "\"dtype\": \"bf16\", \"check\": \"%s\"}\n",
      n, (long long)(count * 2), iters, warmup, per_iter * 1e3, algbw, busbw,
      check ? "pass" : "skipped");

  for (int i = 0; i < n; ++i) NCCLCHECK(ncclCommDestroy(comms[i]));
  for (int i = 0; i < n; ++i) {
    HIPCHECK(hipSetDevice(devs[i]));
    HIPCHECK(hipFree(send[i]));
    HIPCHECK(hipFree(recv[i]));
    HIPCHECK(hipStreamDestroy(streams[i]));
  }
  return 0;
}
