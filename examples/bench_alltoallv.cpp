// A/B benchmark for the all-pairs exchange matrix on one node
// (MI355X-native analog of the reference's bin/bench_alltoallv.cu, which
// compared a cudaMemcpyPeerAsync mesh against MPI_Alltoallv): which
// transport should move an 8-GPU halo-exchange matrix over xGMI?
//
//   kernel : direct-store grid-stride copy kernels launched on the SOURCE
//            device writing through peer-mapped pointers (the engine's
//            translate-job default)
//   mesh   : hipMemcpyPeerAsync per pair (SDMA copy engines)
//   rccl   : single-process RCCL (ncclCommInitAll) grouped send/recv
//
//   build/bench_alltoallv [bytes_per_pair=16777216] [iters=20] [ngpus=all]
//
// CSV: mode,ngpus,bytes_per_pair,pairs,ms,GB_s_total (aggregate goodput
// over all pairs). On 1 GPU the matrix is empty and only a self-copy
// sanity row is emitted.
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <string>
#include <vector>

#define CHECK(x)                                                                                   \
  do {                                                                                             \
    hipError_t e_ = (x);                                                                           \
    if (e_ != hipSuccess) {                                                                        \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e_), __FILE__, __LINE__);       \
      exit(1);                                                                                     \
    }                                                                                              \
  } while (0)
#define NCCLCHECK(x)                                                                               \
  do {                                                                                             \
    ncclResult_t r_ = (x);                                                                         \
    if (r_ != ncclSuccess) {                                                                       \
      fprintf(stderr, "RCCL error %s at %s:%d\n", ncclGetErrorString(r_), __FILE__, __LINE__);     \
      exit(1);                                                                                     \
    }                                                                                              \
  } while (0)

__global__ void copy_kernel(const char *__restrict__ src, char *__restrict__ dst, int64_t words) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t w = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; w < words; w += stride)
    reinterpret_cast<uint4 *>(dst)[w] = reinterpret_cast<const uint4 *>(src)[w];
}

int main(int argc, char **argv) {
  const int64_t bytes = argc > 1 ? atoll(argv[1]) : 16 * 1024 * 1024;
  const int iters = argc > 2 ? atoi(argv[2]) : 20;
  int ndev = 0;
  CHECK(hipGetDeviceCount(&ndev));
  if (argc > 3) ndev = std::min(ndev, atoi(argv[3]));
  const int64_t words = bytes / 16;

  // per device: one send buffer per peer + one recv buffer per peer
  std::vector<std::vector<char *>> sbuf(ndev, std::vector<char *>(ndev, nullptr));
  std::vector<std::vector<char *>> rbuf(ndev, std::vector<char *>(ndev, nullptr));
  std::vector<hipStream_t> stream(ndev);
  for (int d = 0; d < ndev; ++d) {
    CHECK(hipSetDevice(d));
    CHECK(hipStreamCreateWithFlags(&stream[d], hipStreamNonBlocking));
    for (int p = 0; p < ndev; ++p) {
      CHECK(hipMalloc((void **)&sbuf[d][p], bytes));
      CHECK(hipMalloc((void **)&rbuf[d][p], bytes));
      CHECK(hipMemset(sbuf[d][p], 1 + d, bytes));
    }
    for (int p = 0; p < ndev; ++p) {
      if (p == d) continue;
      hipError_t e = hipDeviceEnablePeerAccess(p, 0);
      if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled) CHECK(e);
      (void)hipGetLastError();
    }
  }
  const int pairs = ndev * (ndev - 1);
  const double totalGB = (double)bytes * (pairs ? pairs : 1) / 1e9;

  auto sync_all = [&]() {
    for (int d = 0; d < ndev; ++d) {
      CHECK(hipSetDevice(d));
      CHECK(hipStreamSynchronize(stream[d]));
    }
  };
  auto bench = [&](const char *mode, auto &&run) {
    run(); // warmup
    sync_all();
    const auto t0 = std::chrono::steady_clock::now();
    for (int it = 0; it < iters; ++it) run();
    sync_all();
    const std::chrono::duration<double> dt = std::chrono::steady_clock::now() - t0;
    printf("%s,%d,%lld,%d,%.3f,%.1f\n", mode, ndev, (long long)bytes, pairs,
           dt.count() / iters * 1e3, totalGB * iters / dt.count());
    fflush(stdout);
  };

  printf("mode,ngpus,bytes_per_pair,pairs,ms,GB_s_total\n");

  if (pairs == 0) { // 1 GPU: self-copy sanity only
    bench("kernel_self", [&]() {
      CHECK(hipSetDevice(0));
      hipLaunchKernelGGL(copy_kernel, dim3(1024), dim3(256), 0, stream[0], sbuf[0][0], rbuf[0][0],
                         words);
    });
    return 0;
  }

  // direct-store kernels: launched on the source device, dst is the
  // peer-mapped pointer (xGMI stores; all 7 links of each GPU in flight)
  bench("kernel", [&]() {
    for (int d = 0; d < ndev; ++d) {
      CHECK(hipSetDevice(d));
      for (int p = 0; p < ndev; ++p) {
        if (p == d) continue;
        hipLaunchKernelGGL(copy_kernel, dim3(256), dim3(256), 0, stream[d], sbuf[d][p],
                           rbuf[p][d], words);
      }
    }
  });

  // SDMA mesh
  bench("mesh", [&]() {
    for (int d = 0; d < ndev; ++d) {
      CHECK(hipSetDevice(d));
      for (int p = 0; p < ndev; ++p) {
        if (p == d) continue;
        CHECK(hipMemcpyPeerAsync(rbuf[p][d], p, sbuf[d][p], d, bytes, stream[d]));
      }
    }
  });

  // single-process RCCL: one comm per device, grouped send/recv
  std::vector<ncclComm_t> comms(ndev);
  std::vector<int> devs(ndev);
  for (int d = 0; d < ndev; ++d) devs[d] = d;
  NCCLCHECK(ncclCommInitAll(comms.data(), ndev, devs.data()));
  bench("rccl", [&]() {
    NCCLCHECK(ncclGroupStart());
    for (int d = 0; d < ndev; ++d) {
      CHECK(hipSetDevice(d));
      for (int p = 0; p < ndev; ++p) {
        if (p == d) continue;
        NCCLCHECK(ncclSend(sbuf[d][p], bytes, ncclChar, p, comms[d], stream[d]));
        NCCLCHECK(ncclRecv(rbuf[d][p], bytes, ncclChar, p, comms[d], stream[d]));
      }
    }
    NCCLCHECK(ncclGroupEnd());
  });
  for (int d = 0; d < ndev; ++d) NCCLCHECK(ncclCommDestroy(comms[d]));
  return 0;
}
