// rccl-cell-probe: native RCCL all-reduce probe over a scheduler-placed cell.
//
// Single process, one RCCL communicator per visible GPU (ncclCommInitAll),
// all-reduce over xGMI. Invoked by the scheduler after binding an affinity
// group, with HIP_VISIBLE_DEVICES restricted to the cell's GPU indices;
// measures algbw/busbw and prints one JSON line. A 2-GPU cell probing far
// below one xGMI link's ~153 GB/s means a degraded link => the pair cell is
// marked bad (this component replaces the reference's
// bind-and-hope, reference pkg/scheduler/scheduler.go:594-627).
//
// Build: hipcc -O2 --offload-arch=gfx950 native/rccl_cell_probe.cpp \
//        -I/opt/rocm/include -L/opt/rocm/lib -lrccl -o native/rccl-cell-probe
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#define HIP_CHECK(cmd)                                                              \
  do {                                                                              \
    hipError_t e = (cmd);                                                           \
    if (e != hipSuccess) {                                                          \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e), __FILE__,    \
              __LINE__);                                                            \
      exit(2);                                                                      \
    }                                                                               \
  } while (0)

#define NCCL_CHECK(cmd)                                                             \
  do {                                                                              \
    ncclResult_t r = (cmd);                                                         \
    if (r != ncclSuccess) {                                                         \
      fprintf(stderr, "RCCL error %s at %s:%d\n", ncclGetErrorString(r), __FILE__,  \
              __LINE__);                                                            \
      exit(3);                                                                      \
    }                                                                               \
  } while (0)

int main(int argc, char** argv) {
  long sizeMB = 64;
  long iters = 20;
  long warmup = 5;
  for (int i = 1; i < argc - 1; i++) {
    if (!strcmp(argv[i], "--size-mb")) sizeMB = atol(argv[i + 1]);
    if (!strcmp(argv[i], "--iters")) iters = atol(argv[i + 1]);
    if (!strcmp(argv[i], "--warmup")) warmup = atol(argv[i + 1]);
  }
  int ndev = 0;
  HIP_CHECK(hipGetDeviceCount(&ndev));
  if (ndev < 1) {
    fprintf(stderr, "no visible GPUs\n");
    return 4;
  }
  size_t count = (size_t)sizeMB * 1024 * 1024 / sizeof(float);
  std::vector<int> devs(ndev);
  for (int i = 0; i < ndev; i++) devs[i] = i;
  std::vector<ncclComm_t> comms(ndev);
  NCCL_CHECK(ncclCommInitAll(comms.data(), ndev, devs.data()));

  std::vector<float*> sendbuf(ndev), recvbuf(ndev);
  std::vector<hipStream_t> streams(ndev);
  for (int i = 0; i < ndev; i++) {
    HIP_CHECK(hipSetDevice(i));
    HIP_CHECK(hipMalloc(&sendbuf[i], count * sizeof(float)));
    HIP_CHECK(hipMalloc(&recvbuf[i], count * sizeof(float)));
    HIP_CHECK(hipMemset(sendbuf[i], 1, count * sizeof(float)));
    HIP_CHECK(hipStreamCreate(&streams[i]));
  }
  auto runOnce = [&]() {
    NCCL_CHECK(ncclGroupStart());
    for (int i = 0; i < ndev; i++) {
      NCCL_CHECK(ncclAllReduce(sendbuf[i], recvbuf[i], count, ncclFloat, ncclSum, comms[i],
                               streams[i]));
    }
    NCCL_CHECK(ncclGroupEnd());
  };
  auto syncAll = [&]() {
    for (int i = 0; i < ndev; i++) {
      HIP_CHECK(hipSetDevice(i));
      HIP_CHECK(hipStreamSynchronize(streams[i]));
    }
  };
  for (long w = 0; w < warmup; w++) runOnce();
  syncAll();
  auto t0 = std::chrono::steady_clock::now();
  for (long it = 0; it < iters; it++) runOnce();
  syncAll();
  auto t1 = std::chrono::steady_clock::now();
  double sec = std::chrono::duration<double>(t1 - t0).count() / iters;
  double bytes = (double)count * sizeof(float);
  double algbw = bytes / sec / 1e9;
  double busbw = ndev > 1 ? algbw * 2.0 * (ndev - 1) / ndev : algbw;
  printf(
      "{\"ndev\": %d, \"size_mb\": %ld, \"iters\": %ld, \"ms\": %.4f, "
      "\"algbw_gbps\": %.2f, \"busbw_gbps\": %.2f}\n",
      ndev, sizeMB, iters, sec * 1e3, algbw, busbw);
  for (int i = 0; i < ndev; i++) {
    HIP_CHECK(hipSetDevice(i));
    HIP_CHECK(hipFree(sendbuf[i]));
    HIP_CHECK(hipFree(recvbuf[i]));
    HIP_CHECK(hipStreamDestroy(streams[i]));
    ncclCommDestroy(comms[i]);
  }
  return 0;
}
