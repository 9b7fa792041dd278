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

// Per-pair p2p copy bandwidth (GB/s, worse of the two directions) between
// two visible devices: localizes a low collective busbw to ONE xGMI link
// (every GPU pair on an MI355X node is directly connected).
static double p2pPairGbps(int a, int b, long sizeMB, long iters) {
  const long bytes = sizeMB * 1024 * 1024;
  void* bufA = nullptr;
  void* bufB = nullptr;
  HIP_CHECK(hipSetDevice(a));
  HIP_CHECK(hipMalloc(&bufA, bytes));
  HIP_CHECK(hipSetDevice(b));
  HIP_CHECK(hipMalloc(&bufB, bytes));
  int can = 0;
  HIP_CHECK(hipDeviceCanAccessPeer(&can, b, a));
  if (can) {
    hipError_t e = hipDeviceEnablePeerAccess(a, 0);
    if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled) HIP_CHECK(e);
    HIP_CHECK(hipSetDevice(a));
    e = hipDeviceEnablePeerAccess(b, 0);
    if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled) HIP_CHECK(e);
  }
  double worst = 1e30;
  for (int dir = 0; dir < 2; dir++) {
    int src = dir == 0 ? a : b;
    int dst = dir == 0 ? b : a;
    void* s = dir == 0 ? bufA : bufB;
    void* d = dir == 0 ? bufB : bufA;
    HIP_CHECK(hipSetDevice(src));
    hipStream_t stream;
    HIP_CHECK(hipStreamCreate(&stream));
    HIP_CHECK(hipMemcpyPeerAsync(d, dst, s, src, bytes, stream));  // warmup
    hipEvent_t t0, t1;
    HIP_CHECK(hipEventCreate(&t0));
    HIP_CHECK(hipEventCreate(&t1));
    HIP_CHECK(hipEventRecord(t0, stream));
    for (long it = 0; it < iters; it++) {
      HIP_CHECK(hipMemcpyPeerAsync(d, dst, s, src, bytes, stream));
    }
    HIP_CHECK(hipEventRecord(t1, stream));
    HIP_CHECK(hipEventSynchronize(t1));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    double gbps = (double)bytes * iters / (ms / 1e3) / 1e9;
    if (gbps < worst) worst = gbps;
    HIP_CHECK(hipEventDestroy(t0));
    HIP_CHECK(hipEventDestroy(t1));
    HIP_CHECK(hipStreamDestroy(stream));
  }
  HIP_CHECK(hipSetDevice(a));
  HIP_CHECK(hipFree(bufA));
  HIP_CHECK(hipSetDevice(b));
  HIP_CHECK(hipFree(bufB));
  return worst;
}

int main(int argc, char** argv) {
  long sizeMB = 64;
  long iters = 20;
  long warmup = 5;
  bool p2pMatrix = false;
  for (int i = 1; i < argc; i++) {
    if (!strcmp(argv[i], "--p2p-matrix")) p2pMatrix = true;
    if (i >= argc - 1) continue;
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
  // ndev==1 "busbw" is a device-local HBM copy, not an xGMI number: flag it
  // so consumers label it hbm_copy_gbps (see BENCHMARKS.md).
  std::string matrixJson;
  if (p2pMatrix && ndev > 1) {
    matrixJson = ", \"p2p_matrix\": {";
    bool first = true;
    char buf[64];
    for (int i = 0; i < ndev; i++) {
      for (int j = i + 1; j < ndev; j++) {
        double g = p2pPairGbps(i, j, sizeMB, iters);
        snprintf(buf, sizeof(buf), "%s\"%d-%d\": %.2f", first ? "" : ", ", i, j, g);
        matrixJson += buf;
        first = false;
      }
    }
    matrixJson += "}";
  }
  printf(
      "{\"ndev\": %d, \"size_mb\": %ld, \"iters\": %ld, \"ms\": %.4f, "
      "\"algbw_gbps\": %.2f, \"busbw_gbps\": %.2f, \"hbm_copy\": %s%s}\n",
      ndev, sizeMB, iters, sec * 1e3, algbw, busbw, ndev > 1 ? "false" : "true",
      matrixJson.c_str());
  for (int i = 0; i < ndev; i++) {
    HIP_CHECK(hipSetDevice(i));
    HIP_CHECK(hipFree(sendbuf[i]));
    HIP_CHECK(hipFree(recvbuf[i]));
    HIP_CHECK(hipStreamDestroy(streams[i]));
    ncclCommDestroy(comms[i]);
  }
  return 0;
}
