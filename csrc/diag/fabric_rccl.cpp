// gpud-fabric-check — RCCL xGMI fabric diagnostic for MI355X nodes.
//
// The active fabric check the reference never had (its nccl component is
// monitor-only — reference: components/accelerator/nvidia/nccl/
// kmsg_matcher.go:12; SURVEY.md §2.4 item 4 specifies this binary):
// single-process, one RCCL communicator over all visible GPUs.
//
// xGMI on MI355X is point-to-point — 7 links x ~153 GB/s per GPU, no
// switch — so a ring all-reduce is per-link bound and ONE slow link caps
// the whole ring. The check therefore measures:
//   1. pairwise sendrecv between every GPU pair (per-link bandwidth),
//   2. full all-reduce message sweep (ring bus bandwidth),
// and prints one JSON object on stdout for the diag component to parse.
// Data is verified (all-reduce of ones must equal ndev).
//
// Build: hipcc --offload-arch=gfx950 csrc/diag/fabric_rccl.cpp -lrccl
// Run:   gpud-fabric-check [--min-bytes N] [--max-bytes N] [--pairwise]

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#define CHECK_HIP(x)                                                       \
  do {                                                                     \
    hipError_t e = (x);                                                    \
    if (e != hipSuccess) {                                                 \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e),     \
              __FILE__, __LINE__);                                         \
      exit(2);                                                             \
    }                                                                      \
  } while (0)

#define CHECK_NCCL(x)                                                      \
  do {                                                                     \
    ncclResult_t r = (x);                                                  \
    if (r != ncclSuccess) {                                                \
      fprintf(stderr, "RCCL error %s at %s:%d\n", ncclGetErrorString(r),   \
              __FILE__, __LINE__);                                         \
      exit(3);                                                             \
    }                                                                      \
  } while (0)

struct Buffers {
  float* send = nullptr;
  float* recv = nullptr;
  hipStream_t stream = nullptr;
};

static double now_s() {
  using clock = std::chrono::steady_clock;
  return std::chrono::duration<double>(clock::now().time_since_epoch())
      .count();
}

int main(int argc, char** argv) {
  size_t min_bytes = 8ull << 20;    // 8 MiB
  size_t max_bytes = 1ull << 30;    // 1 GiB
  bool do_pairwise = true;
  int iters = 5;
  for (int i = 1; i < argc; ++i) {
    if (!strcmp(argv[i], "--min-bytes") && i + 1 < argc)
      min_bytes = strtoull(argv[++i], nullptr, 10);
    else if (!strcmp(argv[i], "--max-bytes") && i + 1 < argc)
      max_bytes = strtoull(argv[++i], nullptr, 10);
    else if (!strcmp(argv[i], "--iters") && i + 1 < argc)
      iters = atoi(argv[++i]);
    else if (!strcmp(argv[i], "--no-pairwise"))
      do_pairwise = false;
  }

  int ndev = 0;
  CHECK_HIP(hipGetDeviceCount(&ndev));
  if (ndev < 1) {
    printf("{\"ok\": false, \"error\": \"no GPUs visible\"}\n");
    return 1;
  }

  std::vector<int> devs(ndev);
  for (int i = 0; i < ndev; ++i) devs[i] = i;
  std::vector<ncclComm_t> comms(ndev);
  CHECK_NCCL(ncclCommInitAll(comms.data(), ndev, devs.data()));

  const size_t max_elems = max_bytes / sizeof(float);
  std::vector<Buffers> bufs(ndev);
  for (int i = 0; i < ndev; ++i) {
    CHECK_HIP(hipSetDevice(i));
    CHECK_HIP(hipMalloc(&bufs[i].send, max_elems * sizeof(float)));
    CHECK_HIP(hipMalloc(&bufs[i].recv, max_elems * sizeof(float)));
    CHECK_HIP(hipStreamCreate(&bufs[i].stream));
    // all-reduce of ones verifies to ndev on every element
    std::vector<float> ones(1024, 1.0f);
    for (size_t off = 0; off < max_elems; off += 1024) {
      size_t n = std::min<size_t>(1024, max_elems - off);
      CHECK_HIP(hipMemcpy(bufs[i].send + off, ones.data(), n * sizeof(float),
                          hipMemcpyHostToDevice));
    }
  }

  auto sync_all = [&]() {
    for (int i = 0; i < ndev; ++i) {
      CHECK_HIP(hipSetDevice(i));
      CHECK_HIP(hipStreamSynchronize(bufs[i].stream));
    }
  };

  auto allreduce_once = [&](size_t elems) {
    CHECK_NCCL(ncclGroupStart());
    for (int i = 0; i < ndev; ++i) {
      CHECK_HIP(hipSetDevice(i));
      CHECK_NCCL(ncclAllReduce(bufs[i].send, bufs[i].recv, elems, ncclFloat,
                               ncclSum, comms[i], bufs[i].stream));
    }
    CHECK_NCCL(ncclGroupEnd());
  };

  printf("{\"ok\": true, \"ndev\": %d, \"allreduce\": [", ndev);
  bool first = true;
  bool verified = true;
  for (size_t bytes = min_bytes; bytes <= max_bytes; bytes *= 4) {
    size_t elems = bytes / sizeof(float);
    allreduce_once(elems);  // warmup
    sync_all();
    double t0 = now_s();
    for (int it = 0; it < iters; ++it) allreduce_once(elems);
    sync_all();
    double dt = (now_s() - t0) / iters;
    // ring all-reduce algorithmic bus bandwidth: 2*(n-1)/n * bytes / time
    double busbw = (ndev > 1)
                       ? 2.0 * (ndev - 1) / ndev * (double)bytes / dt / 1e9
                       : (double)bytes / dt / 1e9;
    // verification on device 0, first and last element
    float v[2] = {0, 0};
    CHECK_HIP(hipSetDevice(0));
    CHECK_HIP(hipMemcpy(&v[0], bufs[0].recv, sizeof(float),
                        hipMemcpyDeviceToHost));
    CHECK_HIP(hipMemcpy(&v[1], bufs[0].recv + elems - 1, sizeof(float),
                        hipMemcpyDeviceToHost));
    bool ok = (v[0] == (float)ndev) && (v[1] == (float)ndev);
    verified = verified && ok;
    printf("%s{\"bytes\": %zu, \"seconds\": %.6f, \"busbw_gbps\": %.2f, "
           "\"verified\": %s}",
           first ? "" : ", ", bytes, dt, busbw, ok ? "true" : "false");
    first = false;
  }
  printf("], ");

  // pairwise sendrecv: isolates each point-to-point path (a single slow
  // xGMI link shows up here before it caps the ring)
  printf("\"pairwise\": [");
  if (do_pairwise && ndev > 1) {
    size_t bytes = std::min<size_t>(max_bytes, 256ull << 20);
    size_t elems = bytes / sizeof(float);
    bool pfirst = true;
    for (int a = 0; a < ndev; ++a) {
      for (int b = a + 1; b < ndev; ++b) {
        auto pair_once = [&]() {
          CHECK_NCCL(ncclGroupStart());
          CHECK_HIP(hipSetDevice(a));
          CHECK_NCCL(ncclSend(bufs[a].send, elems, ncclFloat, b, comms[a],
                              bufs[a].stream));
          CHECK_NCCL(ncclRecv(bufs[a].recv, elems, ncclFloat, b, comms[a],
                              bufs[a].stream));
          CHECK_HIP(hipSetDevice(b));
          CHECK_NCCL(ncclSend(bufs[b].send, elems, ncclFloat, a, comms[b],
                              bufs[b].stream));
          CHECK_NCCL(ncclRecv(bufs[b].recv, elems, ncclFloat, a, comms[b],
                              bufs[b].stream));
          CHECK_NCCL(ncclGroupEnd());
        };
        pair_once();  // warmup
        sync_all();
        double t0 = now_s();
        for (int it = 0; it < iters; ++it) pair_once();
        sync_all();
        double dt = (now_s() - t0) / iters;
        // bidirectional: bytes each way simultaneously
        double gbps = (double)bytes / dt / 1e9;
        printf("%s{\"a\": %d, \"b\": %d, \"bytes\": %zu, "
               "\"bidir_gbps_per_dir\": %.2f}",
               pfirst ? "" : ", ", a, b, bytes, gbps);
        pfirst = false;
      }
    }
  }
  printf("], \"verified\": %s}\n", verified ? "true" : "false");

  for (int i = 0; i < ndev; ++i) {
    CHECK_HIP(hipSetDevice(i));
    // best-effort teardown on exit — failures here cannot change the
    // verdict already printed
    (void)hipFree(bufs[i].send);
    (void)hipFree(bufs[i].recv);
    (void)hipStreamDestroy(bufs[i].stream);
    ncclCommDestroy(comms[i]);
  }
  return verified ? 0 : 4;
}
