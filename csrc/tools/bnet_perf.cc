// bnet_perf — native point-to-point benchmark + example client for the
// ncclNetPlugin_v6 vtable (no python overhead; the true transport ceiling).
//
//   bnet_perf loop [sizes]   # fork both roles over loopback: one-way GB/s
//                            # per size, 16-deep pipelined window
//   bnet_perf lat  [sizes]   # fork ping-pong: RTT p50/p90/p99/min per size
//                            # (BNET_PERF_ITERS, default 2000)
//
// BNET_PERF_BYTES bounds the bytes per size in loop mode (default 1 GiB).

#include <arpa/inet.h>
#include <string.h>
#include <sys/wait.h>
#include <unistd.h>

#include <algorithm>
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <string>
#include <vector>

#include "baguanet/nccl_abi.h"

extern "C" ncclNet_v6_t ncclNetPlugin_v6;

namespace {

constexpr int kDepth = 16;
ncclNet_v6_t* net = &ncclNetPlugin_v6;

double now_s() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

#define CHECK(x)                                              \
  do {                                                        \
    ncclResult_t rc_ = (x);                                   \
    if (rc_ != ncclSuccess) {                                 \
      fprintf(stderr, "%s failed: %d\n", #x, (int)rc_);       \
      exit(1);                                                \
    }                                                         \
  } while (0)

void wait_req(void* req) {
  int done = 0, sz = 0;
  while (!done) CHECK(net->test(req, &done, &sz));
}

std::vector<size_t> parse_sizes(int argc, char** argv, int start) {
  std::vector<size_t> sizes;
  for (int i = start; i < argc; i++) sizes.push_back(strtoull(argv[i], 0, 0));
  if (sizes.empty())
    sizes = {4096, 65536, 262144, 1 << 20, 4 << 20};
  return sizes;
}

void run_receiver(void* lcomm, const std::vector<size_t>& sizes,
                  size_t bytes_per_size, int out_fd) {
  void* rcomm = nullptr;
  while (!rcomm) CHECK(net->accept(lcomm, &rcomm));
  void* mh = nullptr;
  CHECK(net->regMr(rcomm, nullptr, 0, NCCL_PTR_HOST, &mh));
  for (size_t size : sizes) {
    size_t n_msgs = std::max<size_t>(8, bytes_per_size / std::max(size, 1ul));
    std::vector<std::vector<char>> bufs(kDepth,
                                        std::vector<char>(std::max(size, 1ul)));
    // tell the sender buffers are ready
    char ok = 1;
    (void)!write(out_fd, &ok, 1);
    size_t done = 0, posted = 0;
    std::vector<void*> reqs;
    while (done < n_msgs) {
      while (posted < n_msgs && reqs.size() < kDepth) {
        void* data = bufs[posted % kDepth].data();
        int sz = (int)size;
        int tag = 0;
        void* req = nullptr;
        CHECK(net->irecv(rcomm, 1, &data, &sz, &tag, &mh, &req));
        if (!req) break;
        reqs.push_back(req);
        posted++;
      }
      for (size_t i = 0; i < reqs.size();) {
        int d = 0, sz = 0;
        CHECK(net->test(reqs[i], &d, &sz));
        if (d) {
          reqs.erase(reqs.begin() + i);
          done++;
        } else {
          i++;
        }
      }
    }
    (void)!write(out_fd, &ok, 1);  // size drained
  }
  CHECK(net->closeRecv(rcomm));
}

void run_sender(void* handle, const std::vector<size_t>& sizes,
                size_t bytes_per_size, int in_fd) {
  void* scomm = nullptr;
  while (!scomm) CHECK(net->connect(0, handle, &scomm));
  void* mh = nullptr;
  CHECK(net->regMr(scomm, nullptr, 0, NCCL_PTR_HOST, &mh));
  for (size_t size : sizes) {
    size_t n_msgs = std::max<size_t>(8, bytes_per_size / std::max(size, 1ul));
    std::vector<char> buf(std::max(size, 1ul), 0x5a);
    char sync;
    (void)!read(in_fd, &sync, 1);  // receiver ready
    double t0 = now_s();
    size_t done = 0, posted = 0;
    std::vector<void*> reqs;
    while (done < n_msgs) {
      while (posted < n_msgs && reqs.size() < kDepth) {
        void* req = nullptr;
        CHECK(net->isend(scomm, buf.data(), (int)size, 0, mh, &req));
        if (!req) break;
        reqs.push_back(req);
        posted++;
      }
      for (size_t i = 0; i < reqs.size();) {
        int d = 0, sz = 0;
        CHECK(net->test(reqs[i], &d, &sz));
        if (d) {
          reqs.erase(reqs.begin() + i);
          done++;
        } else {
          i++;
        }
      }
    }
    (void)!read(in_fd, &sync, 1);  // receiver drained
    double dt = now_s() - t0;
    printf("%10zu B x %6zu msgs: %8.3f GB/s\n", size, n_msgs,
           n_msgs * size / dt / 1e9);
    fflush(stdout);
  }
  CHECK(net->closeSend(scomm));
}

// ---- latency (ping-pong RTT) ---------------------------------------------
// Bidirectional: each process owns one send comm and one recv comm.  The
// echo side mirrors every message; the measuring side records RTTs and
// prints percentiles.  One message in flight — measures per-message
// latency, not throughput.

void setup_duplex(void* my_handle, void* my_lcomm, int peer_handle_fd,
                  void** scomm, void** rcomm) {
  char peer[NCCL_NET_HANDLE_MAXSIZE];
  size_t got = 0;
  while (got < sizeof(peer)) {
    ssize_t n = read(peer_handle_fd, peer + got, sizeof(peer) - got);
    if (n <= 0) exit(1);
    got += (size_t)n;
  }
  (void)my_handle;
  *scomm = nullptr;
  *rcomm = nullptr;
  while (!*scomm || !*rcomm) {
    if (!*scomm) CHECK(net->connect(0, peer, scomm));
    if (!*rcomm) CHECK(net->accept(my_lcomm, rcomm));
  }
}

void run_latency(void* scomm, void* rcomm, const std::vector<size_t>& sizes,
                 int iters, bool measure) {
  void* smh = nullptr;
  void* rmh = nullptr;
  CHECK(net->regMr(scomm, nullptr, 0, NCCL_PTR_HOST, &smh));
  CHECK(net->regMr(rcomm, nullptr, 0, NCCL_PTR_HOST, &rmh));
  for (size_t size : sizes) {
    std::vector<char> sbuf(std::max(size, 1ul), 0x5a);
    std::vector<char> rbuf(std::max(size, 1ul));
    std::vector<double> rtts;
    rtts.reserve(iters);
    int warmup = iters / 10 + 8;
    for (int it = 0; it < iters + warmup; it++) {
      void* rreq = nullptr;
      void* data = rbuf.data();
      int sz = (int)size;
      int tag = 0;
      while (!rreq) CHECK(net->irecv(rcomm, 1, &data, &sz, &tag, &rmh, &rreq));
      double t0 = now_s();
      if (measure) {  // ping → wait echo
        void* sreq = nullptr;
        while (!sreq) CHECK(net->isend(scomm, sbuf.data(), (int)size, 0,
                                       smh, &sreq));
        wait_req(sreq);
        wait_req(rreq);
        if (it >= warmup) rtts.push_back(now_s() - t0);
      } else {  // echo: wait ping → reply
        wait_req(rreq);
        void* sreq = nullptr;
        while (!sreq) CHECK(net->isend(scomm, sbuf.data(), (int)size, 0,
                                       smh, &sreq));
        wait_req(sreq);
      }
    }
    if (measure) {
      std::sort(rtts.begin(), rtts.end());
      auto pct = [&](double p) {
        return rtts[std::min(rtts.size() - 1,
                             (size_t)(p * rtts.size()))] * 1e6;
      };
      printf("%10zu B x %6d iters: RTT p50 %7.1f us  p90 %7.1f us  "
             "p99 %7.1f us  min %7.1f us\n",
             size, iters, pct(0.50), pct(0.90), pct(0.99), rtts.front() * 1e6);
      fflush(stdout);
    }
  }
}

}  // namespace

int main(int argc, char** argv) {
  if (argc < 2) {
    fprintf(stderr, "usage: %s loop|lat [sizes...]\n", argv[0]);
    return 2;
  }
  CHECK(net->init(nullptr));
  size_t bytes_per_size = 1ull << 30;
  if (const char* e = getenv("BNET_PERF_BYTES"))
    bytes_per_size = strtoull(e, 0, 0);

  if (!strcmp(argv[1], "loop")) {
    auto sizes = parse_sizes(argc, argv, 2);
    char handle[NCCL_NET_HANDLE_MAXSIZE] = {};
    void* lcomm = nullptr;
    CHECK(net->listen(0, handle, &lcomm));
    int r2s[2], s2r[2];  // receiver->sender sync pipe (and unused reverse)
    if (pipe(r2s) || pipe(s2r)) return 1;
    pid_t pid = fork();
    if (pid == 0) {
      run_receiver(lcomm, sizes, bytes_per_size, r2s[1]);
      _exit(0);
    }
    run_sender(handle, sizes, bytes_per_size, r2s[0]);
    int st = 0;
    waitpid(pid, &st, 0);
    CHECK(net->closeListen(lcomm));
    return WIFEXITED(st) ? WEXITSTATUS(st) : 1;
  }
  if (!strcmp(argv[1], "lat")) {
    auto sizes = parse_sizes(argc, argv, 2);
    if (argc <= 2) sizes = {8, 4096, 65536};
    int iters = 2000;
    if (const char* e = getenv("BNET_PERF_ITERS")) iters = atoi(e);
    char handle1[NCCL_NET_HANDLE_MAXSIZE] = {};
    void* lcomm1 = nullptr;
    CHECK(net->listen(0, handle1, &lcomm1));
    int c2p[2];  // child sends its listen handle to the parent
    if (pipe(c2p)) return 1;
    pid_t pid = fork();
    if (pid == 0) {  // echo side: own listener, handle over the pipe
      char handle2[NCCL_NET_HANDLE_MAXSIZE] = {};
      void* lcomm2 = nullptr;
      CHECK(net->listen(0, handle2, &lcomm2));
      (void)!write(c2p[1], handle1, sizeof(handle1));  // unused, keeps sym
      (void)!write(c2p[1], handle2, sizeof(handle2));
      void* scomm = nullptr;
      void* rcomm = nullptr;
      // child connects to handle1, accepts on lcomm2
      scomm = nullptr;
      while (!scomm || !rcomm) {
        if (!scomm) CHECK(net->connect(0, handle1, &scomm));
        if (!rcomm) CHECK(net->accept(lcomm2, &rcomm));
      }
      run_latency(scomm, rcomm, sizes, iters, /*measure=*/false);
      CHECK(net->closeSend(scomm));
      CHECK(net->closeRecv(rcomm));
      _exit(0);
    }
    char dummy[NCCL_NET_HANDLE_MAXSIZE];
    void* scomm = nullptr;
    void* rcomm = nullptr;
    {
      size_t got = 0;
      while (got < sizeof(dummy)) {
        ssize_t n = read(c2p[0], dummy + got, sizeof(dummy) - got);
        if (n <= 0) return 1;
        got += (size_t)n;
      }
    }
    setup_duplex(handle1, lcomm1, c2p[0], &scomm, &rcomm);
    run_latency(scomm, rcomm, sizes, iters, /*measure=*/true);
    CHECK(net->closeSend(scomm));
    CHECK(net->closeRecv(rcomm));
    int st = 0;
    waitpid(pid, &st, 0);
    CHECK(net->closeListen(lcomm1));
    return WIFEXITED(st) ? WEXITSTATUS(st) : 1;
  }
  fprintf(stderr, "only 'loop' and 'lat' modes are wired up in this build\n");
  return 2;
}
