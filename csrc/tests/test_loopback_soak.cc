// Full-transport in-process loopback soak: listen/connect/accept +
// randomized verified host traffic through the REAL engine (epoll or
// io_uring per BNET_IMPLEMENT), exactly as RCCL's proxy drives the
// plugin.  Compiled twice: plain (wired into `make native-test`) and
// under ThreadSanitizer (`make tsan-test`) — the TSan build is the
// race-detection pass over the whole engine (claiming, parking,
// header matching, slot reuse), not just the claim protocol.
//
// argv[1]: number of messages (default 20000).

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <ctime>
#include <vector>

#include "../net/transport.h"

using namespace baguanet;

static uint64_t mono_ms() {
  timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return (uint64_t)ts.tv_sec * 1000 + ts.tv_nsec / 1'000'000;
}

#define CHECK(x)                                                     \
  do {                                                               \
    if (!(x)) {                                                      \
      fprintf(stderr, "CHECK failed at %s:%d: %s\n", __FILE__,       \
              __LINE__, #x);                                         \
      return 1;                                                      \
    }                                                                \
  } while (0)

int main(int argc, char** argv) {
  const uint64_t kMsgs =
      argc > 1 ? strtoull(argv[1], nullptr, 10) : 20'000;
  setenv("NCCL_SOCKET_IFNAME", "lo", 0);
  setenv("BNET_MIN_CHUNKSIZE", "8192", 0);
  setenv("BNET_NSTREAMS", "4", 0);

  Net& net = Net::get();
  CHECK(net.ndev() >= 1);

  char handle[128] = {};
  void* lcomm = nullptr;
  CHECK(net.listen(0, handle, &lcomm) == ncclSuccess && lcomm);
  void* scomm = nullptr;
  void* rcomm = nullptr;
  uint64_t t0 = mono_ms();
  while (!scomm || !rcomm) {
    if (!scomm) CHECK(net.connect(0, handle, &scomm) == ncclSuccess);
    if (!rcomm) CHECK(net.accept(lcomm, &rcomm) == ncclSuccess);
    CHECK(mono_ms() - t0 < 10'000);
  }

  const uint32_t sizes[] = {0, 1, 64, 5000, 65536, 300'000, 1u << 20};
  struct Msg {
    std::vector<char> sbuf, rbuf;
    void* sreq = nullptr;
    void* rreq = nullptr;
    bool sdone = false, rdone = false;
    uint32_t size = 0;
  };
  std::vector<Msg*> live;
  srand(42);
  uint64_t completed = 0, posted = 0, bytes = 0;
  uint64_t last_progress = mono_ms();
  uint64_t start = mono_ms();

  while (completed < kMsgs) {
    // keep up to 10 messages in flight
    while (live.size() < 10 && posted < kMsgs) {
      Msg* m = new Msg();
      m->size = sizes[rand() % 7];
      m->sbuf.resize(m->size ? m->size : 1);
      m->rbuf.resize(m->size + 1);
      for (uint32_t i = 0; i < m->size; i++)
        m->sbuf[i] = (char)((posted * 31 + i) & 0xff);
      void* data = m->rbuf.data();
      int sz = (int)m->size;
      int tag = 0;
      void* mh = nullptr;
      CHECK(net.irecv(rcomm, 1, &data, &sz, &tag, &mh, &m->rreq) ==
            ncclSuccess);
      if (!m->rreq) {  // slot backpressure — retry after polling
        delete m;
        break;
      }
      CHECK(net.isend(scomm, m->sbuf.data(), (int)m->size, 0, nullptr,
                      &m->sreq) == ncclSuccess);
      while (!m->sreq) {  // refused: poll completions, retry
        for (Msg* o : live) {
          int done = 0, got = 0;
          if (!o->sdone && net.test(o->sreq, &done, &got) == ncclSuccess)
            o->sdone = done;
        }
        CHECK(net.isend(scomm, m->sbuf.data(), (int)m->size, 0, nullptr,
                        &m->sreq) == ncclSuccess);
        CHECK(mono_ms() - last_progress < 30'000);
      }
      posted++;
      live.push_back(m);
    }
    for (size_t i = 0; i < live.size();) {
      Msg* m = live[i];
      int done = 0, got = 0;
      if (!m->sdone) {
        CHECK(net.test(m->sreq, &done, &got) == ncclSuccess);
        m->sdone = done;
      }
      if (!m->rdone) {
        done = 0;
        CHECK(net.test(m->rreq, &done, &got) == ncclSuccess);
        if (done) {
          CHECK((uint32_t)got == m->size);
          m->rdone = true;
        }
      }
      if (m->sdone && m->rdone) {
        CHECK(memcmp(m->sbuf.data(), m->rbuf.data(), m->size) == 0);
        bytes += m->size;
        delete m;
        live[i] = live.back();
        live.pop_back();
        completed++;
        last_progress = mono_ms();
      } else {
        i++;
      }
    }
    CHECK(mono_ms() - last_progress < 30'000);
  }

  double dt = (mono_ms() - start) / 1000.0;
  printf("loopback soak ok: %lu messages, %.2f GB in %.1fs, verified\n",
         (unsigned long)completed, bytes / 1e9, dt);
  CHECK(net.close_send(scomm) == ncclSuccess);
  CHECK(net.close_recv(rcomm) == ncclSuccess);
  CHECK(net.close_listen(lcomm) == ncclSuccess);
  return 0;
}
