// Minimal native repro of the claim/slot protocol: 4 spinning "socket"
// threads claim chunks and instantly account them; a proxy thread posts
// isends (mixed sizes incl. 0) and test/frees, mimicking net.cc exactly.
// A stall here = pure claim/slot-machine bug (no sockets, no wakeups).

#include <atomic>
#include <cstdio>
#include <cstdlib>
#include <thread>
#include <vector>

#include "../net/transport.h"

using namespace baguanet;

int main(int argc, char** argv) {
  // argv[1]: message target (default 30M; TSan runs use a smaller count)
  const uint64_t kTarget = argc > 1 ? strtoull(argv[1], nullptr, 10)
                                    : 30'000'000;
  SendComm c;
  std::vector<TcpSock*> socks;
  for (int i = 0; i < 4; i++) {
    auto* s = new TcpSock();
    s->idx = i;
    s->scomm = &c;
    c.socks.push_back(s);
  }
  std::atomic<bool> stop{false};
  std::atomic<uint64_t> claims{0};

  // spinning claimers: claim → instantly "send" the chunk
  std::vector<std::thread> workers;
  for (int i = 0; i < 4; i++) {
    workers.emplace_back([&, i] {
      TcpSock* s = c.socks[i];
      while (!stop.load(std::memory_order_relaxed)) {
        uint32_t off = 0, len = 0;
        SendRequest* r = claim_chunk(&c, s->idx, &off, &len);
        if (!r) continue;
        claims.fetch_add(1, std::memory_order_relaxed);
        if (r->total == 0) {
          r->hdr_sent.store(true, std::memory_order_release);
        } else {
          r->sent.fetch_add(len, std::memory_order_acq_rel);
        }
      }
    });
  }

  // proxy: post + test/free, exactly like net.cc isend/test
  const uint32_t sizes[4] = {0, 64, 5000, 65536};
  srand(7);
  uint64_t posted = 0, completed = 0;
  std::vector<uint32_t> outstanding;  // seqs in flight
  uint64_t spins_since_progress = 0;
  while (completed < kTarget) {
    // post while depth < 12
    while (outstanding.size() < 12 && posted < kTarget) {
      SendRequest* r = &c.reqs[c.seq_next % NCCL_NET_MAX_REQUESTS];
      if (ss_state(r->state_seq.load(std::memory_order_acquire)) !=
          REQ_FREE)
        break;  // slot busy → would be NCCL retry; with depth<12 = stall
      uint32_t size = sizes[rand() & 3];
      r->cursor.store(pack_cur(c.seq_next, 0), std::memory_order_relaxed);
      r->total = size;
      r->chunk = pick_chunk_size(size, 32768, 1 << 20, 4);
      r->sent.store(0, std::memory_order_relaxed);
      r->hdr_sent.store(false, std::memory_order_relaxed);
      r->comm = &c;
      r->src = (const char*)&c;  // dummy
      r->avail.store(size, std::memory_order_relaxed);
      uint32_t seq = c.seq_next;
      r->state_seq.store(pack_ss(seq, REQ_ACTIVE),
                         std::memory_order_seq_cst);
      std::atomic_thread_fence(std::memory_order_seq_cst);
      c.seq_next++;
      posted++;
      outstanding.push_back(seq);
    }
    // test/free in order-agnostic fashion
    bool progress = false;
    for (size_t i = 0; i < outstanding.size();) {
      uint32_t seq = outstanding[i];
      SendRequest* r = &c.reqs[seq % NCCL_NET_MAX_REQUESTS];
      if (r->complete()) {
        uint64_t ss = r->state_seq.load(std::memory_order_relaxed);
        r->state_seq.store(pack_ss(ss_seq(ss), REQ_FREE),
                           std::memory_order_release);
        outstanding[i] = outstanding.back();
        outstanding.pop_back();
        completed++;
        progress = true;
      } else {
        i++;
      }
    }
    if (progress) {
      spins_since_progress = 0;
    } else if (++spins_since_progress > 200'000'000) {
      fprintf(stderr,
              "STALL: posted=%lu completed=%lu outstanding=%zu "
              "oldest=%u next=%u\n",
              (unsigned long)posted, (unsigned long)completed,
              outstanding.size(), c.oldest.load(), c.seq_next);
      for (uint32_t seq : outstanding) {
        SendRequest* r = &c.reqs[seq % NCCL_NET_MAX_REQUESTS];
        fprintf(stderr,
                "  seq=%u total=%u chunk=%u cur=%u/g%u sent=%u hdrsent=%d\n",
                seq, r->total.load(), r->chunk.load(), cur_off(r->cursor.load()),
                cur_gen(r->cursor.load()), r->sent.load(),
                (int)r->hdr_sent.load());
      }
      stop = true;
      for (auto& t : workers) t.join();
      return 1;
    }
  }
  stop = true;
  for (auto& t : workers) t.join();
  printf("claim stress ok: %lu messages, %lu claims\n",
         (unsigned long)completed, (unsigned long)claims.load());
  return 0;
}
