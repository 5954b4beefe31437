// uring_engine.cc — io_uring completion-based IO engine (BNET_IMPLEMENT=URING).
//
// The second engine, paralleling the reference's BASIC/TOKIO backend pair
//  the modern-Linux way: the epoll engine is
// readiness-based (wait → syscall read/writev per span), this one is
// completion-based — SQEs for WRITEV/RECV are batched into one
// io_uring_enter and the kernel reports finished transfers through the CQ
// ring.  Raw syscalls + mmap'd rings (no liburing in the image).
//
// Semantics are identical to the epoll engine: same chunk claiming
// (claim_chunk), same header processing (process_recv_header), same
// completion accounting (finish_rx_chunk), same parking rules.  Sends
// are inline-first: a claimed batch is written with one direct
// nonblocking writev and the ring carries only the EAGAIN remainder
// (at most one WRITEV op in flight per socket — concurrent ops would
// interleave the byte stream).  Receives mirror it: drain_recv reads
// greedily and the armed RECV op only waits for new data.

#include <errno.h>
#include <linux/io_uring.h>
#include <string.h>
#include <sys/eventfd.h>
#include <sys/mman.h>
#include <sys/syscall.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <condition_variable>
#include <mutex>
#include <thread>
#include <vector>

#include "baguanet/config.h"
#include "baguanet/log.h"
#include "staging.h"
#include "telemetry.h"
#include "transport.h"

namespace baguanet {

namespace {

// set by submit_send(allow_ring=false) when a batch tail needs a ring op
// the calling (proxy) thread cannot arm; thread-local because several
// proxy threads may inline concurrently
thread_local bool ring_needed_ = false;

int sys_uring_setup(unsigned entries, io_uring_params* p) {
  return (int)syscall(__NR_io_uring_setup, entries, p);
}
int sys_uring_enter(int fd, unsigned to_submit, unsigned min_complete,
                    unsigned flags) {
  return (int)syscall(__NR_io_uring_enter, fd, to_submit, min_complete,
                      flags, nullptr, 0);
}

class UringRing {
 public:
  bool init(unsigned entries) {
    io_uring_params p{};
    fd_ = sys_uring_setup(entries, &p);
    if (fd_ < 0) return false;
    if (!(p.features & IORING_FEAT_SINGLE_MMAP)) {
      close(fd_);
      fd_ = -1;
      return false;  // pre-5.4 kernels — not our targets
    }
    sq_sz_ = p.sq_off.array + p.sq_entries * sizeof(unsigned);
    size_t cq_sz = p.cq_off.cqes + p.cq_entries * sizeof(io_uring_cqe);
    if (cq_sz > sq_sz_) sq_sz_ = cq_sz;
    ring_ = mmap(nullptr, sq_sz_, PROT_READ | PROT_WRITE,
                 MAP_SHARED | MAP_POPULATE, fd_, IORING_OFF_SQ_RING);
    sqes_sz_ = p.sq_entries * sizeof(io_uring_sqe);
    sqes_ = (io_uring_sqe*)mmap(nullptr, sqes_sz_, PROT_READ | PROT_WRITE,
                                MAP_SHARED | MAP_POPULATE, fd_,
                                IORING_OFF_SQES);
    if (ring_ == MAP_FAILED || sqes_ == MAP_FAILED) {
      close(fd_);
      fd_ = -1;
      return false;
    }
    char* r = (char*)ring_;
    sq_head_ = (std::atomic<unsigned>*)(r + p.sq_off.head);
    sq_tail_ = (std::atomic<unsigned>*)(r + p.sq_off.tail);
    sq_mask_ = *(unsigned*)(r + p.sq_off.ring_mask);
    sq_array_ = (unsigned*)(r + p.sq_off.array);
    cq_head_ = (std::atomic<unsigned>*)(r + p.cq_off.head);
    cq_tail_ = (std::atomic<unsigned>*)(r + p.cq_off.tail);
    cq_mask_ = *(unsigned*)(r + p.cq_off.ring_mask);
    cqes_ = (io_uring_cqe*)(r + p.cq_off.cqes);
    entries_ = p.sq_entries;
    return true;
  }
  ~UringRing() {
    if (ring_ && ring_ != MAP_FAILED) munmap(ring_, sq_sz_);
    if (sqes_ && sqes_ != (io_uring_sqe*)MAP_FAILED) munmap(sqes_, sqes_sz_);
    if (fd_ >= 0) close(fd_);
  }

  io_uring_sqe* get_sqe() {
    unsigned tail = sq_tail_->load(std::memory_order_relaxed);
    unsigned head = sq_head_->load(std::memory_order_acquire);
    if (tail - head >= entries_) return nullptr;  // ring full
    io_uring_sqe* sqe = &sqes_[tail & sq_mask_];
    memset(sqe, 0, sizeof(*sqe));
    sq_array_[tail & sq_mask_] = tail & sq_mask_;
    pending_tail_ = tail + 1;
    return sqe;
  }
  void advance_tail() {
    sq_tail_->store(pending_tail_, std::memory_order_release);
    to_submit_++;
  }
  // submit queued SQEs; wait for >= min_complete completions
  int enter(unsigned min_complete) {
    unsigned n = to_submit_;
    to_submit_ = 0;
    int rc = sys_uring_enter(fd_, n, min_complete,
                             min_complete ? IORING_ENTER_GETEVENTS : 0);
    return rc;
  }
  io_uring_cqe* peek() {
    unsigned head = cq_head_->load(std::memory_order_relaxed);
    if (head == cq_tail_->load(std::memory_order_acquire)) return nullptr;
    return &cqes_[head & cq_mask_];
  }
  void seen() {
    cq_head_->fetch_add(1, std::memory_order_release);
  }

 private:
  int fd_ = -1;
  void* ring_ = nullptr;
  io_uring_sqe* sqes_ = nullptr;
  size_t sq_sz_ = 0, sqes_sz_ = 0;
  std::atomic<unsigned>* sq_head_ = nullptr;
  std::atomic<unsigned>* sq_tail_ = nullptr;
  unsigned sq_mask_ = 0;
  unsigned* sq_array_ = nullptr;
  std::atomic<unsigned>* cq_head_ = nullptr;
  std::atomic<unsigned>* cq_tail_ = nullptr;
  unsigned cq_mask_ = 0;
  io_uring_cqe* cqes_ = nullptr;
  unsigned entries_ = 0;
  unsigned pending_tail_ = 0;
  unsigned to_submit_ = 0;
};

constexpr uint64_t kUdEventfd = 1;  // user_data for the eventfd read op

class UringIoThread : public IIoThread {
 public:
  void start(int idx) override {
    idx_ = idx;
    evfd_ = eventfd(0, EFD_CLOEXEC);  // blocking read op via the ring
    if (!ring_.init(256)) {
      // engine factory guarantees availability; treat as fatal for thread
      BNET_WARN("io_uring init failed mid-run: %s", strerror(errno));
      return;
    }
    thr_ = std::thread([this] { run(); });
  }
  void stop() override {
    if (!thr_.joinable()) return;
    stop_.store(true);
    uint64_t one = 1;
    (void)!write(evfd_, &one, sizeof(one));
    thr_.join();
    close(evfd_);
  }
  void add_sock(TcpSock* s) override {
    enqueue({Task::ADD, s, nullptr});
  }
  void remove_sock_sync(TcpSock* s) override {
    auto tok = std::make_shared<SyncToken>();
    enqueue({Task::REMOVE, s, tok});
    tok->wait();
  }
  void kick(TcpSock* s) override { enqueue({Task::KICK, s, nullptr}); }

 private:
  struct Task {
    enum { ADD, REMOVE, KICK } kind;
    TcpSock* s;
    std::shared_ptr<SyncToken> tok;  // REMOVE only
  };

  void enqueue(Task t) {
    {
      std::lock_guard<std::mutex> lk(task_mu_);
      tasks_.push_back(t);
    }
    uint64_t one = 1;
    (void)!write(evfd_, &one, sizeof(one));
  }

  void arm_eventfd() {
    io_uring_sqe* sqe = ring_.get_sqe();
    if (!sqe) return;  // ring full — re-armed on next drain
    sqe->opcode = IORING_OP_READ;
    sqe->fd = evfd_;
    sqe->addr = (uint64_t)(uintptr_t)&ev_buf_;
    sqe->len = sizeof(ev_buf_);
    sqe->user_data = kUdEventfd;
    ring_.advance_tail();
    evfd_armed_ = true;
  }

  // ---- send path -------------------------------------------------------
  // Claim up to kUrBatch chunks and submit them as ONE ordered WRITEV —
  // multiple independent ops per socket would complete concurrently and
  // interleave bytes, and a single chunk per op starves the kernel
  // between completion and resubmission (measured 24 vs 47 GB/s against
  // the epoll engine before batching).
  //
  // owner_busy exclusion (same protocol as the epoll engine): every
  // send-side entry — ADD/KICK tasks, the spin retry loop, the CQE
  // handler's resubmission, and the PROXY's inline attempt — goes
  // through submit_send_gated / try_inline_send.  The proxy must never
  // touch the ring, so its path runs submit_send(allow_ring=false):
  // an EAGAIN tail marks ring_needed_ and falls back to a kick so the
  // engine thread arms the WRITEV.
  void submit_send_gated(TcpSock* s) {
    if (s->owner_busy.exchange(1, std::memory_order_acquire) != 0) {
      s->rekick.store(true, std::memory_order_release);
      return;
    }
    submit_send(s);
    s->owner_busy.store(0, std::memory_order_release);
    if (s->rekick.exchange(false, std::memory_order_acq_rel)) kick(s);
  }

  bool try_inline_send(TcpSock* s) override {
    if (s->fd < 0) return false;
    if (s->owner_busy.exchange(1, std::memory_order_acquire) != 0)
      return false;  // engine thread owns it right now — caller kicks
    bool ok = false;
    if (s->ur.op == 0 && !s->ur.closing &&
        !s->scomm->error.load(std::memory_order_relaxed)) {
      ring_needed_ = false;
      submit_send(s, /*allow_ring=*/false);
      ok = !ring_needed_;  // partial batch: owner must arm the WRITEV
    }
    s->owner_busy.store(0, std::memory_order_release);
    if (s->rekick.exchange(false, std::memory_order_acq_rel)) kick(s);
    return ok;
  }

  void submit_send(TcpSock* s, bool allow_ring = true) {
    SendComm* c = s->scomm;
    if (s->ur.op || s->ur.closing ||
        c->error.load(std::memory_order_relaxed))
      return;
    if (s->ur.nchunks == 0) {
      s->ur.batch_bytes = 0;
      s->ur.done = 0;
      while (s->ur.nchunks < TcpSock::kUrBatch) {
        uint32_t off = 0, len = 0;
        SendRequest* r = claim_chunk(c, s->idx, &off, &len);
        if (!r && s->ur.nchunks == 0) {
          // same idle publish-then-recheck handshake as the epoll engine
          s->snd_idle.store(true, std::memory_order_seq_cst);
          std::atomic_thread_fence(std::memory_order_seq_cst);
          r = claim_chunk(c, s->idx, &off, &len);
          if (r) s->snd_idle.store(false, std::memory_order_relaxed);
        }
        if (!r) break;
        int i = s->ur.nchunks++;
        s->ur.reqs[i] = r;
        s->ur.hdrs[i] = {ss_seq(r->state_seq.load(std::memory_order_relaxed)),
                         off, len, r->total, r->tag};
        s->ur.payloads[i] = r->src + off;
        s->ur.batch_bytes += (uint32_t)sizeof(ChunkHdr) + len;
      }
      if (s->ur.nchunks == 0) return;  // idle (snd_idle set) until kicked
    s->snd_idle.store(false, std::memory_order_relaxed);
    }
    // Build the iovec for the unwritten tail of the batch's virtual stream
    // [hdr0|pay0|hdr1|pay1|...].
    int iovn = 0;
    uint32_t skip = s->ur.done;
    for (int i = 0; i < s->ur.nchunks; i++) {
      uint32_t span = (uint32_t)sizeof(ChunkHdr) + s->ur.hdrs[i].len;
      if (skip >= span) {
        skip -= span;
        continue;
      }
      uint32_t o = skip;
      skip = 0;
      if (o < sizeof(ChunkHdr))
        s->ur.iov[iovn++] = {(char*)&s->ur.hdrs[i] + o, sizeof(ChunkHdr) - o};
      uint32_t pay_off =
          o > sizeof(ChunkHdr) ? o - (uint32_t)sizeof(ChunkHdr) : 0;
      if (s->ur.hdrs[i].len > pay_off)
        s->ur.iov[iovn++] = {(void*)(s->ur.payloads[i] + pay_off),
                             s->ur.hdrs[i].len - pay_off};
    }
    if (iovn == 0) {
      finish_batch(s);
      return;
    }
    // Inline-first: try one direct nonblocking writev before arming a
    // ring op.  On an idle ring, CQE delivery costs ~20 us per wakeup
    // (poll-arm -> task_work), which dominated single-message latency
    // (measured 70 us RTT vs 17 us on epoll); writing synchronously from
    // the kicked thread puts the bytes on the wire immediately, and the
    // ring carries only the remainder when the socket buffer fills.
    // Mirrors the receive side, where drain_recv already reads greedily
    // and the RECV op only waits for new data.
    ssize_t n = writev(s->fd, s->ur.iov, iovn);
    if (n > 0) {
      s->ur.done += (uint32_t)n;
      if (s->ur.done == s->ur.batch_bytes) {
        finish_batch(s);
        submit_send(s, allow_ring);  // claim + send the next batch too
        return;
      }
      // partial: rebuild the iovec for the unwritten tail
      submit_send_tail(s, allow_ring);
      return;
    }
    if (n < 0 && errno != EAGAIN && errno != EWOULDBLOCK &&
        errno != EINTR) {
      s->scomm->error.store(errno);
      BNET_WARN("bnet(uring) inline send error: %s", strerror(errno));
      return;
    }
    if (!allow_ring) {
      ring_needed_ = true;  // proxy path: the engine thread must arm
      return;
    }
    arm_writev(s, iovn);
  }

  // arm the WRITEV SQE for the current iov (already built)
  void arm_writev(TcpSock* s, int iovn) {
    io_uring_sqe* sqe = ring_.get_sqe();
    if (!sqe) return;  // ring full; retried after next drain
    sqe->opcode = IORING_OP_WRITEV;
    sqe->fd = s->fd;
    sqe->addr = (uint64_t)(uintptr_t)s->ur.iov;
    sqe->len = iovn;
    sqe->user_data = (uint64_t)(uintptr_t)s;
    ring_.advance_tail();
    s->ur.op = 1;
  }

  // rebuild the iov for the unwritten tail of the batch and arm it
  void submit_send_tail(TcpSock* s, bool allow_ring = true) {
    int iovn = 0;
    uint32_t skip = s->ur.done;
    for (int i = 0; i < s->ur.nchunks; i++) {
      uint32_t span = (uint32_t)sizeof(ChunkHdr) + s->ur.hdrs[i].len;
      if (skip >= span) {
        skip -= span;
        continue;
      }
      uint32_t o = skip;
      skip = 0;
      if (o < sizeof(ChunkHdr))
        s->ur.iov[iovn++] = {(char*)&s->ur.hdrs[i] + o, sizeof(ChunkHdr) - o};
      uint32_t pay_off =
          o > sizeof(ChunkHdr) ? o - (uint32_t)sizeof(ChunkHdr) : 0;
      if (s->ur.hdrs[i].len > pay_off)
        s->ur.iov[iovn++] = {(void*)(s->ur.payloads[i] + pay_off),
                             s->ur.hdrs[i].len - pay_off};
    }
    if (iovn == 0) {
      finish_batch(s);
      return;
    }
    if (!allow_ring) {
      ring_needed_ = true;
      return;
    }
    arm_writev(s, iovn);
  }

  void finish_batch(TcpSock* s) {
    SendComm* c = s->scomm;
    for (int i = 0; i < s->ur.nchunks; i++) {
      SendRequest* r = s->ur.reqs[i];
      uint32_t len = s->ur.hdrs[i].len;
      if (r->total == 0) {
        r->hdr_sent.store(true, std::memory_order_release);
      } else {
        c->stats.bytes_sent.fetch_add(len, std::memory_order_relaxed);
        r->sent.fetch_add(len, std::memory_order_acq_rel);
      }
    }
    s->ur.nchunks = 0;
    s->ur.done = 0;
    s->ur.batch_bytes = 0;
  }

  void on_send_cqe(TcpSock* s, int res) {
    // ur.* is mutated while ur.op is STILL 1 — the proxy's inline path
    // bails on op != 0, so the armed op doubles as the CQE handler's
    // exclusion; op clears (release) only after the state is consistent
    if (res <= 0) {
      s->ur.op.store(0, std::memory_order_release);
      if (res == -ECANCELED) return;  // teardown cancel
      if (res == -EAGAIN || res == -EINTR) {
        submit_send_gated(s);  // retry
        return;
      }
      s->scomm->error.store(res ? -res : EPIPE);
      BNET_WARN("bnet(uring) send error: %s", strerror(-res));
      return;
    }
    s->ur.done += (uint32_t)res;
    if (s->ur.done == s->ur.batch_bytes) finish_batch(s);
    s->ur.op.store(0, std::memory_order_release);
    submit_send_gated(s);  // remainder / next batch
  }

  // ---- recv path -------------------------------------------------------
  void submit_recv(TcpSock* s) {
    RecvComm* c = s->rcomm;
    if (s->ur.op || s->ur.closing ||
        c->error.load(std::memory_order_relaxed))
      return;
    if (s->parked.load(std::memory_order_relaxed)) return;
    char* buf;
    uint32_t want;
    if (!s->rx.in_payload) {
      buf = (char*)&s->rx.hdr + s->rx.hdr_got;
      want = sizeof(ChunkHdr) - s->rx.hdr_got;
    } else {
      buf = s->rx.target;
      want = s->rx.remaining;
    }
    io_uring_sqe* sqe = ring_.get_sqe();
    if (!sqe) return;
    sqe->opcode = IORING_OP_RECV;
    sqe->fd = s->fd;
    sqe->addr = (uint64_t)(uintptr_t)buf;
    sqe->len = want;
    sqe->user_data = (uint64_t)(uintptr_t)s;
    ring_.advance_tail();
    s->ur.op = 2;
  }

  void on_recv_cqe(TcpSock* s, int res) {
    RecvComm* c = s->rcomm;
    s->ur.op = 0;
    if (res < 0) {
      if (res == -ECANCELED) return;  // teardown cancel
      if (res == -EAGAIN || res == -EINTR) {
        submit_recv(s);
        return;
      }
      c->error.store(-res);
      BNET_WARN("bnet(uring) recv error: %s", strerror(-res));
      return;
    }
    if (res == 0) {  // EOF
      if (!s->rx.in_payload && s->rx.hdr_got == 0 &&
          !recv_socket_incomplete(c)) {
        s->ur.eof = true;
        return;  // orderly shutdown: never re-arm
      }
      c->error.store(ECONNRESET);
      BNET_WARN("bnet(uring) recv eof mid-protocol");
      return;
    }
    // absorb the CQE's bytes into the rx state, then drain the socket
    // greedily with plain read() — one RECV op per burst of buffered
    // data (a RECV op per span left the kernel buffer idle between
    // completion and resubmission: 2-3x slower at 1 MiB, measured)
    if (!s->rx.in_payload) {
      s->rx.hdr_got += (uint32_t)res;
    } else {
      s->rx.target += res;
      s->rx.remaining -= (uint32_t)res;
      if (s->rx.remaining == 0) finish_rx_chunk(s);
    }
    if (drain_recv(s) == RX_WAIT) submit_recv(s);
    // RX_PARKED: irecv()'s kick resumes; RX_CLOSED: stop re-arming
  }

  void resume_parked(TcpSock* s) {
    // header already buffered in rx.hdr — drain retries the match
    if (drain_recv(s) == RX_WAIT) submit_recv(s);
  }

  void handle_tasks() {
    std::vector<Task> batch;
    {
      std::lock_guard<std::mutex> lk(task_mu_);
      batch.swap(tasks_);
    }
    for (auto& t : batch) {
      switch (t.kind) {
        case Task::ADD:
          socks_.push_back(t.s);
          if (t.s->is_recv)
            submit_recv(t.s);
          else
            submit_send_gated(t.s);
          break;
        case Task::REMOVE: {
          TcpSock* s = t.s;
          s->ur.closing = true;  // block resubmission from completions
          // cancel any outstanding op on this fd, then reap its CQE
          if (s->ur.op) {
            io_uring_sqe* sqe = ring_.get_sqe();
            if (sqe) {
              sqe->opcode = IORING_OP_ASYNC_CANCEL;
              sqe->addr = (uint64_t)(uintptr_t)s;
              sqe->user_data = (uint64_t)(uintptr_t)s | 2;  // cancel marker
              ring_.advance_tail();
              ring_.enter(0);
            }
            // drain until the op's CQE arrives
            int spins = 0;
            while (s->ur.op && spins++ < 1000000) {
              drain_cqes();
              if (s->ur.op) ring_.enter(1);
            }
          }
          close(s->fd);
          s->fd = -1;
          socks_.erase(std::remove(socks_.begin(), socks_.end(), s),
                       socks_.end());
          if (s->scomm) s->scomm->live_socks.fetch_sub(1);
          if (s->rcomm) s->rcomm->live_socks.fetch_sub(1);
          t.tok->signal();
          break;
        }
        case Task::KICK:
          if (std::find(socks_.begin(), socks_.end(), t.s) != socks_.end()) {
            if (t.s->is_recv) {
              if (t.s->parked.load(std::memory_order_relaxed))
                resume_parked(t.s);
              else
                submit_recv(t.s);  // no-op if an op is already in flight
            } else {
              submit_send_gated(t.s);
            }
          }
          break;
      }
    }
  }

  void drain_cqes() {
    io_uring_cqe* cqe;
    while ((cqe = ring_.peek())) {
      uint64_t ud = cqe->user_data;
      int res = cqe->res;
      ring_.seen();
      if (ud == kUdEventfd) {
        evfd_armed_ = false;
        continue;
      }
      if (ud & 2) continue;  // ASYNC_CANCEL result — ignore
      TcpSock* s = (TcpSock*)(uintptr_t)ud;
      if (s->fd < 0) {
        s->ur.op = 0;  // completion for a socket being removed
        continue;
      }
      if (s->ur.op == 1)
        on_send_cqe(s, res);
      else if (s->ur.op == 2)
        on_recv_cqe(s, res);
    }
  }

  void run() {
    char tname[16];
    snprintf(tname, sizeof(tname), "bnet-ur%d", idx_);
    pthread_setname_np(pthread_self(), tname);
    arm_eventfd();
    ring_.enter(0);
    uint64_t last_active_ns = 0;
    while (!stop_.load(std::memory_order_relaxed)) {
      bool staging_busy = false;
      for (TcpSock* s : socks_) {
        if (s->scomm && s->scomm->stage_pool &&
            stage_pending(s->scomm->stage_pool)) {
          stage_poll(s->scomm->stage_pool);
          staging_busy = true;
        }
      }
      uint64_t now = now_ns();
      bool spin = staging_busy ||
                  (now - last_active_ns < Config::get().spin_us * 1000ull);
      if (!evfd_armed_) arm_eventfd();
      int rc = ring_.enter(spin ? 0 : 1);
      (void)rc;
      io_uring_cqe* had = ring_.peek();
      if (had) last_active_ns = now_ns();  // fresh: `now` predates the wait
      drain_cqes();
      handle_tasks();
      if (spin || had) {
        // jobs/watermarks may have advanced: retry any socket without an
        // op in flight.  Senders with a prepared batch (nchunks > 0) are
        // included — a ring-full get_sqe() at submit time would otherwise
        // strand the batch until an external kick (reachable only with
        // >255 sockets on one thread, but cheap to close).
        for (TcpSock* s : socks_) {
          if (s->ur.op) continue;
          if (s->is_recv) {
            if (!s->ur.eof && !s->parked.load(std::memory_order_relaxed))
              submit_recv(s);
          } else {
            submit_send_gated(s);
          }
        }
        ring_.enter(0);
      }
    }
  }

  int idx_ = 0;
  int evfd_ = -1;
  uint64_t ev_buf_ = 0;
  bool evfd_armed_ = false;
  UringRing ring_;
  std::thread thr_;
  std::atomic<bool> stop_{false};
  std::mutex task_mu_;
  std::vector<Task> tasks_;
  std::vector<TcpSock*> socks_;
};

}  // namespace

bool uring_available() {
  static int avail = -1;
  if (avail < 0) {
    // test hook: exercise the seccomp-fallback path without seccomp
    if (const char* v = getenv("BNET_FORCE_NO_URING"); v && *v == '1') {
      avail = 0;
      return false;
    }
    io_uring_params p{};
    int fd = sys_uring_setup(4, &p);
    if (fd >= 0) {
      avail = (p.features & IORING_FEAT_SINGLE_MMAP) ? 1 : 0;
      close(fd);
    } else {
      avail = 0;
    }
  }
  return avail == 1;
}

IIoThread* make_uring_thread() { return new UringIoThread(); }

}  // namespace baguanet
