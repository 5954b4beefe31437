// transport.cc — see transport.h for the design rationale.

#include "transport.h"

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <netinet/tcp.h>
#include <string.h>
#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <sys/uio.h>
#include <unistd.h>

#include <algorithm>
#include <random>

#include "baguanet/log.h"
#include "staging.h"
#include "telemetry.h"

namespace baguanet {

// ------------------------------------------------------------- helpers ----

uint32_t pick_chunk_size(uint32_t total, uint32_t min_chunk,
                         uint32_t max_chunk, int nstreams) {
  if (nstreams < 1) nstreams = 1;
  uint32_t per = total ? (total + nstreams - 1) / nstreams : min_chunk;
  per = std::max(per, min_chunk);
  per = std::min(per, std::max(max_chunk, min_chunk));
  return per;
}

// ------------------------------------------------------------ IoThread ----

void IoThread::start(int idx) {
  idx_ = idx;
  epfd_ = epoll_create1(EPOLL_CLOEXEC);
  evfd_ = eventfd(0, EFD_NONBLOCK | EFD_CLOEXEC);
  epoll_event ev{};
  ev.events = EPOLLIN;
  ev.data.ptr = nullptr;  // nullptr marks the eventfd
  epoll_ctl(epfd_, EPOLL_CTL_ADD, evfd_, &ev);
  thr_ = std::thread([this] { run(); });
}

void IoThread::stop() {
  if (!thr_.joinable()) return;
  stop_.store(true);
  uint64_t one = 1;
  (void)!write(evfd_, &one, sizeof(one));
  thr_.join();
  close(epfd_);
  close(evfd_);
}

void IoThread::add_sock(TcpSock* s) {
  std::lock_guard<std::mutex> lk(task_mu_);
  tasks_.push_back({Task::ADD, s, nullptr});
  uint64_t one = 1;
  (void)!write(evfd_, &one, sizeof(one));
}

void IoThread::remove_sock_sync(TcpSock* s) {
  auto tok = std::make_shared<SyncToken>();
  {
    std::lock_guard<std::mutex> lk(task_mu_);
    tasks_.push_back({Task::REMOVE, s, tok});
    uint64_t one = 1;
    (void)!write(evfd_, &one, sizeof(one));
  }
  tok->wait();
}

void IoThread::kick(TcpSock* s) {
  s->dbg_kicks.fetch_add(1, std::memory_order_relaxed);
  std::lock_guard<std::mutex> lk(task_mu_);
  tasks_.push_back({Task::KICK, s, nullptr});
  uint64_t one = 1;
  (void)!write(evfd_, &one, sizeof(one));
}

void IoThread::handle_tasks() {
  std::vector<Task> batch;
  {
    std::lock_guard<std::mutex> lk(task_mu_);
    batch.swap(tasks_);
  }
  for (auto& t : batch) {
    switch (t.kind) {
      case Task::ADD: {
        TcpSock* s = t.s;
        socks_.push_back(s);
        epoll_event ev{};
        ev.events = s->is_recv ? EPOLLIN : 0;
        ev.data.ptr = s;
        s->epollin_on = s->is_recv;
        epoll_ctl(epfd_, EPOLL_CTL_ADD, s->fd, &ev);
        progress(s);
        break;
      }
      case Task::REMOVE: {
        TcpSock* s = t.s;
        epoll_ctl(epfd_, EPOLL_CTL_DEL, s->fd, nullptr);
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
        // The socket may have been removed between enqueue and drain.
        if (std::find(socks_.begin(), socks_.end(), t.s) != socks_.end()) {
          t.s->dbg_kick_run.fetch_add(1, std::memory_order_relaxed);
          progress(t.s);
        } else {
          t.s->dbg_find_fail.fetch_add(1, std::memory_order_relaxed);
        }
        break;
    }
  }
}

void IoThread::run() {
  char tname[16];
  snprintf(tname, sizeof(tname), "bnet-io%d", idx_);
  pthread_setname_np(pthread_self(), tname);
  epoll_event evs[64];
  uint64_t last_active_ns = 0;
  while (!stop_.load(std::memory_order_relaxed)) {
    // Spin (timeout 0) while staging copies are pending (GPU watermarks
    // must advance promptly) or within a short window after traffic —
    // a blocked epoll_wait costs an eventfd wakeup (~5-10 us) per
    // message, which dominates small-message rates.  NCCL's own proxy
    // threads spin the same way.
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
    int n = epoll_wait(epfd_, evs, 64, spin ? 0 : 100);
    if (n < 0 && errno != EINTR) {
      // A hard epoll failure would silently strand every socket on this
      // thread; surface it as a comm error on all of them and say so.
      BNET_WARN("bnet-io%d: epoll_wait failed: %s — marking %zu socket(s) "
                "errored", idx_, strerror(errno), socks_.size());
      for (TcpSock* s : socks_) {
        if (s->scomm) s->scomm->error.store(errno ? errno : EIO);
        if (s->rcomm) s->rcomm->error.store(errno ? errno : EIO);
      }
      break;
    }
    // stamp with a FRESH timestamp: `now` predates a blocking wait, and a
    // stale stamp would fail to arm the spin window for the burst that
    // just started
    if (n > 0) last_active_ns = now_ns();
    for (int i = 0; i < n; i++) {
      if (evs[i].data.ptr == nullptr) {
        uint64_t v;
        (void)!read(evfd_, &v, sizeof(v));
        continue;
      }
      progress(static_cast<TcpSock*>(evs[i].data.ptr));
    }
    // Drain tasks EVERY iteration, not only on eventfd wakes: gating the
    // drain on (had_ev || spin) left enqueued kicks stranded in rare
    // interleavings (observed: kicks enqueued == executed + N with all
    // threads asleep), stalling a message until its slot wrapped.  The
    // eventfd still guarantees a sleeping thread wakes; draining is
    // unconditional so no wake can race past its task.
    handle_tasks();
    if (spin) {
      // watermarks/jobs may have advanced without an epoll event; also
      // resume any mid-write socket that is not EPOLLOUT-armed (progress
      // on an active tx just continues its writev)
      for (TcpSock* s : socks_)
        if (!s->is_recv && !s->want_epollout) progress(s);
    } else {
      // Pre-block sweep: unconditional forward-progress guarantee.  Kicks
      // are an optimization; before sleeping, re-scan idle senders and
      // parked receivers so that even a lost wakeup costs at most one
      // idle-loop period (100 ms), never a stall.  (A residual ~1-in-200k
      // lost kick was still observed after the drain fix; this bounds it.)
      for (TcpSock* s : socks_) {
        if (!s->is_recv && !s->want_epollout) {
          uint32_t before = s->dbg_claims.load(std::memory_order_relaxed);
          progress(s);
          if (s->dbg_claims.load(std::memory_order_relaxed) != before)
            s->dbg_sweep_rescue.fetch_add(1, std::memory_order_relaxed);
        } else if (s->is_recv) {
          // parked sockets re-check their slot; armed ones cost one
          // EAGAIN read per idle period — total immunity to any missed
          // readiness edge
          progress(s);
        }
      }
    }
  }
}

void IoThread::set_epollout(TcpSock* s, bool on) {
  if (s->want_epollout == on || s->fd < 0) return;
  s->want_epollout = on;
  epoll_event ev{};
  ev.events = (s->is_recv ? EPOLLIN : 0) | (on ? EPOLLOUT : 0);
  ev.data.ptr = s;
  epoll_ctl(epfd_, EPOLL_CTL_MOD, s->fd, &ev);
}

void IoThread::progress(TcpSock* s) {
  if (s->fd < 0) return;
  if (s->is_recv) {
    progress_recv(s);
    return;
  }
  // owner_busy exclusion against a proxy-thread inline send.  The loser
  // marks rekick; the winner re-kicks after releasing, so a transition
  // that raced with the release is never lost.
  if (s->owner_busy.exchange(1, std::memory_order_acquire) != 0) {
    s->rekick.store(true, std::memory_order_release);
    return;
  }
  progress_send(s);
  s->owner_busy.store(0, std::memory_order_release);
  if (s->rekick.exchange(false, std::memory_order_acq_rel)) kick(s);
}

bool IoThread::try_inline_send(TcpSock* s) {
  if (s->fd < 0) return false;
  if (s->owner_busy.exchange(1, std::memory_order_acquire) != 0)
    return false;  // owner busy right now — caller falls back to kick()
  // progress_send is thread-agnostic under owner_busy exclusion: tx state
  // is exclusively owned, epoll_ctl (set_epollout) is kernel-thread-safe,
  // kick()/debug counters are atomic or mutex-protected.
  progress_send(s);
  s->owner_busy.store(0, std::memory_order_release);
  if (s->rekick.exchange(false, std::memory_order_acq_rel)) kick(s);
  return true;
}

// Claim the next unsent chunk across the comm's active requests, oldest
// first (FIFO completion; dynamic stream assignment — reference TODO
// nthread:335 realized).  Single-chunk messages are statically routed to
// socket (seq % nsocks): letting all sockets race for a one-chunk message
// just bounces the slot cache lines across 4 IO threads and halves the
// small-message rate (measured on the MI355X box: 3.8 vs 11.4 GB/s at
// 64 KiB).  Multi-chunk messages keep fully dynamic assignment.
SendRequest* claim_chunk(SendComm* c, int sock_idx, uint32_t* off,
                         uint32_t* len, TcpSock* dbg) {
  int nsocks = (int)c->socks.size();
  // Single-chunk messages fan out over at most 2 sockets: measured on the
  // MI355X box, 64 KiB message rate is ~11 GB/s at 1-2 sockets but drops
  // to 3-5 GB/s when spread over 4 (cross-thread handoff per message
  // outweighs parallelism below the striping threshold).
  int nsmall = nsocks < 2 ? nsocks : 2;
  uint32_t oldest = c->oldest.load(std::memory_order_acquire);
  uint32_t newest = oldest + NCCL_NET_MAX_REQUESTS;
  for (uint32_t s = oldest; s != newest; s++) {
    SendRequest* r = &c->reqs[s % NCCL_NET_MAX_REQUESTS];
    uint64_t ss = r->state_seq.load(std::memory_order_acquire);
    uint32_t st = ss_state(ss);
    uint32_t rseq = ss_seq(ss);
    if (st != REQ_ACTIVE || rseq != s) {
      if (rseq == s || (st == REQ_ACTIVE && rseq != s)) {
        // seq s already completed (slot FREE with our seq, or reused by a
        // newer request) — advance the frontier and keep scanning.
        if (s == oldest) {
          uint32_t expect = oldest;
          c->oldest.compare_exchange_strong(expect, s + 1);
        }
        continue;
      }
      if (dbg) {
        dbg->dbg_breaks.fetch_add(1, std::memory_order_relaxed);
        dbg->dbg_break_s.store(s, std::memory_order_relaxed);
        dbg->dbg_break_ss.store(ss, std::memory_order_relaxed);
        dbg->dbg_break_old.store(oldest, std::memory_order_relaxed);
      }
      break;  // seq s was never posted → nothing newer exists either
    }
    if (r->total <= r->chunk) {
      // single-chunk (or empty) message: statically owned
      if ((int)(s % (uint32_t)nsmall) != sock_idx) continue;
    }
    if (r->total == 0) {
      // header-only claim: offset field doubles as the claim flag; the
      // generation tag rejects stale claims after slot reuse
      uint64_t cur64 = pack_cur(s, 0);
      if (r->cursor.compare_exchange_strong(cur64, pack_cur(s, 1))) {
        *off = 0;
        *len = 0;
        return r;
      }
      continue;
    }
    uint32_t avail = std::min(r->avail.load(std::memory_order_acquire),
                              r->total.load(std::memory_order_relaxed));
    uint64_t cur64 = r->cursor.load(std::memory_order_relaxed);
    while (cur_gen(cur64) == s && cur_off(cur64) < avail) {
      uint32_t cur = cur_off(cur64);
      uint32_t end = std::min(cur + r->chunk.load(std::memory_order_relaxed), avail);
      if (r->cursor.compare_exchange_weak(cur64, pack_cur(s, end))) {
        *off = cur;
        *len = end - cur;
        return r;
      }
      // cur64 reloaded by the failed CAS; the gen check guards reuse
    }
    // nothing claimable in this request (fully claimed or staging-limited);
    // move on to the next one
  }
  if (dbg) dbg->dbg_full.fetch_add(1, std::memory_order_relaxed);
  return nullptr;
}

void IoThread::progress_send(TcpSock* s) {
  SendComm* c = s->scomm;
  s->dbg_progress.fetch_add(1, std::memory_order_relaxed);
  if (s->dbg_inprog.fetch_add(1, std::memory_order_acq_rel) != 0)
    s->dbg_reent.fetch_add(1, std::memory_order_relaxed);
  struct Guard {
    TcpSock* s;
    ~Guard() { s->dbg_inprog.fetch_sub(1, std::memory_order_acq_rel); }
  } guard{s};
  if (c->error.load(std::memory_order_relaxed)) {
    s->dbg_exit.store(5, std::memory_order_relaxed);
    return;
  }
  // Fairness credit: one dispatch writes at most `inflight_per_stream`
  // bytes before yielding the IO thread to its other sockets (other
  // comms sharing the NIC), then reschedules itself.
  uint32_t budget = Config::get().inflight_per_stream;
  if (budget < 65536) budget = 65536;
  while (true) {
    if (!s->tx.active) {
      uint32_t off = 0, len = 0;
      SendRequest* r = claim_chunk(c, s->idx, &off, &len, s);
      if (!r) {
        // going idle: publish-then-recheck so a concurrent isend either
        // sees snd_idle (and kicks) or we see its request now
        s->snd_idle.store(true, std::memory_order_seq_cst);
        std::atomic_thread_fence(std::memory_order_seq_cst);
        r = claim_chunk(c, s->idx, &off, &len, s);
        if (!r) {
          set_epollout(s, false);
          s->dbg_exit.store(1, std::memory_order_relaxed);
          return;
        }
        s->snd_idle.store(false, std::memory_order_relaxed);
      }
      s->dbg_claims.fetch_add(1, std::memory_order_relaxed);
      s->tx.active = true;
      s->tx.req = r;
      s->tx.hdr = {ss_seq(r->state_seq.load(std::memory_order_relaxed)), off,
                   len, r->total, r->tag};
      s->tx.payload = r->src + off;
      s->tx.done = 0;
    }
    // Write the 16-byte header + payload as one virtual stream.
    while (true) {
      uint32_t hdr_left =
          s->tx.done < sizeof(ChunkHdr) ? sizeof(ChunkHdr) - s->tx.done : 0;
      uint32_t pay_done = s->tx.done >= sizeof(ChunkHdr)
                              ? s->tx.done - (uint32_t)sizeof(ChunkHdr)
                              : 0;
      uint32_t pay_left = s->tx.hdr.len - pay_done;
      if (hdr_left == 0 && pay_left == 0) break;
      struct iovec iov[2];
      int iovn = 0;
      if (hdr_left)
        iov[iovn++] = {(char*)&s->tx.hdr + (sizeof(ChunkHdr) - hdr_left),
                       hdr_left};
      if (pay_left)
        iov[iovn++] = {(void*)(s->tx.payload + pay_done), pay_left};
      ssize_t w = writev(s->fd, iov, iovn);
      if (w > 0) {
        s->tx.done += (uint32_t)w;
        continue;
      }
      if (w < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
        set_epollout(s, true);
        s->dbg_exit.store(2, std::memory_order_relaxed);
        return;
      }
      if (w < 0 && errno == EINTR) continue;
      c->error.store(errno ? errno : EPIPE);
      BNET_WARN("bnet send socket error: %s", strerror(errno));
      s->dbg_exit.store(3, std::memory_order_relaxed);
      return;
    }
    // chunk fully written
    SendRequest* r = s->tx.req;
    uint32_t len = s->tx.hdr.len;
    s->tx.active = false;
    s->tx.req = nullptr;
    if (r->total == 0) {
      r->hdr_sent.store(true, std::memory_order_release);
    } else {
      c->stats.bytes_sent.fetch_add(len, std::memory_order_relaxed);
      r->sent.fetch_add(len, std::memory_order_acq_rel);
    }
    if (len >= budget) {
      kick(s);  // yield: requeue ourselves behind other sockets' work
      s->dbg_exit.store(4, std::memory_order_relaxed);
      return;
    }
    budget -= len;
  }
}

// Greedy nonblocking recv drain shared by both engines: loop the rx state
// machine with plain read() until the kernel buffer is empty.  Returns:
//   RX_WAIT   — EAGAIN: caller waits for readability (epoll arm / RECV op)
//   RX_PARKED — a chunk for an un-posted request arrived; irecv() kicks
//   RX_CLOSED — error (comm error set) or benign end-of-stream EOF
int drain_recv(TcpSock* s) {
  RecvComm* c = s->rcomm;
  if (c->error.load(std::memory_order_relaxed)) return RX_CLOSED;
  while (true) {
    if (!s->rx.in_payload) {
      // read header
      while (s->rx.hdr_got < sizeof(ChunkHdr)) {
        ssize_t n = read(s->fd, (char*)&s->rx.hdr + s->rx.hdr_got,
                         sizeof(ChunkHdr) - s->rx.hdr_got);
        if (n > 0) {
          s->rx.hdr_got += (uint32_t)n;
          continue;
        }
        if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK))
          return RX_WAIT;
        if (n < 0 && errno == EINTR) continue;
        if (n == 0 && s->rx.hdr_got == 0) {
          // EOF on a message boundary: benign iff no posted recv still
          // expects socket bytes
          if (!recv_socket_incomplete(c)) return RX_CLOSED;
        }
        c->error.store(n == 0 ? ECONNRESET : (errno ? errno : EIO));
        BNET_WARN("bnet recv socket %s", n == 0 ? "eof mid-protocol"
                                                : strerror(errno));
        return RX_CLOSED;
      }
      // header complete → locate the posted request
      int hr = process_recv_header(s);
      if (hr < 0) return RX_CLOSED;  // protocol error (comm error set)
      if (hr == 1) {
        // Not posted yet — park.  Dekker-style handshake with irecv():
        // we publish `parked` (seq_cst) BEFORE re-checking the slot;
        // irecv publishes the slot (seq_cst) BEFORE checking `parked`.
        // At least one side must see the other, so either we proceed now
        // or the kick finds parked==true.
        s->parked.store(true, std::memory_order_seq_cst);
        // the re-check load below is acquire; without a seq_cst fence it
        // may be ordered before the store above (store-buffer litmus) and
        // both sides can miss each other
        std::atomic_thread_fence(std::memory_order_seq_cst);
        hr = process_recv_header(s);
        if (hr < 0) return RX_CLOSED;
        if (hr == 1) return RX_PARKED;
        // posted concurrently: fall through and continue
      }
      if (s->parked.load(std::memory_order_relaxed))
        s->parked.store(false, std::memory_order_relaxed);
      if (s->rx.hdr.len == 0) {
        // empty chunk (only for zero-byte messages)
        finish_rx_chunk(s);
        continue;
      }
    }
    while (s->rx.remaining) {
      ssize_t n = read(s->fd, s->rx.target, s->rx.remaining);
      if (n > 0) {
        s->rx.target += n;
        s->rx.remaining -= (uint32_t)n;
        continue;
      }
      if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK))
        return RX_WAIT;
      if (n < 0 && errno == EINTR) continue;
      c->error.store(n == 0 ? ECONNRESET : (errno ? errno : EIO));
      BNET_WARN("bnet recv payload %s", n == 0 ? "eof mid-chunk"
                                               : strerror(errno));
      return RX_CLOSED;
    }
    finish_rx_chunk(s);
  }
}

void IoThread::progress_recv(TcpSock* s) {
  bool was_parked = s->parked.load(std::memory_order_relaxed);
  int rc = drain_recv(s);
  if (rc == RX_WAIT) {
    if (was_parked && !s->parked.load(std::memory_order_relaxed)) {
      epoll_event ev{};  // resumed from parked: re-arm EPOLLIN
      ev.events = EPOLLIN;
      ev.data.ptr = s;
      s->epollin_on = true;
      epoll_ctl(epfd_, EPOLL_CTL_MOD, s->fd, &ev);
    }
    return;
  }
  if (rc == RX_PARKED || rc == RX_CLOSED) {
    // stop polling this socket (parked: irecv's kick resumes; closed:
    // nothing more to read — errors surface via the comm error)
    epoll_event ev{};
    ev.events = 0;
    ev.data.ptr = s;
    s->epollin_on = false;
    epoll_ctl(epfd_, EPOLL_CTL_MOD, s->fd, &ev);
  }
}

bool recv_socket_incomplete(RecvComm* c) {
  for (auto& r : c->reqs) {
    if (ss_state(r.state_seq.load(std::memory_order_acquire)) != REQ_ACTIVE)
      continue;
    // active but all bytes received (awaiting test()/H2D) is not pending
    if (!r.socket_complete()) return true;
  }
  return false;
}

// rx.hdr is complete: locate/validate the posted request (see transport.h).
int process_recv_header(TcpSock* s) {
  RecvComm* c = s->rcomm;
  const ChunkHdr& h = s->rx.hdr;
  // Frame sanity BEFORE slot matching: a corrupted header must error the
  // comm, not park the socket waiting for a request that can never match.
  constexpr uint32_t kMaxMsg = 1u << 30;  // NCCL's MAX_NET_SIZE
  if (h.total > kMaxMsg || h.len > h.total || h.offset > h.total ||
      h.offset + h.len > h.total || (h.len == 0 && h.total != 0)) {
    BNET_WARN("bnet: malformed chunk header (seq=%u off=%u len=%u total=%u)",
              h.seq, h.offset, h.len, h.total);
    c->error.store(EPROTO);
    return -1;
  }
  uint32_t seq = s->rx.hdr.seq;
  RecvRequest* r = &c->reqs[seq % NCCL_NET_MAX_REQUESTS];
  if (r->state_seq.load(std::memory_order_acquire) !=
      pack_ss(seq, REQ_ACTIVE))
    return 1;  // not posted yet
  int64_t expect = -1;
  r->total.compare_exchange_strong(expect, (int64_t)s->rx.hdr.total);
  if ((int64_t)s->rx.hdr.total != r->total.load()) {
    BNET_WARN("bnet: inconsistent total in chunk headers (%u vs %ld)",
              s->rx.hdr.total, (long)r->total.load());
    c->error.store(EPROTO);
    return -1;
  }
  if (s->rx.hdr.total > r->capacity) {
    BNET_WARN("bnet: message (%u B) exceeds posted buffer (%u B)",
              s->rx.hdr.total, r->capacity);
    c->error.store(EMSGSIZE);
    return -1;
  }
  if (s->rx.hdr.tag != r->tag) {
    // Matching is seq-ordinal (like NCCL's bundled socket transport); the
    // echoed tag VERIFIES that assumption — a cross-match is a loud
    // protocol error instead of silent data corruption.
    BNET_WARN("bnet: tag mismatch on seq=%u (sent %d, posted %d)", seq,
              s->rx.hdr.tag, r->tag);
    c->error.store(EPROTO);
    return -1;
  }
  s->rx.req = r;
  char* base = stage_recv_base(r);  // bounce for CUDA dst, else dst
  s->rx.target = base + s->rx.hdr.offset;
  s->rx.remaining = s->rx.hdr.len;
  s->rx.in_payload = true;
  return 0;
}

// Called with a fully-received chunk in s->rx.
void finish_rx_chunk(TcpSock* s) {
  RecvComm* c = s->rcomm;
  RecvRequest* r = s->rx.req;
  ChunkHdr h = s->rx.hdr;
  s->rx.in_payload = false;
  s->rx.hdr_got = 0;
  s->rx.req = nullptr;
  c->stats.bytes_recv.fetch_add(h.len, std::memory_order_relaxed);
  // ORDER MATTERS: issue this chunk's H2D copy BEFORE counting it, so when
  // any thread observes received == total, every chunk's copy has been
  // submitted and the completion event recorded after that covers them all.
  if (r->stage) stage_recv_issue(r, h.offset, h.len);
  uint32_t got = r->received.fetch_add(h.len, std::memory_order_acq_rel) +
                 h.len;
  if (r->stage && got == h.total) stage_recv_last(r);
}

// -------------------------------------------------------------- Engine ----

Engine::Engine() {
  int n = Config::get().io_threads;
  bool want_uring = Config::get().implement == "URING";
  if (want_uring && !uring_available()) {
    BNET_WARN("BNET_IMPLEMENT=URING but io_uring is unavailable here "
              "(seccomp?) — falling back to the epoll engine");
    want_uring = false;
  }
  impl_ = want_uring ? "URING" : "EPOLL";
  threads_.reserve(n);
  for (int i = 0; i < n; i++) {
    if (want_uring)
      threads_.emplace_back(make_uring_thread());
    else
      threads_.emplace_back(new IoThread());
    threads_.back()->start(i);
  }
  BNET_INFO("baguanet engine: %s, %d IO thread(s)", impl_, n);
}

Engine::~Engine() {
  for (auto& t : threads_) t->stop();
}

Engine& Engine::get() {
  static Engine e;
  return e;
}

int Engine::assign() {
  return rr_.fetch_add(1) % threads_.size();
}

void Engine::register_sock(TcpSock* s) {
  s->io_thread = assign();
  threads_[s->io_thread]->add_sock(s);
}

void Engine::unregister_sock_sync(TcpSock* s) {
  if (s->io_thread >= 0) threads_[s->io_thread]->remove_sock_sync(s);
}

void Engine::kick_comm(SendComm* c, int max_socks) {
  // Kick only IDLE sockets (snd_idle), up to the message's chunk count —
  // a busy socket keeps claiming work itself, and its idle transition is
  // covered by the publish-then-recheck handshake in progress_send /
  // submit_send (it re-scans after setting snd_idle, so either it finds
  // this request or this kick sees snd_idle).
  int n = max_socks < 0 ? (int)c->socks.size() : max_socks;
  for (TcpSock* s : c->socks) {
    if (n-- <= 0) break;
    threads_[s->io_thread]->kick(s);  // unconditional — see kick_sock
  }
}

void Engine::kick_sock(TcpSock* s) {
  // UNCONDITIONAL: the snd_idle gate was observed (rarely, ~1e-5 of
  // messages under soak) to miss the idle transition despite the
  // seq_cst publish-then-recheck pair, stalling a message until the
  // pre-block sweep.  A redundant kick to a busy socket is a cheap
  // no-op; correctness wins.
  threads_[s->io_thread]->kick(s);
}

void Engine::post_send(TcpSock* s) {
  // Proxy-inline first: the posting thread writes the bytes itself when
  // the socket's owner is idle, skipping the kick->eventfd->IO-thread
  // wake (~10-20 us).  Contention (or an engine without inline support)
  // falls back to the unconditional kick.
  if (threads_[s->io_thread]->try_inline_send(s)) return;
  threads_[s->io_thread]->kick(s);
}

void Engine::kick_comm(RecvComm* c) {
  // Only parked sockets need a wakeup; the park/post race is closed by
  // the seq_cst publish-then-recheck handshake (see progress_recv): the
  // parking thread re-checks the slot after publishing `parked`, and
  // irecv publishes the slot before loading `parked` here.
  for (TcpSock* s : c->socks)
    if (s->parked.load(std::memory_order_seq_cst))
      threads_[s->io_thread]->kick(s);
}

}  // namespace baguanet
