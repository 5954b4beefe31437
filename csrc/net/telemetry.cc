#include "telemetry.h"

#include <arpa/inet.h>
#include <errno.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <string.h>
#include <sys/socket.h>
#include <time.h>
#include <unistd.h>

#include <cstdio>
#include <cstdlib>
#include <thread>

#include "baguanet/config.h"
#include "baguanet/log.h"
#include "transport.h"

namespace baguanet {

constexpr uint64_t Telemetry::kBounds[7];

uint64_t now_ns() {
  timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return (uint64_t)ts.tv_sec * 1000000000ull + ts.tv_nsec;
}

// Live pull endpoint: a detached thread serving the Prometheus text
// rendering over HTTP on 127.0.0.1:BNET_METRICS_PORT (the reference
// PUSHED to a gateway every 200 us, nthread:183-211; this image has no
// egress, so scrape-style pull is the live equivalent).  Multi-rank jobs
// use port+rank so every rank stays scrapable.
static void metrics_server(int port) {
  int fd = socket(AF_INET, SOCK_STREAM | SOCK_CLOEXEC, 0);
  if (fd < 0) return;
  int one = 1;
  setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  sockaddr_in a{};
  a.sin_family = AF_INET;
  a.sin_port = htons((uint16_t)port);
  a.sin_addr.s_addr = htonl(INADDR_LOOPBACK);
  if (bind(fd, (sockaddr*)&a, sizeof(a)) < 0 || listen(fd, 16) < 0) {
    BNET_WARN("metrics endpoint bind(127.0.0.1:%d) failed: %s", port,
              strerror(errno));
    close(fd);
    return;
  }
  BNET_INFO("metrics endpoint live on 127.0.0.1:%d", port);
  while (true) {
    int c = accept(fd, nullptr, nullptr);
    if (c < 0) {
      if (errno == EINTR) continue;
      break;
    }
    char req[1024];
    (void)!read(c, req, sizeof(req));  // drain the request line
    char* body = nullptr;
    size_t blen = 0;
    FILE* mem = open_memstream(&body, &blen);
    if (mem) {
      Telemetry::get().render_metrics(mem);
      fclose(mem);
      char hdr[160];
      int hn = snprintf(hdr, sizeof(hdr),
                        "HTTP/1.0 200 OK\r\nContent-Type: text/plain; "
                        "version=0.0.4\r\nContent-Length: %zu\r\n\r\n",
                        blen);
      (void)!write(c, hdr, hn);
      (void)!write(c, body, blen);
      free(body);
    }
    close(c);
  }
  close(fd);
}

Telemetry& Telemetry::get() {
  static Telemetry* t = [] {
    auto* p = new Telemetry();
    p->spans_on = !Config::get().trace_file.empty();
    atexit([] {
      const Config& c = Config::get();
      if (!c.metrics_file.empty())
        Telemetry::get().dump_metrics(c.metrics_file.c_str());
      if (!c.trace_file.empty())
        Telemetry::get().dump_trace(c.trace_file.c_str());
    });
    if (Config::get().metrics_port > 0) {
      int port = Config::get().metrics_port +
                 (Config::get().rank > 0 ? Config::get().rank : 0);
      std::thread(metrics_server, port).detach();
    }
    return p;
  }();
  return *t;
}

void Telemetry::hist_add(std::atomic<uint64_t>* h, uint64_t bytes) {
  int i = 0;
  while (i < 7 && bytes > kBounds[i]) i++;
  h[i].fetch_add(1, std::memory_order_relaxed);
}

uint32_t Telemetry::span_begin(uint8_t kind, uint64_t comm, uint32_t seq,
                               uint32_t nbytes) {
  if (!spans_on) return UINT32_MAX;
  uint32_t slot = span_next.fetch_add(1, std::memory_order_relaxed) %
                  kSpanCap;
  Span& s = spans[slot];
  s.t0 = now_ns();
  s.t1 = 0;
  s.comm = comm;
  s.seq = seq;
  s.nbytes = nbytes;
  s.kind = kind;
  return slot;
}

void Telemetry::span_end(uint32_t slot) {
  if (slot == UINT32_MAX) return;
  spans[slot % kSpanCap].t1 = now_ns();
}

void Telemetry::dump_metrics(const char* path) {
  FILE* f = fopen(path, "w");
  if (!f) return;
  render_metrics(f);
  fclose(f);
}

void Telemetry::render_metrics(FILE* f) {
  // rank label (reference pushed Prometheus with a `rank` label,
  // nthread:183-211) — empty when the launcher exported no RANK
  char rl[32] = "";
  int rank = Config::get().rank;
  if (rank >= 0) snprintf(rl, sizeof(rl), "{rank=\"%d\"}", rank);
  fprintf(f, "# baguanet transport metrics (prometheus text format)\n");
  fprintf(f, "bnet_isend_total%s %llu\n", rl,
          (unsigned long long)isend_count.load());
  fprintf(f, "bnet_irecv_total%s %llu\n", rl,
          (unsigned long long)irecv_count.load());
  fprintf(f, "bnet_bytes_sent_total%s %llu\n", rl,
          (unsigned long long)bytes_sent.load());
  fprintf(f, "bnet_bytes_recv_total%s %llu\n", rl,
          (unsigned long long)bytes_recv.load());
  fprintf(f, "bnet_send_comms_total%s %llu\n", rl,
          (unsigned long long)send_comms.load());
  fprintf(f, "bnet_recv_comms_total%s %llu\n", rl,
          (unsigned long long)recv_comms.load());
  fprintf(f, "bnet_staged_d2h_bytes_total%s %llu\n", rl,
          (unsigned long long)staged_d2h_bytes.load());
  fprintf(f, "bnet_staged_h2d_bytes_total%s %llu\n", rl,
          (unsigned long long)staged_h2d_bytes.load());
  char rli[32] = "";  // histogram label prefix merging with `le`
  if (rank >= 0) snprintf(rli, sizeof(rli), "rank=\"%d\",", rank);
  const char* names[2] = {"bnet_isend_nbytes", "bnet_irecv_nbytes"};
  std::atomic<uint64_t>* hists[2] = {isend_hist, irecv_hist};
  for (int h = 0; h < 2; h++) {
    uint64_t cum = 0;
    for (int i = 0; i < 7; i++) {
      cum += hists[h][i].load();
      fprintf(f, "%s_bucket{%sle=\"%llu\"} %llu\n", names[h], rli,
              (unsigned long long)kBounds[i], (unsigned long long)cum);
    }
    cum += hists[h][7].load();
    fprintf(f, "%s_bucket{%sle=\"+Inf\"} %llu\n", names[h], rli,
            (unsigned long long)cum);
  }
}

void Telemetry::dump_trace(const char* path) {
  FILE* f = fopen(path, "w");
  if (!f) return;
  fprintf(f, "[\n");
  uint32_t n = span_next.load();
  uint32_t count = n < kSpanCap ? n : kSpanCap;
  bool first = true;
  for (uint32_t i = 0; i < count; i++) {
    const Span& s = spans[i];
    if (s.t1 == 0 || s.t1 < s.t0) continue;
    if (!first) fprintf(f, ",\n");
    first = false;
    fprintf(f,
            "{\"name\":\"%s seq=%u %uB\",\"ph\":\"X\",\"pid\":%d,"
            "\"tid\":%llu,\"ts\":%.3f,\"dur\":%.3f}",
            s.kind == 0 ? "isend" : "irecv", s.seq, s.nbytes,
            Config::get().rank < 0 ? 0 : Config::get().rank,
            (unsigned long long)(s.comm & 0xffff), s.t0 / 1000.0,
            (s.t1 - s.t0) / 1000.0);
  }
  fprintf(f, "\n]\n");
  fclose(f);
}

}  // namespace baguanet

extern "C" {
// Live config snapshot as JSON (tests verify env knobs are honored).
__attribute__((visibility("default"))) int bnet_config_json(char* buf,
                                                            int len) {
  const baguanet::Config& c = baguanet::Config::get();
  return snprintf(buf, (size_t)len,
                  "{\"nstreams\":%d,\"min_chunk\":%u,\"max_chunk\":%u,"
                  "\"io_threads\":%d,\"sockbuf\":%d,\"cuda_ptr\":%d,"
                  "\"stage_pool\":%zu,\"stage_chunk\":%u,"
                  "\"stage_kernel\":%d,\"backlog\":%d,\"spin_us\":%u,"
                  "\"hello_timeout_ms\":%u,\"connect_abandon_ms\":%u,"
                  "\"rank\":%d,\"implement\":\"%s\"}",
                  c.nstreams, c.min_chunk, c.max_chunk, c.io_threads,
                  c.sockbuf, (int)c.cuda_ptr, c.stage_pool, c.stage_chunk,
                  c.stage_kernel, c.backlog, c.spin_us, c.hello_timeout_ms,
                  c.connect_abandon_ms, c.rank, c.implement.c_str());
}

// Debug: dump a recv comm's request slots + socket rx state (stall
// triage from test harnesses).
__attribute__((visibility("default"))) int bnet_dump_recv_state(
    void* recv_comm, char* buf, int len) {
  using namespace baguanet;
  auto* c = (RecvComm*)recv_comm;
  int off = 0;
  off += snprintf(buf + off, len - off,
                  "post_next=%u refusal@%u=(%u,%u) slots:", c->post_next,
                  c->last_refusal_at, ss_seq(c->last_refusal_ss),
                  ss_state(c->last_refusal_ss));
  for (int i = 0; i < NCCL_NET_MAX_REQUESTS; i++) {
    auto& r = c->reqs[i];
    uint64_t ss = r.state_seq.load();
    if (ss_state(ss) != REQ_ACTIVE) continue;
    off += snprintf(buf + off, len - off,
                    " [%d@%p seq=%u total=%ld recvd=%u stage=%d done_sock=%d]",
                    i, (void*)((uintptr_t)&r | 2),
                    ss_seq(ss), (long)r.total.load(), r.received.load(),
                    r.stage ? 1 : 0, r.socket_complete() ? 1 : 0);
    if (off >= len - 128) break;
  }
  off += snprintf(buf + off, len - off, " socks:");
  for (auto* s : c->socks) {
    off += snprintf(buf + off, len - off,
                    " {idx=%d parked=%d inpay=%d hdrgot=%u rem=%u seq=%u "
                    "op=%d epollin=%d}",
                    s->idx, (int)s->parked.load(), (int)s->rx.in_payload,
                    s->rx.hdr_got, s->rx.remaining, s->rx.hdr.seq,
                    (int)s->ur.op, (int)s->epollin_on);
    if (off >= len - 160) break;
  }
  return off;
}

__attribute__((visibility("default"))) int bnet_dump_send_state(
    void* send_comm, char* buf, int len) {
  using namespace baguanet;
  auto* c = (SendComm*)send_comm;
  int off = 0;
  off += snprintf(buf + off, len - off,
                  "oldest=%u next=%u refusal@%u=(%u,%u) slots:",
                  c->oldest.load(), c->seq_next, c->last_refusal_at,
                  ss_seq(c->last_refusal_ss), ss_state(c->last_refusal_ss));
  for (int i = 0; i < NCCL_NET_MAX_REQUESTS; i++) {
    auto& r = c->reqs[i];
    uint64_t ss = r.state_seq.load();
    if (ss_state(ss) != REQ_ACTIVE) continue;
    off += snprintf(buf + off, len - off,
                    " [%d seq=%u total=%u chunk=%u cur=%u/g%u avail=%u "
                    "sent=%u]",
                    i, ss_seq(ss), r.total.load(), r.chunk.load(),
                    cur_off(r.cursor.load()), cur_gen(r.cursor.load()),
                    r.avail.load(), r.sent.load());
    if (off >= len - 128) break;
  }
  off += snprintf(buf + off, len - off, " socks:");
  for (auto* s : c->socks) {
    off += snprintf(buf + off, len - off,
                    " {idx=%d txact=%d txdone=%u txlen=%u wantout=%d "
                    "sndidle=%d op=%d nch=%d kicks=%u prog=%u claims=%u "
                    "brk=%u full=%u brkS=%u brkSS=%u/%u brkOld=%u "
                    "gfail=%u ffail=%u exit=%u reent=%u krun=%u rescue=%u}",
                    s->idx, (int)s->tx.active, s->tx.done, s->tx.hdr.len,
                    (int)s->want_epollout, (int)s->snd_idle.load(),
                    (int)s->ur.op, s->ur.nchunks, s->dbg_kicks.load(),
                    s->dbg_progress.load(), s->dbg_claims.load(),
                    s->dbg_breaks.load(), s->dbg_full.load(),
                    s->dbg_break_s.load(),
                    ss_seq(s->dbg_break_ss.load()),
                    ss_state(s->dbg_break_ss.load()),
                    s->dbg_break_old.load(), s->dbg_gate_fail.load(),
                    s->dbg_find_fail.load(), s->dbg_exit.load(),
                    s->dbg_reent.load(), s->dbg_kick_run.load(),
                    s->dbg_sweep_rescue.load());
    if (off >= len - 160) break;
  }
  return off;
}

// Force-kick every socket of a send comm (stall-recovery probe).
__attribute__((visibility("default"))) void bnet_force_kick(void* send_comm) {
  using namespace baguanet;
  auto* c = (SendComm*)send_comm;
  for (auto* s : c->socks)
    Engine::get().thread(s->io_thread).kick(s);
}

__attribute__((visibility("default"))) void bnet_dump_metrics(const char* path) {
  baguanet::Telemetry::get().dump_metrics(path);
}
__attribute__((visibility("default"))) void bnet_dump_trace(const char* path) {
  baguanet::Telemetry::get().dump_trace(path);
}
}
