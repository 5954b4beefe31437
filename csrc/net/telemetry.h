// telemetry.h — metrics counters + per-request span tracing.
//
// Equivalent of the reference's cross-cutting telemetry (SURVEY §2.1 R6:
// OpenTelemetry Jaeger spans per isend/irecv + Prometheus push metrics,
// reference nthread:108-211).  This image has no network, so export is
// file-based and pull-style instead of push:
//   * BNET_METRICS_FILE — Prometheus text-format counters, written at
//     process exit (and on demand via bnet_dump_metrics()).
//   * BNET_TRACE_FILE   — Chrome trace-event JSON of isend/irecv spans
//     (id + nbytes, start at post, end at completion — the same span
//     structure as the reference's isend-{comm}/irecv-{comm} spans).

#pragma once

#include <atomic>
#include <cstdint>
#include <cstdio>

namespace baguanet {

uint64_t now_ns();

struct Telemetry {
  std::atomic<uint64_t> isend_count{0}, irecv_count{0};
  std::atomic<uint64_t> bytes_sent{0}, bytes_recv{0};
  std::atomic<uint64_t> send_comms{0}, recv_comms{0};
  std::atomic<uint64_t> staged_d2h_bytes{0}, staged_h2d_bytes{0};
  // histogram boundaries (bytes): reference used {16,1024,4096,1048576}
  // (nthread:139-141); extended upward for modern message sizes
  static constexpr uint64_t kBounds[7] = {16,      1024,     4096,    65536,
                                          1048576, 16777216, 134217728};
  std::atomic<uint64_t> isend_hist[8]{};
  std::atomic<uint64_t> irecv_hist[8]{};

  // span ring (lock-free, overwrites oldest)
  struct Span {
    uint64_t t0 = 0, t1 = 0;
    uint64_t comm = 0;
    uint32_t seq = 0;
    uint32_t nbytes = 0;
    uint8_t kind = 0;  // 0=isend 1=irecv
  };
  static constexpr uint32_t kSpanCap = 1 << 15;
  Span spans[kSpanCap];
  std::atomic<uint32_t> span_next{0};
  bool spans_on = false;

  static Telemetry& get();
  void hist_add(std::atomic<uint64_t>* h, uint64_t bytes);
  uint32_t span_begin(uint8_t kind, uint64_t comm, uint32_t seq,
                      uint32_t nbytes);
  void span_end(uint32_t slot);
  void dump_metrics(const char* path);
  void render_metrics(FILE* f);  // Prometheus text (file or live endpoint)
  void dump_trace(const char* path);
};

}  // namespace baguanet

extern "C" {
// on-demand dumps + config introspection (used by tests via ctypes)
void bnet_dump_metrics(const char* path);
void bnet_dump_trace(const char* path);
int bnet_config_json(char* buf, int len);
int bnet_dump_recv_state(void* recv_comm, char* buf, int len);
int bnet_dump_send_state(void* send_comm, char* buf, int len);
}
