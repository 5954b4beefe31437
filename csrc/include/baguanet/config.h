// baguanet/config.h — env-var configuration, read once at init.
//
// Mirrors the reference's knob surface (BAGUA_NET_NSTREAMS,
// BAGUA_NET_MIN_CHUNKSIZE, BAGUA_NET_IMPLEMENT, NCCL_SOCKET_IFNAME/FAMILY —
// reference src/implement/nthread_per_socket_backend.rs:228-235,
// src/utils.rs:33-38) under the BNET_ prefix, plus the knobs this design
// adds (IO threads, dynamic chunking, staging, credit window).

#pragma once

#include <cstdint>
#include <string>

namespace baguanet {

struct Config {
  // Number of parallel data TCP streams per peer connection.
  // Reference default: 2 (nthread_per_socket_backend.rs:228-231).
  int nstreams = 4;
  // Lower bound on a stripe chunk; payload below this is not split.
  // Reference default: 1 MiB (nthread_per_socket_backend.rs:232-235).
  uint32_t min_chunk = 128 * 1024;
  // Upper bound on a stripe chunk — keeps dynamic stream assignment
  // fine-grained enough to balance (reference TODO at nthread:335).
  uint32_t max_chunk = 1 * 1024 * 1024;
  // Shared epoll IO threads servicing all comms' sockets.
  int io_threads = 4;
  // Cap on bytes in flight (queued to the kernel) per data socket before a
  // writer yields to other messages — the fairness/credit window.
  uint32_t inflight_per_stream = 4 * 1024 * 1024;
  // SO_SNDBUF/SO_RCVBUF request; 0 = kernel autotuning.
  int sockbuf = 0;
  // Enable NCCL_PTR_CUDA staging through pinned ring buffers (GPU present).
  bool cuda_ptr = true;
  // Pinned staging pool per comm direction, bytes.
  size_t stage_pool = 64ull * 1024 * 1024;
  // Process-wide cap on pinned staging memory across ALL comms (pools +
  // dedicated oversize allocations).  Pools shrink (down to 8 MiB) when
  // the budget runs low instead of pinning GBs on many-comm nodes.
  size_t pinned_budget = 2ull * 1024 * 1024 * 1024;
  // D2H/H2D pipeline chunk for staging copies.
  uint32_t stage_chunk = 512 * 1024;
  // Staging copy mode: 0 = SDMA (hipMemcpyAsync), 1 = pack kernel.
  int stage_kernel = 0;
  // Listen backlog (reference: 16384, nthread:101).
  int backlog = 16384;
  // Accepted connections must complete their WireHello within this
  // window or be reaped (dead client protection on long-lived listeners).
  uint32_t hello_timeout_ms = 30000;
  // Connector-side in-progress connect() state untouched for this long is
  // considered abandoned by RCCL and its sockets are reaped (0 = never).
  uint32_t connect_abandon_ms = 120000;
  // Process rank for telemetry labeling, from BNET_RANK or RANK (the
  // launcher env; cf. reference nthread:104-107); -1 = unknown.
  int rank = -1;
  // Busy-spin window after the last observed traffic before an IO
  // thread blocks, in microseconds.  Spinning keeps latency-critical
  // ping-pong patterns off the scheduler wake path.
  uint32_t spin_us = 200;
  // IO engine: "EPOLL" (default) or "URING" (io_uring; falls back to
  // epoll when unavailable).  The reference's BAGUA_NET_IMPLEMENT
  // BASIC/TOKIO selector, rebuilt as readiness- vs completion-based IO.
  std::string implement = "EPOLL";
  // Metrics dump file ("" = disabled); written at process exit.
  std::string metrics_file;
  // Live Prometheus pull endpoint on 127.0.0.1:(port + rank); 0 = off.
  // (The reference pushed to a gateway; this image has no egress, so the
  // live equivalent is scrape-style pull.)
  int metrics_port = 0;
  // Chrome-trace span dump file ("" = disabled).
  std::string trace_file;

  static const Config& get();  // parsed once, cached
};

}  // namespace baguanet
