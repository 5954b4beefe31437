#include "baguanet/config.h"

#include <cstdio>
#include <cstdlib>

namespace baguanet {

static long env_long(const char* name, long dflt) {
  const char* v = getenv(name);
  if (!v || !*v) return dflt;
  char* end = nullptr;
  long x = strtol(v, &end, 10);
  return end && *end == '\0' ? x : dflt;
}

static std::string env_str(const char* name, const char* dflt) {
  const char* v = getenv(name);
  return v ? v : dflt;
}

const Config& Config::get() {
  static Config cfg = [] {
    Config c;
    c.nstreams = (int)env_long("BNET_NSTREAMS", c.nstreams);
    if (c.nstreams < 1) c.nstreams = 1;
    if (c.nstreams > 64) c.nstreams = 64;
    c.min_chunk = (uint32_t)env_long("BNET_MIN_CHUNKSIZE", c.min_chunk);
    c.max_chunk = (uint32_t)env_long("BNET_MAX_CHUNKSIZE", c.max_chunk);
    if (c.min_chunk < 4096) c.min_chunk = 4096;
    if (c.max_chunk < c.min_chunk) c.max_chunk = c.min_chunk;
    c.io_threads = (int)env_long("BNET_IO_THREADS", c.io_threads);
    if (c.io_threads < 1) c.io_threads = 1;
    if (c.io_threads > 32) c.io_threads = 32;
    c.inflight_per_stream =
        (uint32_t)env_long("BNET_INFLIGHT", c.inflight_per_stream);
    c.sockbuf = (int)env_long("BNET_SOCKBUF", c.sockbuf);
    c.cuda_ptr = env_long("BNET_CUDA_PTR", 1) != 0;
    c.stage_pool = (size_t)env_long("BNET_STAGE_POOL", (long)c.stage_pool);
    c.pinned_budget =
        (size_t)env_long("BNET_PINNED_BUDGET", (long)c.pinned_budget);
    c.stage_chunk = (uint32_t)env_long("BNET_STAGE_CHUNK", c.stage_chunk);
    if (c.stage_chunk < 65536) c.stage_chunk = 65536;
    c.stage_kernel = (int)env_long("BNET_STAGE_KERNEL", 0);
    c.backlog = (int)env_long("BNET_BACKLOG", c.backlog);
    c.spin_us = (uint32_t)env_long("BNET_SPIN_US", c.spin_us);
    c.hello_timeout_ms = (uint32_t)env_long("BNET_HELLO_TIMEOUT_MS",
                                            c.hello_timeout_ms);
    c.connect_abandon_ms = (uint32_t)env_long("BNET_CONNECT_ABANDON_MS",
                                              c.connect_abandon_ms);
    c.implement = env_str("BNET_IMPLEMENT", "EPOLL");
    c.rank = (int)env_long("BNET_RANK", env_long("RANK", -1));
    c.metrics_file = env_str("BNET_METRICS_FILE", "");
    c.metrics_port = (int)env_long("BNET_METRICS_PORT", 0);
    c.trace_file = env_str("BNET_TRACE_FILE", "");
    // Rank-template the dump paths so N-rank jobs leave N distinguishable
    // files instead of overwriting one another: "%r" substitutes the rank;
    // with no template and a known rank, ".r<rank>" is appended.
    auto rankify = [&](std::string& path) {
      if (path.empty()) return;
      auto pos = path.find("%r");
      char buf[16];
      snprintf(buf, sizeof(buf), "%d", c.rank < 0 ? 0 : c.rank);
      if (pos != std::string::npos)
        path = path.substr(0, pos) + buf + path.substr(pos + 2);
      else if (c.rank >= 0)
        path += std::string(".r") + buf;
    };
    rankify(c.metrics_file);
    rankify(c.trace_file);
    return c;
  }();
  return cfg;
}

}  // namespace baguanet
