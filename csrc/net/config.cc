#include "baguanet/config.h"

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
    c.stage_chunk = (uint32_t)env_long("BNET_STAGE_CHUNK", c.stage_chunk);
    if (c.stage_chunk < 65536) c.stage_chunk = 65536;
    c.stage_kernel = (int)env_long("BNET_STAGE_KERNEL", 0);
    c.backlog = (int)env_long("BNET_BACKLOG", c.backlog);
    c.spin_us = (uint32_t)env_long("BNET_SPIN_US", c.spin_us);
    c.hello_timeout_ms = (uint32_t)env_long("BNET_HELLO_TIMEOUT_MS",
                                            c.hello_timeout_ms);
    c.implement = env_str("BNET_IMPLEMENT", "EPOLL");
    c.metrics_file = env_str("BNET_METRICS_FILE", "");
    c.trace_file = env_str("BNET_TRACE_FILE", "");
    return c;
  }();
  return cfg;
}

}  // namespace baguanet
