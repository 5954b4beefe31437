// baguanet/log.h — logging through RCCL's debug logger with stderr fallback.
//
// The reference logged through the captured ncclDebugLogger_t on the C++
// side (reference: cc/v4/nccl_net_v4.cc:13-16) and `tracing` on the Rust
// side; here there is one native layer, so one logger.

#pragma once

#include <cstdarg>
#include <cstdio>
#include <cstdlib>

#include "baguanet/nccl_abi.h"

namespace baguanet {

// Set once at plugin init; safe to read unsynchronized afterwards.
extern ncclDebugLogger_t g_logger;
extern int g_log_level;  // fallback stderr level: 0=off 1=warn 2=info 3=trace

void log_impl(ncclDebugLogLevel level, unsigned long subsys, const char* file,
              int line, const char* fmt, ...) __attribute__((format(printf, 5, 6)));

#define BNET_WARN(...) \
  ::baguanet::log_impl(NCCL_LOG_WARN, NCCL_ALL, __FILE__, __LINE__, __VA_ARGS__)
#define BNET_INFO(...) \
  ::baguanet::log_impl(NCCL_LOG_INFO, NCCL_NET, __FILE__, __LINE__, __VA_ARGS__)
#define BNET_TRACE(...) \
  ::baguanet::log_impl(NCCL_LOG_TRACE, NCCL_NET, __FILE__, __LINE__, __VA_ARGS__)

}  // namespace baguanet
