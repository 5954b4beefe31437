#include "baguanet/log.h"

#include <cstring>

namespace baguanet {

ncclDebugLogger_t g_logger = nullptr;
int g_log_level = [] {
  const char* v = getenv("BNET_LOG");
  if (!v) return 1;  // warn by default
  if (!strcmp(v, "off")) return 0;
  if (!strcmp(v, "info")) return 2;
  if (!strcmp(v, "trace")) return 3;
  return 1;
}();

void log_impl(ncclDebugLogLevel level, unsigned long subsys, const char* file,
              int line, const char* fmt, ...) {
  va_list ap;
  if (g_logger) {
    va_start(ap, fmt);
    // ncclDebugLogger_t is itself variadic; forward by formatting first.
    char buf[1024];
    vsnprintf(buf, sizeof(buf), fmt, ap);
    va_end(ap);
    g_logger(level, subsys, file, line, "%s", buf);
    return;
  }
  int want = level == NCCL_LOG_WARN ? 1 : level == NCCL_LOG_INFO ? 2 : 3;
  if (g_log_level < want) return;
  va_start(ap, fmt);
  fprintf(stderr, "[baguanet %s %s:%d] ",
          level == NCCL_LOG_WARN ? "WARN"
          : level == NCCL_LOG_INFO ? "INFO"
                                   : "TRACE",
          file, line);
  vfprintf(stderr, fmt, ap);
  fprintf(stderr, "\n");
  va_end(ap);
}

}  // namespace baguanet
