// staging.cc — pinned-ring staging implementation (see staging.h).

#include "staging.h"

#include "freelist.h"

#include <hip/hip_runtime.h>
#include <rocprofiler-sdk-roctx/roctx.h>

#include <algorithm>
#include <mutex>
#include <vector>

#include "baguanet/config.h"
#include "baguanet/log.h"
#include "telemetry.h"
#include "transport.h"

namespace baguanet {

// pack_kernels.hip — vectorized CDNA4 copy kernels (BNET_STAGE_KERNEL=1).
void launch_copy_kernel(void* dst, const void* src, size_t bytes,
                        hipStream_t stream);

#define HIP_WARN(call)                                             \
  do {                                                             \
    hipError_t e_ = (call);                                        \
    if (e_ != hipSuccess)                                          \
      BNET_WARN("%s failed: %s", #call, hipGetErrorString(e_));    \
  } while (0)

// ---- process-wide pinned-memory budget ------------------------------------
// Per-comm pools and dedicated oversize allocations all draw from one cap
// (BNET_PINNED_BUDGET): an 8-GPU node with many channels x peers must not
// pin GBs unboundedly.
static std::atomic<size_t> g_pinned{0};

static bool pinned_reserve(size_t n) {
  size_t budget = Config::get().pinned_budget;
  size_t cur = g_pinned.load(std::memory_order_relaxed);
  while (true) {
    if (cur + n > budget) return false;
    if (g_pinned.compare_exchange_weak(cur, cur + n)) return true;
  }
}

static void pinned_release(size_t n) {
  g_pinned.fetch_sub(n, std::memory_order_relaxed);
}

struct StageAlloc {
  char* host = nullptr;
  size_t pool_off = 0;
  bool dedicated = false;  // oversize: own hipHostMalloc, not from the arena
  size_t ded_cap = 0;      // pinned bytes behind a dedicated allocation
  uint32_t size = 0;
  uint64_t t_begin_ns = 0;   // for slow-event diagnostics
  bool diag_logged = false;
  // send pipeline
  const char* gpu_src = nullptr;
  uint32_t total = 0;
  uint32_t copy_chunk = 0;
  std::vector<hipEvent_t> events;
  uint32_t events_done = 0;
  SendRequest* sreq = nullptr;
  // recv pipeline
  char* gpu_dst = nullptr;
  StagePool* pool = nullptr;
  // published (release) only AFTER hipEventRecord — readers load acquire;
  // a cached/reused event queried before its re-record reports complete,
  // so visibility must imply recorded
  std::atomic<hipEvent_t> done_ev{nullptr};
};

class StagePool {
 public:
  char* base = nullptr;
  size_t size = 0;
  int device = 0;
  hipStream_t d2h = nullptr, h2d = nullptr;
  std::mutex mu;
  FreeList arena{0};
  std::vector<hipEvent_t> ev_cache;
  std::vector<StageAlloc*> inflight;  // send stagings with copies pending
  std::atomic<int> pending{0};
  // one cached dedicated buffer for oversize messages (> pool size), so a
  // soak of oversize messages does not hipHostMalloc per message
  char* big_cache = nullptr;
  size_t big_cache_size = 0;
  bool warned_oversize = false;

  // Oversize path: take the cached buffer if it fits, else allocate a
  // dedicated pinned buffer under the global budget.  Returns nullptr if
  // the budget cannot fit it right now (caller retries); *cap is the
  // actual pinned allocation size (what free_dedicated must release).
  char* alloc_dedicated(size_t sz, size_t* cap) {
    if (big_cache && big_cache_size >= sz) {
      char* b = big_cache;
      *cap = big_cache_size;
      big_cache = nullptr;
      big_cache_size = 0;
      return b;
    }
    if (!pinned_reserve(sz)) return nullptr;
    char* b = nullptr;
    hipError_t e = hipHostMalloc((void**)&b, sz, hipHostMallocDefault);
    if (e != hipSuccess) {
      BNET_WARN("oversize hipHostMalloc(%zu) failed: %s", sz,
                hipGetErrorString(e));
      pinned_release(sz);
      return nullptr;
    }
    *cap = sz;
    return b;
  }

  void free_dedicated(char* b, size_t sz) {
    if (!big_cache) {  // keep the largest one around for reuse
      big_cache = b;
      big_cache_size = sz;
      return;
    }
    if (big_cache_size < sz) {
      std::swap(big_cache, b);
      std::swap(big_cache_size, sz);
    }
    (void)hipHostFree(b);
    pinned_release(sz);
  }

  hipEvent_t get_event() {
    if (!ev_cache.empty()) {
      hipEvent_t e = ev_cache.back();
      ev_cache.pop_back();
      return e;
    }
    hipEvent_t e = nullptr;
    HIP_WARN(hipEventCreateWithFlags(&e, hipEventDisableTiming));
    return e;
  }
  void put_event(hipEvent_t e) { ev_cache.push_back(e); }

  char* alloc(uint32_t sz, size_t* off_out) {
    size_t off = arena.alloc(sz);
    if (off == SIZE_MAX) return nullptr;
    *off_out = off;
    return base + off;
  }
  void free(size_t off, uint32_t sz) { arena.free(off, sz); }
};

bool staging_available() {
  if (!Config::get().cuda_ptr) return false;
  static int avail = -1;
  if (avail < 0) {
    int n = 0;
    avail = (hipGetDeviceCount(&n) == hipSuccess && n > 0) ? 1 : 0;
  }
  return avail == 1;
}

StagePool* stage_pool_create(bool* retry_later) {
  if (retry_later) *retry_later = false;
  auto* p = new StagePool();
  // Budget-clamped pool: shrink (halving, floor 8 MiB) instead of blowing
  // the process-wide pinned cap when many comms are live.  A pool takes at
  // most HALF the remaining budget so its peer direction (and later comms)
  // can still get one — a send pool grabbing the whole budget would
  // livelock its own recv side behind endless NCCL retries.
  constexpr size_t kFloor = 8ull * 1024 * 1024;
  size_t budget = Config::get().pinned_budget;
  size_t used = g_pinned.load(std::memory_order_relaxed);
  size_t half_left = budget > used ? (budget - used) / 2 : 0;
  size_t want =
      std::min(Config::get().stage_pool, std::max(half_left, kFloor));
  while (!pinned_reserve(want)) {
    if (want <= kFloor) {
      BNET_WARN("pinned budget exhausted (%zu in use of %zu) — staging "
                "pool creation deferred; closing comms frees budget",
                g_pinned.load(), Config::get().pinned_budget);
      delete p;
      if (retry_later) *retry_later = true;
      return nullptr;
    }
    want /= 2;
  }
  if (want < Config::get().stage_pool)
    BNET_INFO("staging pool clamped to %zu MiB by BNET_PINNED_BUDGET",
              want >> 20);
  p->size = want;
  hipError_t e = hipHostMalloc((void**)&p->base, p->size, hipHostMallocDefault);
  if (e != hipSuccess) {
    BNET_WARN("hipHostMalloc(%zu) failed: %s", p->size, hipGetErrorString(e));
    pinned_release(want);
    delete p;
    return nullptr;
  }
  HIP_WARN(hipGetDevice(&p->device));
  HIP_WARN(hipStreamCreateWithFlags(&p->d2h, hipStreamNonBlocking));
  HIP_WARN(hipStreamCreateWithFlags(&p->h2d, hipStreamNonBlocking));
  p->arena = FreeList(p->size);
  return p;
}

void stage_pool_destroy(StagePool* p) {
  if (!p) return;
  if (p->d2h) (void)hipStreamSynchronize(p->d2h);
  if (p->h2d) (void)hipStreamSynchronize(p->h2d);
  for (auto e : p->ev_cache) (void)hipEventDestroy(e);
  if (p->d2h) (void)hipStreamDestroy(p->d2h);
  if (p->h2d) (void)hipStreamDestroy(p->h2d);
  if (p->base) (void)hipHostFree(p->base);
  pinned_release(p->size);
  if (p->big_cache) {
    (void)hipHostFree(p->big_cache);
    pinned_release(p->big_cache_size);
  }
  delete p;
}

static void issue_copy(void* dst, const void* src, size_t n, hipMemcpyKind k,
                       hipStream_t s) {
  if (Config::get().stage_kernel) {
    launch_copy_kernel(dst, src, n, s);
  } else {
    HIP_WARN(hipMemcpyAsync(dst, src, n, k, s));
  }
}

// roctx range markers make the staging pipeline visible in rocprofv3
// traces ("add rocprof markers around the HIP staging path").
static bool roctx_on() {
  static int on = [] {
    const char* v = getenv("BNET_ROCTX");
    return v && *v == '1' ? 1 : 0;
  }();
  return on == 1;
}

// Bounce-space acquisition shared by send/recv begin (pool mutex held):
// arena for messages that fit the pool, dedicated budget-accounted pinned
// allocation for oversize ones (no hard failure — the reference-era
// "exceeds staging pool" ncclInternalError is gone; see staging.h).
static bool acquire_bounce(StagePool* p, StageAlloc* a, uint32_t need) {
  if ((size_t)need > p->size) {
    if (!p->warned_oversize) {
      p->warned_oversize = true;
      BNET_WARN("staged message (%u B) exceeds the %zu MiB staging pool — "
                "using dedicated pinned allocations (raise BNET_STAGE_POOL "
                "to keep oversize messages in the ring)",
                need, p->size >> 20);
    }
    a->host = p->alloc_dedicated(need, &a->ded_cap);
    if (!a->host) return false;  // budget full right now — caller retries
    a->dedicated = true;
    return true;
  }
  a->host = p->alloc(need, &a->pool_off);
  return a->host != nullptr;
}

bool stage_send_begin(StagePool* p, SendRequest* req, const void* src,
                      uint32_t total) {
  if (roctx_on()) roctxRangePush("bnet_stage_send_d2h");
  std::lock_guard<std::mutex> lk(p->mu);
  auto* a = new StageAlloc();
  if (!acquire_bounce(p, a, std::max(total, 1u))) {
    delete a;
    if (roctx_on()) roctxRangePop();
    return false;
  }
  a->size = std::max(total, 1u);
  a->gpu_src = (const char*)src;
  a->total = total;
  a->copy_chunk = Config::get().stage_chunk;
  a->sreq = req;
  a->t_begin_ns = now_ns();
  req->src = a->host;
  req->stage = a;
  for (uint32_t off = 0; off < total; off += a->copy_chunk) {
    uint32_t n = std::min(a->copy_chunk, total - off);
    issue_copy(a->host + off, a->gpu_src + off, n, hipMemcpyDeviceToHost,
               p->d2h);
    Telemetry::get().staged_d2h_bytes.fetch_add(n, std::memory_order_relaxed);
    hipEvent_t ev = p->get_event();
    HIP_WARN(hipEventRecord(ev, p->d2h));
    a->events.push_back(ev);
  }
  if (total > 0) {
    p->inflight.push_back(a);
    p->pending.fetch_add(1, std::memory_order_release);
  }
  if (roctx_on()) roctxRangePop();
  return true;
}

bool stage_poll(StagePool* p) {
  if (p->pending.load(std::memory_order_acquire) == 0) return false;
  SendComm* advanced_comm = nullptr;
  bool advanced = false;
  {
    std::lock_guard<std::mutex> lk(p->mu);
    for (auto it = p->inflight.begin(); it != p->inflight.end();) {
      StageAlloc* a = *it;
      hipError_t qe = hipSuccess;
      while (a->events_done < a->events.size() &&
             (qe = hipEventQuery(a->events[a->events_done])) ==
                 hipSuccess) {
        a->events_done++;
        advanced = true;
        uint32_t avail =
            std::min<uint64_t>((uint64_t)a->events_done * a->copy_chunk,
                               a->total);
        a->sreq->avail.store(avail, std::memory_order_release);
        advanced_comm = a->sreq->comm;
      }
      if (a->events_done == a->events.size()) {
        for (auto e : a->events) p->put_event(e);
        a->events.clear();
        it = p->inflight.erase(it);
        p->pending.fetch_sub(1, std::memory_order_release);
      } else {
        // Slow-event diagnostics + one-shot recovery: a D2H copy of a
        // few hundred KB completes in microseconds; >2 s pending means
        // the query errors, the stream wedged, or the copy was lost.
        // Log the exact state once and re-issue the remaining copies.
        if (!a->diag_logged && now_ns() - a->t_begin_ns > 2'000'000'000ull) {
          a->diag_logged = true;
          hipError_t sq = hipStreamQuery(p->d2h);
          BNET_WARN(
              "staged D2H stuck >2s: total=%u done=%u/%zu query=%d(%s) "
              "streamQuery=%d(%s) — re-issuing remaining copies",
              a->total, a->events_done, a->events.size(), (int)qe,
              hipGetErrorString(qe), (int)sq, hipGetErrorString(sq));
          for (uint32_t i = a->events_done; i < a->events.size(); i++) {
            uint32_t off = i * a->copy_chunk;
            uint32_t n = std::min(a->copy_chunk, a->total - off);
            issue_copy(a->host + off, a->gpu_src + off, n,
                       hipMemcpyDeviceToHost, p->d2h);
            HIP_WARN(hipEventRecord(a->events[i], p->d2h));
          }
        }
        ++it;
      }
    }
  }
  // Watermark advances must WAKE idle claimer sockets: the advancing
  // thread is not necessarily the one hosting the owning socket, and once
  // `pending` hits zero other threads stop spin-retrying their senders —
  // without this kick a just-completed staged send could sit unclaimed
  // forever (observed as a rare soak stall).
  if (advanced_comm) Engine::get().kick_comm(advanced_comm);
  return advanced;
}

bool stage_pending(StagePool* p) {
  return p->pending.load(std::memory_order_acquire) > 0;
}

void stage_send_watchdog(StagePool* p, SendRequest* req) {
  StageAlloc* a = (StageAlloc*)req->stage;
  if (!a) return;
  std::lock_guard<std::mutex> lk(p->mu);
  if (req->avail.load(std::memory_order_acquire) >= a->total) return;
  if (now_ns() - a->t_begin_ns < 2'000'000'000ull) return;
  bool in_list = false;
  for (auto* x : p->inflight)
    if (x == a) in_list = true;
  hipError_t sq = hipStreamQuery(p->d2h);
  BNET_WARN(
      "staged send watchdog: total=%u avail=%u events=%zu done=%u "
      "in_inflight=%d pool_pending=%d streamQuery=%d(%s) — re-staging",
      a->total, req->avail.load(), a->events.size(), a->events_done,
      (int)in_list, p->pending.load(), (int)sq, hipGetErrorString(sq));
  // re-issue all remaining copies and re-register for polling
  uint32_t start = a->events_done;
  for (uint32_t i = start; i < a->events.size(); i++) {
    uint32_t off = i * a->copy_chunk;
    uint32_t n = std::min(a->copy_chunk, a->total - off);
    issue_copy(a->host + off, a->gpu_src + off, n, hipMemcpyDeviceToHost,
               p->d2h);
    HIP_WARN(hipEventRecord(a->events[i], p->d2h));
  }
  if (a->events.empty()) {
    // pathological: no events recorded at all — restage from scratch
    for (uint32_t off = 0; off < a->total; off += a->copy_chunk) {
      uint32_t n = std::min(a->copy_chunk, a->total - off);
      issue_copy(a->host + off, a->gpu_src + off, n, hipMemcpyDeviceToHost,
                 p->d2h);
      hipEvent_t ev = p->get_event();
      HIP_WARN(hipEventRecord(ev, p->d2h));
      a->events.push_back(ev);
    }
    a->events_done = 0;
  }
  if (!in_list) {
    p->inflight.push_back(a);
    p->pending.fetch_add(1, std::memory_order_release);
  }
  a->t_begin_ns = now_ns();  // rearm the watchdog
}

bool stage_recv_begin(StagePool* p, RecvRequest* req, void* dst,
                      uint32_t capacity) {
  std::lock_guard<std::mutex> lk(p->mu);
  auto* a = new StageAlloc();
  if (!acquire_bounce(p, a, std::max(capacity, 1u))) {
    delete a;
    return false;
  }
  a->size = std::max(capacity, 1u);
  a->gpu_dst = (char*)dst;
  a->pool = p;
  req->stage = a;
  return true;
}

char* stage_recv_base(RecvRequest* req) {
  return req->stage ? ((StageAlloc*)req->stage)->host : req->dst;
}

void stage_recv_issue(RecvRequest* req, uint32_t offset, uint32_t len) {
  StageAlloc* a = (StageAlloc*)req->stage;
  StagePool* p = a->pool;
  if (!len) return;
  if (roctx_on()) roctxMarkA("bnet_stage_recv_h2d");
  std::lock_guard<std::mutex> lk(p->mu);
  issue_copy(a->gpu_dst + offset, a->host + offset, len,
             hipMemcpyHostToDevice, p->h2d);
  Telemetry::get().staged_h2d_bytes.fetch_add(len, std::memory_order_relaxed);
}

void stage_recv_last(RecvRequest* req) {
  StageAlloc* a = (StageAlloc*)req->stage;
  StagePool* p = a->pool;
  hipEvent_t ev;
  {
    std::lock_guard<std::mutex> lk(p->mu);
    ev = p->get_event();
    HIP_WARN(hipEventRecord(ev, p->h2d));
  }
  a->done_ev.store(ev, std::memory_order_release);
}

bool stage_recv_done(RecvRequest* req) {
  StageAlloc* a = (StageAlloc*)req->stage;
  if (!a) return true;
  hipEvent_t ev = a->done_ev.load(std::memory_order_acquire);
  if (!ev) return false;  // last chunk not yet issued+recorded
  return hipEventQuery(ev) == hipSuccess;
}

void stage_release(StagePool* p, SendRequest* req) {
  StageAlloc* a = (StageAlloc*)req->stage;
  if (!a) return;
  std::lock_guard<std::mutex> lk(p->mu);
  if (a->dedicated)
    p->free_dedicated(a->host, a->ded_cap);
  else
    p->free(a->pool_off, a->size);
  req->stage = nullptr;
  delete a;
}

void stage_release(StagePool* p, RecvRequest* req) {
  StageAlloc* a = (StageAlloc*)req->stage;
  if (!a) return;
  std::lock_guard<std::mutex> lk(p->mu);
  hipEvent_t ev = a->done_ev.load(std::memory_order_relaxed);
  if (ev) p->put_event(ev);
  if (a->dedicated)
    p->free_dedicated(a->host, a->ded_cap);
  else
    p->free(a->pool_off, a->size);
  req->stage = nullptr;
  delete a;
}

// test hooks: process-wide pinned accounting (tests/test_gpu_plugin_cuda.py
// asserts the budget is respected across comm churn)
extern "C" {
__attribute__((visibility("default"))) void bnet_pinned_stats(
    size_t* used, size_t* budget) {
  if (used) *used = g_pinned.load(std::memory_order_relaxed);
  if (budget) *budget = Config::get().pinned_budget;
}
}

}  // namespace baguanet
