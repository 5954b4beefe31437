// staging.h — GPU<->host staging for NCCL_PTR_CUDA payloads.
//
// The reference plugin only supported host pointers and let NCCL's proxy
// stage GPU data into NCCL-owned pinned buffers (reference
// cc/v4/nccl_net_v4.cc:105-109 rejects NCCL_PTR_CUDA).  Here the plugin
// advertises NCCL_PTR_CUDA and owns the staging hot path, MI355X-style:
//
//   send:  HBM src --(chunk-pipelined hipMemcpyAsync, dedicated
//          side stream)--> pinned ring slot --(socket writers, started per
//          chunk as its copy event lands)--> TCP.  Socket writes overlap
//          the remaining D2H copies.
//   recv:  TCP --> pinned ring slot --(per-chunk hipMemcpyAsync H2D as each
//          chunk finishes on its socket)--> HBM dst; the request completes
//          when the last copy's event lands, so iflush is trivially
//          satisfied (returns a NULL request).
//
// Copies use the SDMA copy engines via hipMemcpyAsync by default;
// BNET_STAGE_KERNEL=1 switches to the hand-written CDNA4 pack kernels in
// csrc/hip/pack_kernels.hip (vectorized uint4 copies) — both are
// benchmarked in tests/gpu and profiles/.

#pragma once

#include <cstdint>

namespace baguanet {

struct SendRequest;
struct RecvRequest;
class StagePool;

// True if a HIP device is present and staging is enabled by config.
bool staging_available();

// ---- pool lifecycle (per comm, lazily created on first CUDA request) ----
// Pool size is clamped by the process-wide BNET_PINNED_BUDGET (halving to
// an 8 MiB floor); *retry_later distinguishes a temporarily-exhausted
// budget (caller should retry: closing comms frees budget) from a hard
// hipHostMalloc failure.
//
// Messages larger than the pool take a DEDICATED budget-accounted pinned
// allocation (one cached per pool) instead of failing.  Deliberately not a
// modular windowed ring: RCCL's proxy never posts messages beyond a few
// MiB (its net messages are bounded by buffSize/NCCL_STEPS slices), so
// oversize is a configuration-robustness path — a simple contiguous
// fallback is preferred over wrap-around offset mapping plus sender/
// receiver window pacing inside the transport hot path, which round-1
// soak testing showed is the most race-sensitive code in the plugin.
StagePool* stage_pool_create(bool* retry_later = nullptr);
void stage_pool_destroy(StagePool* p);

// ---- send path ----
// Begin staging `total` bytes from GPU `src` for `req`.  Sets req->src to a
// pinned bounce region and req->stage; req->avail advances as chunk copies
// complete (via stage_poll).  Returns false if pool space is exhausted
// (caller should make isend return request=NULL so NCCL retries).
bool stage_send_begin(StagePool* p, SendRequest* req, const void* src,
                      uint32_t total);

// Advance watermarks of all in-flight send stagings (event queries).
// Returns true if anything advanced.
bool stage_poll(StagePool* p);

// Any send staging still in flight? (IO threads spin while true.)
bool stage_pending(StagePool* p);

// Request-level watchdog called from test(): if req's staging has not
// advanced for >2 s (alloc missing from the poll list, wedged stream,
// errored event), log the pool/alloc state and re-stage the remaining
// copies, re-registering the alloc for polling.
void stage_send_watchdog(StagePool* p, SendRequest* req);

// ---- recv path ----
// Reserve a pinned bounce of `capacity` bytes for a CUDA recv.  Returns
// false if pool space is exhausted.
bool stage_recv_begin(StagePool* p, RecvRequest* req, void* dst,
                      uint32_t capacity);

// Where socket readers should land bytes for `req` (bounce for CUDA,
// req->dst for plain host recvs).
char* stage_recv_base(RecvRequest* req);

// A chunk [offset, offset+len) just finished on a socket: issue its H2D
// copy.  Callers must issue BEFORE counting the chunk as received.
void stage_recv_issue(RecvRequest* req, uint32_t offset, uint32_t len);

// All chunks are issued (received == total): record the completion event.
void stage_recv_last(RecvRequest* req);

// True when all H2D copies for the request have drained (event landed).
bool stage_recv_done(RecvRequest* req);

// Release a request's staging allocation (send or recv).
void stage_release(StagePool* p, SendRequest* req);
void stage_release(StagePool* p, RecvRequest* req);

}  // namespace baguanet
