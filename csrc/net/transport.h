// transport.h — MI355X-native multi-stream TCP transport core.
//
// This is the from-scratch equivalent of the reference's Rust core
// (trait Net, reference src/interface.rs:34-74; BASIC backend
// src/implement/nthread_per_socket_backend.rs), redesigned rather than
// translated:
//
//   * Self-describing chunk frames {seq, offset, len, total} on every data
//     socket replace the reference's ctrl-socket length frames + fixed
//     chunk->stream mapping (reference nthread:395-413).  Receivers scatter
//     by offset, so senders may assign chunks to streams DYNAMICALLY
//     (least-loaded / work-stealing) — the reference's own TODO
//     (nthread:335) — and the extra ctrl connection disappears.
//   * A shared epoll IO-thread pool replaces one-OS-thread-per-socket with
//     spin-on-EWOULDBLOCK (reference utils.rs:132-178).
//   * Fixed lock-free request slots (NCCL_NET_MAX_REQUESTS=32 per comm)
//     replace the global Arc<Mutex<dyn Net>> taken on every isend/test
//     (reference src/lib.rs:15, :312).
//   * connect/accept are nonblocking state machines, as required by the
//     ncclNet_v6+ contract (the reference's blocking v4 accept,
//     nthread:433-447, would deadlock the RCCL proxy).
//   * NCCL_PTR_CUDA payloads are staged through pinned ring buffers with
//     chunk-pipelined hipMemcpyAsync on side streams (staging.h) — the GPU
//     path the reference delegated to NCCL.
//
// Wire endianness is x86-64 native (little-endian); the cluster is
// homogeneous.

#pragma once

#include <netinet/in.h>
#include <sys/socket.h>
#include <sys/uio.h>

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <memory>
#include <mutex>
#include <thread>
#include <vector>

#include "baguanet/config.h"
#include "baguanet/nccl_abi.h"
#include "ifdiscovery.h"

namespace baguanet {

class StagePool;    // staging.h
struct StageAlloc;  // staging.h

// ---------------------------------------------------------------- wire ----

constexpr uint32_t kMagic = 0xBA60A4E7;
constexpr uint32_t kWireVersion = 2;  // v2: ChunkHdr carries the NCCL tag

struct WireHello {  // connector -> acceptor, once per data socket
  uint32_t magic;
  uint32_t version;
  uint64_t conn_id;    // groups the nstreams sockets of one connect()
  uint16_t stream_id;  // 0..nstreams-1
  uint16_t nstreams;
  uint32_t reserved;
};
static_assert(sizeof(WireHello) == 24, "wire layout");

struct ChunkHdr {  // precedes every payload chunk on a data socket
  uint32_t seq;     // per-comm message sequence number
  uint32_t offset;  // byte offset of this chunk within the message
  uint32_t len;     // chunk payload bytes (0 only for empty messages)
  uint32_t total;   // total message bytes
  int32_t tag;      // NCCL isend tag, echoed so the receiver can verify
                    // seq-ordinal matching against the posted irecv tag
                    // (a silent cross-match becomes a loud EPROTO)
};
static_assert(sizeof(ChunkHdr) == 20, "wire layout");

// NCCL handle (<= NCCL_NET_HANDLE_MAXSIZE = 128 bytes).  `stage` stashes the
// connector-side in-progress state across nonblocking connect() retries
// (NCCL keeps the handle buffer stable between calls).
struct ListenHandle {
  uint32_t magic;
  uint16_t family;  // AF_INET / AF_INET6
  uint16_t port;    // network byte order
  uint8_t addr[16];
  uint64_t stage;        // ConnectTask* — validated against a live-task
  uint64_t stage_token;  // registry + this conn_id before any dereference
};
static_assert(sizeof(ListenHandle) <= NCCL_NET_HANDLE_MAXSIZE, "handle size");

// ------------------------------------------------------------ requests ----

enum ReqState : uint32_t { REQ_FREE = 0, REQ_ACTIVE = 1 };

// Grouped-receive ceiling advertised as properties.maxRecvs (the proxy
// aggregates small recvs when this is > 1; NCCL's IB plugin uses 8).
// Each group member consumes one of the NCCL_NET_MAX_REQUESTS seq slots.
constexpr int kMaxRecvs = 4;

// state+seq packed into one atomic word so observers can never see a torn
// (state, seq) pair.  Without this, a claim scan could read stale
// state=FREE together with a freshly-written new seq during slot re-posting
// and misclassify an in-flight post as "completed", advancing the send
// frontier past a request that was never transmitted.
inline uint64_t pack_ss(uint32_t seq, uint32_t state) {
  return ((uint64_t)seq << 32) | state;
}
inline uint32_t ss_state(uint64_t ss) { return (uint32_t)ss; }
inline uint32_t ss_seq(uint64_t ss) { return (uint32_t)(ss >> 32); }

// The claim cursor is generation-tagged the same way: (seq << 32) | offset.
// A claimer validates state_seq, but between that check and its cursor CAS
// the slot can complete -> free -> repost (observed under soak: a stale
// CAS claimed a 32 KiB chunk of a freshly-posted 5000 B message,
// corrupting the stream and wedging completion).  With the generation in
// the CAS word, a cross-generation CAS can never commit; the repost bumps
// the cursor generation BEFORE writing any other field.
inline uint64_t pack_cur(uint32_t seq, uint32_t off) {
  return ((uint64_t)seq << 32) | off;
}
inline uint32_t cur_off(uint64_t c) { return (uint32_t)c; }
inline uint32_t cur_gen(uint64_t c) { return (uint32_t)(c >> 32); }

struct SendComm;
struct RecvComm;

struct SendRequest {
  std::atomic<uint64_t> state_seq{pack_ss(UINT32_MAX, REQ_FREE)};
  // total/chunk are atomics ONLY because a stale claimer may read them
  // while the slot is being reposted for the next generation (the
  // gen-tagged cursor CAS then rejects the claim, so the value is
  // discarded — but the concurrent plain read would still be UB; TSan
  // flags it).  All accesses are relaxed; ordering comes from the
  // state_seq publish.
  std::atomic<uint32_t> total{0};
  std::atomic<uint32_t> chunk{0};  // stripe chunk size for this message
  const char* src = nullptr;    // host source (user buffer or staging bounce)
  // generation-tagged claim cursor: (seq << 32) | next_unclaimed_offset.
  // For total == 0 the "offset" doubles as the header-claim flag (0 -> 1).
  std::atomic<uint64_t> cursor{0};
  std::atomic<uint32_t> avail{0};   // staged watermark; == total for host src
  std::atomic<uint32_t> sent{0};    // bytes fully handed to the kernel
  std::atomic<bool> hdr_sent{false};
  int tag = 0;  // NCCL tag, echoed in every ChunkHdr (written before the
                // state_seq publish; read only after a confirmed claim)
  StageAlloc* stage = nullptr;  // non-null for NCCL_PTR_CUDA sends
  SendComm* comm = nullptr;
  uint32_t span_slot = UINT32_MAX;

  bool complete() const {
    uint32_t t = total.load(std::memory_order_relaxed);
    return t == 0 ? hdr_sent.load(std::memory_order_acquire)
                  : sent.load(std::memory_order_acquire) == t;
  }
};

struct RecvRequest {
  std::atomic<uint64_t> state_seq{pack_ss(UINT32_MAX, REQ_FREE)};
  char* dst = nullptr;     // user destination (host) — staging writes here
  uint32_t capacity = 0;   // posted buffer size (recv may be smaller)
  int tag = 0;             // posted NCCL tag; checked against ChunkHdr.tag
  // Grouped receive (irecv n>1, maxRecvs): the FIRST member records the
  // group size; members occupy n consecutive seq slots and the returned
  // request completes when all members do.  0/1 = ungrouped.
  uint8_t group_n = 0;
  std::atomic<int64_t> total{-1};      // from first chunk header
  std::atomic<uint32_t> received{0};   // socket bytes landed
  std::atomic<bool> gpu_done{false};   // H2D staging drained (CUDA dst)
  StageAlloc* stage = nullptr;         // bounce buffer for CUDA recv
  RecvComm* comm = nullptr;
  uint32_t span_slot = UINT32_MAX;

  bool socket_complete() const {
    int64_t t = total.load(std::memory_order_acquire);
    return t >= 0 && received.load(std::memory_order_acquire) == (uint64_t)t;
  }
};

// Completion token for synchronous engine-task handshakes (socket
// removal).  shared_ptr ownership: the waiter and the IO thread each hold
// a reference, so neither side can destroy the cv while the other is
// still inside wait()/notify_all() (a stack-allocated cv here was a
// TSan-caught destroy-during-broadcast race).
struct SyncToken {
  std::mutex mu;
  std::condition_variable cv;
  bool done = false;
  void signal() {
    {
      std::lock_guard<std::mutex> lk(mu);
      done = true;
    }
    cv.notify_all();
  }
  void wait() {
    std::unique_lock<std::mutex> lk(mu);
    cv.wait(lk, [this] { return done; });
  }
};

// -------------------------------------------------------------- socket ----

struct TcpSock {
  int fd = -1;
  int io_thread = -1;
  int idx = 0;  // position within its comm's socket list
  bool is_recv = false;
  bool want_epollout = false;
  bool epollin_on = false;  // diagnostic mirror of the epoll interest
  // stall-triage counters
  std::atomic<uint32_t> dbg_kicks{0};      // kick() enqueued for this sock
  std::atomic<uint32_t> dbg_progress{0};   // progress_send/submit entries
  std::atomic<uint32_t> dbg_claims{0};     // successful chunk claims
  std::atomic<uint32_t> dbg_breaks{0};     // scans ended via break
  std::atomic<uint32_t> dbg_full{0};       // scans ended at window end
  std::atomic<uint32_t> dbg_break_s{0};    // last break: scan seq
  std::atomic<uint64_t> dbg_break_ss{0};   // last break: slot state_seq
  std::atomic<uint32_t> dbg_break_old{0};  // last break: oldest at entry
  std::atomic<uint32_t> dbg_gate_fail{0};  // kick_sock gate saw idle=false
  std::atomic<uint32_t> dbg_find_fail{0};  // KICK task: sock not in socks_
  std::atomic<uint32_t> dbg_exit{0};       // progress_send last exit point
  std::atomic<int> dbg_inprog{0};          // reentrancy detector
  std::atomic<uint32_t> dbg_reent{0};      // progress_send reentered!
  std::atomic<uint32_t> dbg_kick_run{0};   // KICK tasks executed for sock
  std::atomic<uint32_t> dbg_sweep_rescue{0};  // pre-block sweep found work
  // send-side idle flag for the kick fast path: set (seq_cst) by the IO
  // thread right before its final failed claim re-check; isend publishes
  // the request (seq_cst + fence) then kicks only sockets with snd_idle
  // set — same Dekker publish-then-recheck handshake as recv parking
  std::atomic<bool> snd_idle{false};
  // proxy-inline-send exclusion: every progress_send entry (IO-thread
  // event/kick/sweep AND the posting thread's inline attempt from isend)
  // must win owner_busy first.  The loser marks rekick; the winner re-runs
  // via kick() after releasing, so no transition is lost.  Removes the
  // kick->eventfd->IO-thread hop (~10-20 us) from the small-message path —
  // the same trick as NCCL's own net_socket inline sends.
  std::atomic<int> owner_busy{0};
  std::atomic<bool> rekick{false};
  // io_uring engine per-socket state (unused by the epoll engine)
  static constexpr int kUrBatch = 8;  // chunks per WRITEV submission
  struct {
    // 0 none, 1 send(writev), 2 recv.  Atomic because the proxy's inline
    // send reads it under owner_busy while the engine thread transitions
    // it: an armed op (op==1) also serves as the CQE handler's mutual
    // exclusion — the proxy bails whenever op != 0, so the handler may
    // mutate ur.* freely until it clears op (release) as its last step.
    std::atomic<uint8_t> op{0};
    bool closing = false;  // removal in progress: no resubmission
    bool eof = false;      // recv: orderly peer shutdown — never re-arm
    // batched send: up to kUrBatch chunks in one ordered WRITEV
    int nchunks = 0;
    ChunkHdr hdrs[8];
    const char* payloads[8];
    SendRequest* reqs[8];
    uint32_t batch_bytes = 0;  // sum of (16 + len) over the batch
    uint32_t done = 0;         // bytes of the batch written so far
    struct iovec iov[16];
  } ur;
  std::atomic<bool> parked{false};  // recv: waiting for a not-yet-posted seq
  SendComm* scomm = nullptr;
  RecvComm* rcomm = nullptr;

  // TX state: one chunk in flight (header + payload), partial-write capable.
  struct {
    bool active = false;
    ChunkHdr hdr{};
    const char* payload = nullptr;
    uint32_t done = 0;  // bytes of (16 + len) virtual stream written
    SendRequest* req = nullptr;
  } tx;

  // RX state machine.
  struct {
    bool in_payload = false;
    ChunkHdr hdr{};
    uint32_t hdr_got = 0;
    char* target = nullptr;  // dst+offset or bounce+offset
    uint32_t remaining = 0;
    RecvRequest* req = nullptr;
  } rx;
};

// --------------------------------------------------------------- comms ----

struct CommStats {
  std::atomic<uint64_t> isend_count{0}, irecv_count{0};
  std::atomic<uint64_t> bytes_sent{0}, bytes_recv{0};
};

struct SendComm {
  std::vector<TcpSock*> socks;
  SendRequest reqs[NCCL_NET_MAX_REQUESTS];
  uint32_t seq_next = 0;         // proxy thread only
  uint64_t last_refusal_ss = 0;
  uint32_t last_refusal_at = 0;
  std::atomic<uint32_t> oldest{0};  // completion frontier (lazily advanced)
  std::atomic<int> error{0};
  std::atomic<int> live_socks{0};
  int dev = 0;
  StagePool* stage_pool = nullptr;  // lazily created for CUDA sends
  CommStats stats;
};

struct RecvComm {
  std::vector<TcpSock*> socks;
  RecvRequest reqs[NCCL_NET_MAX_REQUESTS];
  uint32_t post_next = 0;  // proxy thread only
  uint64_t last_refusal_ss = 0;  // blocking slot's state_seq at refusal
  uint32_t last_refusal_at = 0;  // post_next at refusal
  std::atomic<int> error{0};
  std::atomic<int> live_socks{0};
  int dev = 0;
  StagePool* stage_pool = nullptr;
  CommStats stats;
};

// -------------------------------------------------- connection states -----

struct ConnectTask {  // connector side, lives in ListenHandle::stage
  int dev = 0;
  uint64_t conn_id = 0;
  uint64_t last_touch_ns = 0;  // abandoned-task reaping (connect registry)
  sockaddr_storage peer{};
  socklen_t peer_len = 0;
  struct Pending {
    int fd = -1;
    bool connected = false;
    uint32_t hello_sent = 0;
  };
  std::vector<Pending> socks;
};

struct ListenComm {
  int fd = -1;
  int dev = 0;
  // sockets accepted but whose WireHello is not fully read yet
  struct HalfConn {
    int fd;
    WireHello hello;
    uint32_t got = 0;
    uint64_t t0_ns = 0;  // accept time — stalled hellos are reaped
  };
  std::vector<HalfConn> half;
  // complete hellos grouped by conn_id
  struct Group {
    uint64_t conn_id;
    uint16_t nstreams;
    std::vector<int> fds;  // indexed by stream_id, -1 = missing
    int have = 0;
  };
  std::vector<Group> groups;
};

// -------------------------------------------------------------- engine ----

// Shared transport state-machine helpers (used by both engines).
SendRequest* claim_chunk(SendComm* c, int sock_idx, uint32_t* off,
                         uint32_t* len, TcpSock* dbg = nullptr);
enum RxResult { RX_WAIT = 0, RX_PARKED = 1, RX_CLOSED = 2 };
// Greedy nonblocking drain of the rx state machine (see transport.cc).
int drain_recv(TcpSock* s);
// Any posted recv still waiting for socket bytes?  (A fully-received
// request that merely awaits test()/H2D drain does NOT make peer EOF an
// error.)
bool recv_socket_incomplete(RecvComm* c);
// rx.hdr is complete: locate/validate the posted request and prime
// rx.target/remaining.  Returns 0 = proceed, 1 = park (request not yet
// posted), -1 = protocol error (comm error set).
int process_recv_header(TcpSock* s);
// A fully-received chunk in s->rx: account + staging hand-off.
void finish_rx_chunk(TcpSock* s);

class IIoThread {
 public:
  virtual ~IIoThread() = default;
  virtual void start(int idx) = 0;
  virtual void stop() = 0;
  virtual void add_sock(TcpSock* s) = 0;          // thread-safe
  virtual void remove_sock_sync(TcpSock* s) = 0;  // blocks until removed
  virtual void kick(TcpSock* s) = 0;              // re-run progress for s
  // Attempt the first send progress inline on the CALLING thread (the
  // NCCL proxy posting an isend).  Returns false if the engine does not
  // support it or the socket's owner is busy — caller falls back to
  // kick().  Only valid for send sockets.
  virtual bool try_inline_send(TcpSock* s) { (void)s; return false; }
};

class IoThread : public IIoThread {
 public:
  void start(int idx) override;
  void stop() override;
  void add_sock(TcpSock* s) override;
  void remove_sock_sync(TcpSock* s) override;
  void kick(TcpSock* s) override;
  bool try_inline_send(TcpSock* s) override;

 private:
  void run();
  void progress(TcpSock* s);
  void progress_send(TcpSock* s);
  void progress_recv(TcpSock* s);
  void set_epollout(TcpSock* s, bool on);
  void handle_tasks();

  int idx_ = 0;
  int epfd_ = -1, evfd_ = -1;
  std::thread thr_;
  std::atomic<bool> stop_{false};
  std::mutex task_mu_;
  struct Task {
    enum { ADD, REMOVE, KICK } kind;
    TcpSock* s;
    std::shared_ptr<SyncToken> tok;  // REMOVE only
  };
  std::vector<Task> tasks_;
  std::vector<TcpSock*> socks_;  // owned set (IO thread only)
};

// io_uring-based engine thread (uring_engine.cc); selected with
// BNET_IMPLEMENT=URING, falls back to epoll when io_uring is unavailable
// (e.g. seccomp-filtered containers).
IIoThread* make_uring_thread();
bool uring_available();

class Engine {
 public:
  static Engine& get();
  IIoThread& thread(int idx) { return *threads_[idx]; }
  int assign();  // round-robin IO thread index
  int nthreads() const { return (int)threads_.size(); }
  void register_sock(TcpSock* s);
  void unregister_sock_sync(TcpSock* s);
  void kick_comm(SendComm* c, int max_socks = -1);
  void kick_comm(RecvComm* c);
  void kick_sock(TcpSock* s);
  // inline-first send from the posting thread; falls back to kick_sock
  void post_send(TcpSock* s);
  const char* impl() const { return impl_; }

 private:
  Engine();
  ~Engine();
  std::vector<std::unique_ptr<IIoThread>> threads_;
  std::atomic<uint32_t> rr_{0};
  const char* impl_ = "EPOLL";
};

// ------------------------------------------------------------- net API ----
// The plugin-facing API (plugin.cc adapts it to the ncclNet_v6 vtable).

struct NetDevice {
  NetIf nif;
  ncclNetProperties_v6_t props;  // name/pciPath point into `nif`/strings
  std::string name_str;
};

class Net {
 public:
  static Net& get();
  int ndev() const { return (int)devs_.size(); }
  ncclResult_t get_properties(int dev, ncclNetProperties_v6_t* props);
  ncclResult_t listen(int dev, void* handle, void** listen_comm);
  ncclResult_t connect(int dev, void* handle, void** send_comm);
  ncclResult_t accept(void* listen_comm, void** recv_comm);
  ncclResult_t isend(void* send_comm, void* data, int size, int tag,
                     void* mhandle, void** request);
  ncclResult_t irecv(void* recv_comm, int n, void** data, int* sizes,
                     int* tags, void** mhandles, void** request);
  ncclResult_t iflush(void* recv_comm, int n, void** data, int* sizes,
                      void** mhandles, void** request);
  ncclResult_t test(void* request, int* done, int* sizes);
  ncclResult_t close_send(void* send_comm);
  ncclResult_t close_recv(void* recv_comm);
  ncclResult_t close_listen(void* listen_comm);
  int ptr_support() const { return ptr_support_; }

 private:
  Net();
  std::vector<NetDevice> devs_;
  int ptr_support_ = NCCL_PTR_HOST;
};

// Chunk-size policy: stripe into roughly nstreams equal parts, clamped to
// [min_chunk, max_chunk] (cf. reference utils.rs:200-205 `chunk_size`, which
// had no upper clamp — the max keeps dynamic assignment balanced).
uint32_t pick_chunk_size(uint32_t total, uint32_t min_chunk,
                         uint32_t max_chunk, int nstreams);

}  // namespace baguanet
