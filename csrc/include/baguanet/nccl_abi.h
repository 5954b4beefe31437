// baguanet/nccl_abi.h — vendored RCCL/NCCL network-plugin ABI definitions.
//
// RCCL (ROCm 7.x, /opt/rocm/lib/librccl.so.1) dlopens `libnccl-net.so` (or
// `libnccl-net-<NCCL_NET_PLUGIN>.so`) and probes the exported symbols
// `ncclNetPlugin_v10` … `ncclNetPlugin_v6`.  The installed
// /opt/rocm/include/rccl/nccl_net.h #includes `net/net_v6.h`…`net_v10.h`,
// but those sub-headers are NOT shipped in the image, so — exactly as the
// reference vendored its nccl types (reference: cc/nccl_types.h:1-3,
// cc/v4/nccl_net_v4.h) — we vendor the public, stable plugin ABI here,
// self-contained.  Only what the plugin actually needs is defined.
//
// Constants below (HANDLE_MAXSIZE=128, MAX_REQUESTS=32, NCCL_PTR_DMABUF)
// match the installed /opt/rocm/include/rccl/nccl_net.h and differ from the
// reference's NCCL-2.6-era values (64 / 8; reference cc/nccl_types.h:44,50).

#pragma once

#include <cstddef>
#include <cstdint>

// ---------------------------------------------------------------------------
// Core result / logging types (public ABI, from nccl.h / nccl_common.h)
// ---------------------------------------------------------------------------

typedef enum {
  ncclSuccess = 0,
  ncclUnhandledCudaError = 1,
  ncclSystemError = 2,
  ncclInternalError = 3,
  ncclInvalidArgument = 4,
  ncclInvalidUsage = 5,
  ncclRemoteError = 6,
  ncclInProgress = 7,
  ncclNumResults = 8
} ncclResult_t;

typedef enum {
  NCCL_LOG_NONE = 0,
  NCCL_LOG_VERSION = 1,
  NCCL_LOG_WARN = 2,
  NCCL_LOG_INFO = 3,
  NCCL_LOG_ABORT = 4,
  NCCL_LOG_TRACE = 5
} ncclDebugLogLevel;

typedef enum {
  NCCL_INIT = 0x1,
  NCCL_COLL = 0x2,
  NCCL_P2P = 0x4,
  NCCL_SHM = 0x8,
  NCCL_NET = 0x10,
  NCCL_GRAPH = 0x20,
  NCCL_TUNING = 0x40,
  NCCL_ENV = 0x80,
  NCCL_ALLOC = 0x100,
  NCCL_CALL = 0x200,
  NCCL_PROXY = 0x400,
  NCCL_NVLS = 0x800,
  NCCL_ALL = ~0
} ncclDebugLogSubSys;

typedef void (*ncclDebugLogger_t)(ncclDebugLogLevel level, unsigned long flags,
                                  const char* file, int line, const char* fmt,
                                  ...);

// ---------------------------------------------------------------------------
// Net plugin constants (match installed /opt/rocm/include/rccl/nccl_net.h)
// ---------------------------------------------------------------------------

#define NCCL_NET_HANDLE_MAXSIZE 128
#define NCCL_PTR_HOST 0x1
#define NCCL_PTR_CUDA 0x2
#define NCCL_PTR_DMABUF 0x4
#define NCCL_NET_MAX_REQUESTS 32

// Device-side networking (GPUDirect-style in-kernel networking) is not used
// by this plugin: it declares NCCL_NET_DEVICE_HOST (all networking done by
// the host proxy), which every RCCL version accepts.
#define NCCL_NET_DEVICE_HOST 0x0
#define NCCL_NET_DEVICE_INVALID_VERSION 0x0

typedef struct {
  int netDeviceType;     // NCCL_NET_DEVICE_HOST for host-proxy plugins
  int netDeviceVersion;  // NCCL_NET_DEVICE_INVALID_VERSION
  void* handle;
  size_t size;
  int needsProxyProgress;
} ncclNetDeviceHandle_v7_t;

typedef ncclNetDeviceHandle_v7_t ncclNetDeviceHandle_v8_t;

// ---------------------------------------------------------------------------
// v6 ABI (NCCL >= 2.13 layout of v6, incl. regMrDmaBuf) — what we export.
// RCCL probes v10..v6 and wraps older versions internally; v6 is the
// simplest fully-supported surface.
// ---------------------------------------------------------------------------

typedef struct {
  char* name;      // Used mostly for logging
  char* pciPath;   // Path to the PCI device in /sys
  uint64_t guid;   // Unique identifier for the NIC chip
  int ptrSupport;  // [NCCL_PTR_HOST|NCCL_PTR_CUDA|NCCL_PTR_DMABUF]
  int speed;       // Port speed in Mbps
  int port;        // Port number
  float latency;   // Network latency
  int maxComms;    // Maximum number of comms we can create
  int maxRecvs;    // Maximum number of grouped receives
} ncclNetProperties_v6_t;

typedef struct {
  // Name of the network (mainly for logs)
  const char* name;
  // Initialize the network.
  ncclResult_t (*init)(ncclDebugLogger_t logFunction);
  // Return the number of adapters.
  ncclResult_t (*devices)(int* ndev);
  // Get various device properties.
  ncclResult_t (*getProperties)(int dev, ncclNetProperties_v6_t* props);
  // Create a receiving object and provide a handle to connect to it. The
  // handle can be up to NCCL_NET_HANDLE_MAXSIZE bytes and will be exchanged
  // between ranks to create a connection.
  ncclResult_t (*listen)(int dev, void* handle, void** listenComm);
  // Connect to a handle and return a sending comm object for that peer.
  // This call must not block for the connection to be established, and
  // instead should return successfully with sendComm == NULL with the
  // expectation that it will be called again until sendComm != NULL.
  ncclResult_t (*connect)(int dev, void* handle, void** sendComm);
  // Finalize connection establishment after remote peer has called connect.
  // This call must not block for the connection to be established, and
  // instead should return successfully with recvComm == NULL with the
  // expectation that it will be called again until recvComm != NULL.
  ncclResult_t (*accept)(void* listenComm, void** recvComm);
  // Register/Deregister memory.
  ncclResult_t (*regMr)(void* comm, void* data, int size, int type,
                        void** mhandle);
  // DMA-BUF support
  ncclResult_t (*regMrDmaBuf)(void* comm, void* data, size_t size, int type,
                              uint64_t offset, int fd, void** mhandle);
  ncclResult_t (*deregMr)(void* comm, void* mhandle);
  // Asynchronous send to a peer.  May return request == NULL if the call
  // cannot be performed (or would block).
  ncclResult_t (*isend)(void* sendComm, void* data, int size, int tag,
                        void* mhandle, void** request);
  // Asynchronous recv from a peer.  May return request == NULL if the call
  // cannot be performed (or would block).
  ncclResult_t (*irecv)(void* recvComm, int n, void** data, int* sizes,
                        int* tags, void** mhandles, void** request);
  // Perform a flush/fence to make sure all data received with NCCL_PTR_CUDA
  // is visible to the GPU.
  ncclResult_t (*iflush)(void* recvComm, int n, void** data, int* sizes,
                         void** mhandles, void** request);
  // Test whether a request is complete. If size is not NULL, it returns the
  // number of bytes sent/received.
  ncclResult_t (*test)(void* request, int* done, int* sizes);
  // Close and free send/recv comm objects
  ncclResult_t (*closeSend)(void* sendComm);
  ncclResult_t (*closeRecv)(void* recvComm);
  ncclResult_t (*closeListen)(void* listenComm);
} ncclNet_v6_t;

// ---------------------------------------------------------------------------
// v7 ABI (NCCL 2.18 era): properties gain netDeviceType/netDeviceVersion;
// connect/accept gain a device-handle out-param (NULL from host-proxy
// plugins); getDeviceMr/irecvConsumed appended for device-offload plugins.
// These layouts are frozen public ABI — identical across NCCL and RCCL
// (third-party plugins such as aws-ofi are compiled against the same
// structs for both stacks).
// ---------------------------------------------------------------------------

typedef struct {
  char* name;
  char* pciPath;
  uint64_t guid;
  int ptrSupport;  // [NCCL_PTR_HOST|NCCL_PTR_CUDA|NCCL_PTR_DMABUF]
  int speed;       // Mbps
  int port;
  float latency;
  int maxComms;
  int maxRecvs;
  int netDeviceType;     // ncclNetDeviceType (NCCL_NET_DEVICE_HOST)
  int netDeviceVersion;
} ncclNetProperties_v7_t;

typedef struct {
  const char* name;
  ncclResult_t (*init)(ncclDebugLogger_t logFunction);
  ncclResult_t (*devices)(int* ndev);
  ncclResult_t (*getProperties)(int dev, ncclNetProperties_v7_t* props);
  ncclResult_t (*listen)(int dev, void* handle, void** listenComm);
  ncclResult_t (*connect)(int dev, void* handle, void** sendComm,
                          ncclNetDeviceHandle_v7_t** sendDevComm);
  ncclResult_t (*accept)(void* listenComm, void** recvComm,
                         ncclNetDeviceHandle_v7_t** recvDevComm);
  ncclResult_t (*regMr)(void* comm, void* data, int size, int type,
                        void** mhandle);
  ncclResult_t (*regMrDmaBuf)(void* comm, void* data, size_t size, int type,
                              uint64_t offset, int fd, void** mhandle);
  ncclResult_t (*deregMr)(void* comm, void* mhandle);
  ncclResult_t (*isend)(void* sendComm, void* data, int size, int tag,
                        void* mhandle, void** request);
  ncclResult_t (*irecv)(void* recvComm, int n, void** data, int* sizes,
                        int* tags, void** mhandles, void** request);
  ncclResult_t (*iflush)(void* recvComm, int n, void** data, int* sizes,
                         void** mhandles, void** request);
  ncclResult_t (*test)(void* request, int* done, int* sizes);
  ncclResult_t (*closeSend)(void* sendComm);
  ncclResult_t (*closeRecv)(void* recvComm);
  ncclResult_t (*closeListen)(void* listenComm);
  // device-offload hooks (host-proxy plugins return ncclInternalError)
  ncclResult_t (*getDeviceMr)(void* comm, void* mhandle, void** dptr_mhandle);
  ncclResult_t (*irecvConsumed)(void* recvComm, int n, void* request);
} ncclNet_v7_t;

// ---------------------------------------------------------------------------
// v8 ABI (NCCL 2.19-2.21 era): properties gain regIsGlobal; regMr's size
// widens to size_t.  Everything else matches v7.
// ---------------------------------------------------------------------------

typedef struct {
  char* name;
  char* pciPath;
  uint64_t guid;
  int ptrSupport;
  int regIsGlobal;  // regMr()s are valid for all comms (we say no: the
                    // mhandle is just the pointer type, but per-comm is the
                    // conservative answer)
  int speed;
  int port;
  float latency;
  int maxComms;
  int maxRecvs;
  int netDeviceType;
  int netDeviceVersion;
} ncclNetProperties_v8_t;

typedef struct {
  const char* name;
  ncclResult_t (*init)(ncclDebugLogger_t logFunction);
  ncclResult_t (*devices)(int* ndev);
  ncclResult_t (*getProperties)(int dev, ncclNetProperties_v8_t* props);
  ncclResult_t (*listen)(int dev, void* handle, void** listenComm);
  ncclResult_t (*connect)(int dev, void* handle, void** sendComm,
                          ncclNetDeviceHandle_v8_t** sendDevComm);
  ncclResult_t (*accept)(void* listenComm, void** recvComm,
                         ncclNetDeviceHandle_v8_t** recvDevComm);
  ncclResult_t (*regMr)(void* comm, void* data, size_t size, int type,
                        void** mhandle);
  ncclResult_t (*regMrDmaBuf)(void* comm, void* data, size_t size, int type,
                              uint64_t offset, int fd, void** mhandle);
  ncclResult_t (*deregMr)(void* comm, void* mhandle);
  ncclResult_t (*isend)(void* sendComm, void* data, int size, int tag,
                        void* mhandle, void** request);
  ncclResult_t (*irecv)(void* recvComm, int n, void** data, int* sizes,
                        int* tags, void** mhandles, void** request);
  ncclResult_t (*iflush)(void* recvComm, int n, void** data, int* sizes,
                         void** mhandles, void** request);
  ncclResult_t (*test)(void* request, int* done, int* sizes);
  ncclResult_t (*closeSend)(void* sendComm);
  ncclResult_t (*closeRecv)(void* recvComm);
  ncclResult_t (*closeListen)(void* listenComm);
  ncclResult_t (*getDeviceMr)(void* comm, void* mhandle, void** dptr_mhandle);
  ncclResult_t (*irecvConsumed)(void* recvComm, int n, void* request);
} ncclNet_v8_t;
