// plugin.cc — the exported ncclNetPlugin_v6 vtable (ABI adapter layer).
//
// Equivalent of the reference's cc/v4/nccl_net_v4.cc + cc/v3/nccl_net_v3.cc
// adapters, targeting the v6 ABI that this image's RCCL probes
// (librccl.so.1 looks for ncclNetPlugin_v6..v10).  RCCL wraps a
// v6 plugin internally for its newer call paths.

#include "baguanet/log.h"
#include "baguanet/nccl_abi.h"
#include "staging.h"
#include "telemetry.h"
#include "transport.h"

namespace {

using baguanet::Net;

ncclResult_t bnInit(ncclDebugLogger_t logFunction) {
  baguanet::g_logger = logFunction;
  Net::get();  // construct: NIC discovery, staging probe
  baguanet::Telemetry::get();  // registers the atexit metric/trace dumps
  BNET_INFO("baguanet (MI355X-native multi-stream TCP transport) loaded: "
            "%d device(s), ptrSupport 0x%x",
            Net::get().ndev(), Net::get().ptr_support());
  return ncclSuccess;
}

ncclResult_t bnDevices(int* ndev) {
  *ndev = Net::get().ndev();
  return ncclSuccess;
}

ncclResult_t bnGetProperties(int dev, ncclNetProperties_v6_t* props) {
  return Net::get().get_properties(dev, props);
}

ncclResult_t bnListen(int dev, void* handle, void** listenComm) {
  return Net::get().listen(dev, handle, listenComm);
}

ncclResult_t bnConnect(int dev, void* handle, void** sendComm) {
  return Net::get().connect(dev, handle, sendComm);
}

ncclResult_t bnAccept(void* listenComm, void** recvComm) {
  return Net::get().accept(listenComm, recvComm);
}

ncclResult_t bnRegMr(void* comm, void* data, int size, int type,
                     void** mhandle) {
  (void)comm, (void)data, (void)size;
  if (type != NCCL_PTR_HOST &&
      !(type == NCCL_PTR_CUDA && baguanet::staging_available()))
    return ncclInternalError;
  // The registration IS the pointer type: isend/irecv use it to pick the
  // staging path.  No pinning needed: host buffers are used in place and
  // GPU buffers are staged through the comm's pinned ring.
  *mhandle = (void*)(uintptr_t)type;
  return ncclSuccess;
}

ncclResult_t bnRegMrDmaBuf(void* comm, void* data, size_t size, int type,
                           uint64_t offset, int fd, void** mhandle) {
  (void)offset, (void)fd;
  return bnRegMr(comm, data, (int)size, type, mhandle);
}

ncclResult_t bnDeregMr(void* comm, void* mhandle) {
  (void)comm, (void)mhandle;
  return ncclSuccess;
}

ncclResult_t bnIsend(void* sendComm, void* data, int size, int tag,
                     void* mhandle, void** request) {
  return Net::get().isend(sendComm, data, size, tag, mhandle, request);
}

ncclResult_t bnIrecv(void* recvComm, int n, void** data, int* sizes,
                     int* tags, void** mhandles, void** request) {
  return Net::get().irecv(recvComm, n, data, sizes, tags, mhandles, request);
}

ncclResult_t bnIflush(void* recvComm, int n, void** data, int* sizes,
                      void** mhandles, void** request) {
  return Net::get().iflush(recvComm, n, data, sizes, mhandles, request);
}

ncclResult_t bnTest(void* request, int* done, int* sizes) {
  return Net::get().test(request, done, sizes);
}

ncclResult_t bnCloseSend(void* sendComm) {
  return Net::get().close_send(sendComm);
}

ncclResult_t bnCloseRecv(void* recvComm) {
  return Net::get().close_recv(recvComm);
}

ncclResult_t bnCloseListen(void* listenComm) {
  return Net::get().close_listen(listenComm);
}

// ---- v7 / v8 adapters ------------------------------------------------------
// RCCL probes ncclNetPlugin_v10..v6 newest-first; exporting v8 (+v7) lets
// newer RCCLs use a native vtable instead of their internal v6 wrapper
// (reference shipped floor+current the same way: cc/v3 + cc/v4).  The only
// deltas from v6 are the properties tail, the device-handle out-params on
// connect/accept (NULL: all networking is host-proxy), v8's size_t regMr,
// and the device-offload hooks (unreachable at NCCL_NET_DEVICE_HOST).

ncclResult_t bnGetProperties_v7(int dev, ncclNetProperties_v7_t* props) {
  ncclNetProperties_v6_t p6;
  ncclResult_t rc = Net::get().get_properties(dev, &p6);
  if (rc != ncclSuccess) return rc;
  props->name = p6.name;
  props->pciPath = p6.pciPath;
  props->guid = p6.guid;
  props->ptrSupport = p6.ptrSupport;
  props->speed = p6.speed;
  props->port = p6.port;
  props->latency = p6.latency;
  props->maxComms = p6.maxComms;
  props->maxRecvs = p6.maxRecvs;
  props->netDeviceType = NCCL_NET_DEVICE_HOST;
  props->netDeviceVersion = NCCL_NET_DEVICE_INVALID_VERSION;
  return ncclSuccess;
}

ncclResult_t bnGetProperties_v8(int dev, ncclNetProperties_v8_t* props) {
  ncclNetProperties_v6_t p6;
  ncclResult_t rc = Net::get().get_properties(dev, &p6);
  if (rc != ncclSuccess) return rc;
  props->name = p6.name;
  props->pciPath = p6.pciPath;
  props->guid = p6.guid;
  props->ptrSupport = p6.ptrSupport;
  props->regIsGlobal = 0;
  props->speed = p6.speed;
  props->port = p6.port;
  props->latency = p6.latency;
  props->maxComms = p6.maxComms;
  props->maxRecvs = p6.maxRecvs;
  props->netDeviceType = NCCL_NET_DEVICE_HOST;
  props->netDeviceVersion = NCCL_NET_DEVICE_INVALID_VERSION;
  return ncclSuccess;
}

ncclResult_t bnConnect_v7(int dev, void* handle, void** sendComm,
                          ncclNetDeviceHandle_v7_t** sendDevComm) {
  if (sendDevComm) *sendDevComm = nullptr;  // host-proxy: no device comm
  return Net::get().connect(dev, handle, sendComm);
}

ncclResult_t bnAccept_v7(void* listenComm, void** recvComm,
                         ncclNetDeviceHandle_v7_t** recvDevComm) {
  if (recvDevComm) *recvDevComm = nullptr;
  return Net::get().accept(listenComm, recvComm);
}

ncclResult_t bnRegMr_v8(void* comm, void* data, size_t size, int type,
                        void** mhandle) {
  if (size > (size_t)INT32_MAX) {
    // registration is type-tagging only (no pinning), so size never
    // matters — but refuse obviously bogus inputs rather than truncate
    return bnRegMr(comm, data, INT32_MAX, type, mhandle);
  }
  return bnRegMr(comm, data, (int)size, type, mhandle);
}

ncclResult_t bnGetDeviceMr(void* comm, void* mhandle, void** dptr_mhandle) {
  (void)comm, (void)mhandle, (void)dptr_mhandle;
  return ncclInternalError;  // never called at NCCL_NET_DEVICE_HOST
}

ncclResult_t bnIrecvConsumed(void* recvComm, int n, void* request) {
  (void)recvComm, (void)n, (void)request;
  return ncclInternalError;  // never called at NCCL_NET_DEVICE_HOST
}

}  // namespace

// Compile-time ABI ceiling: the main .so exports v6+v7+v8; a companion
// v6-only .so (libnccl-net-bagua6.so, -DBNET_ABI_MAX=6) is shipped as an
// escape hatch selectable via NCCL_NET_PLUGIN=bagua6 in case a future RCCL
// disagrees about the newer frozen layouts.
#ifndef BNET_ABI_MAX
#define BNET_ABI_MAX 8
#endif

extern "C" {

// RCCL dlsyms this struct by name (cf. reference export of ncclNetPlugin_v4,
// cc/v4/nccl_net_v4.cc:210-226).
__attribute__((visibility("default"))) ncclNet_v6_t ncclNetPlugin_v6 = {
    "BaguaNetAMD",  bnInit,        bnDevices,   bnGetProperties,
    bnListen,       bnConnect,     bnAccept,    bnRegMr,
    bnRegMrDmaBuf,  bnDeregMr,     bnIsend,     bnIrecv,
    bnIflush,       bnTest,        bnCloseSend, bnCloseRecv,
    bnCloseListen,
};

#if BNET_ABI_MAX >= 7
__attribute__((visibility("default"))) ncclNet_v7_t ncclNetPlugin_v7 = {
    "BaguaNetAMD",  bnInit,          bnDevices,     bnGetProperties_v7,
    bnListen,       bnConnect_v7,    bnAccept_v7,   bnRegMr,
    bnRegMrDmaBuf,  bnDeregMr,       bnIsend,       bnIrecv,
    bnIflush,       bnTest,          bnCloseSend,   bnCloseRecv,
    bnCloseListen,  bnGetDeviceMr,   bnIrecvConsumed,
};
#endif

#if BNET_ABI_MAX >= 8
__attribute__((visibility("default"))) ncclNet_v8_t ncclNetPlugin_v8 = {
    "BaguaNetAMD",  bnInit,          bnDevices,     bnGetProperties_v8,
    bnListen,       bnConnect_v7,    bnAccept_v7,   bnRegMr_v8,
    bnRegMrDmaBuf,  bnDeregMr,       bnIsend,       bnIrecv,
    bnIflush,       bnTest,          bnCloseSend,   bnCloseRecv,
    bnCloseListen,  bnGetDeviceMr,   bnIrecvConsumed,
};
#endif

}  // extern "C"
