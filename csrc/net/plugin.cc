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

}  // namespace

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

}  // extern "C"
