"""ctypes binding to the plugin's exported ``ncclNetPlugin_v6`` vtable, plus
env helpers to make RCCL (torch.distributed "nccl" backend on ROCm) load it.

The binding exercises the *real* ABI surface RCCL consumes — the same
struct-of-function-pointers RCCL dlsyms (cf. reference SURVEY §2.2 C2) — so
the CPU loopback tests cover exactly what RCCL will call.
"""

from __future__ import annotations

import ctypes as C
import os
from typing import Optional

from . import PLUGIN_DIR, PLUGIN_PATH

NCCL_PTR_HOST = 0x1
NCCL_PTR_CUDA = 0x2
NCCL_NET_HANDLE_MAXSIZE = 128

ncclSuccess = 0


class NetProperties(C.Structure):
    _fields_ = [
        ("name", C.c_char_p),
        ("pciPath", C.c_char_p),
        ("guid", C.c_uint64),
        ("ptrSupport", C.c_int),
        ("speed", C.c_int),
        ("port", C.c_int),
        ("latency", C.c_float),
        ("maxComms", C.c_int),
        ("maxRecvs", C.c_int),
    ]


class NetPropertiesV7(C.Structure):
    _fields_ = NetProperties._fields_ + [
        ("netDeviceType", C.c_int),
        ("netDeviceVersion", C.c_int),
    ]


class NetPropertiesV8(C.Structure):
    _fields_ = [
        ("name", C.c_char_p),
        ("pciPath", C.c_char_p),
        ("guid", C.c_uint64),
        ("ptrSupport", C.c_int),
        ("regIsGlobal", C.c_int),
        ("speed", C.c_int),
        ("port", C.c_int),
        ("latency", C.c_float),
        ("maxComms", C.c_int),
        ("maxRecvs", C.c_int),
        ("netDeviceType", C.c_int),
        ("netDeviceVersion", C.c_int),
    ]


_F = C.CFUNCTYPE


class NcclNetV6(C.Structure):
    _fields_ = [
        ("name", C.c_char_p),
        ("init", _F(C.c_int, C.c_void_p)),
        ("devices", _F(C.c_int, C.POINTER(C.c_int))),
        ("getProperties", _F(C.c_int, C.c_int, C.POINTER(NetProperties))),
        ("listen", _F(C.c_int, C.c_int, C.c_void_p, C.POINTER(C.c_void_p))),
        ("connect", _F(C.c_int, C.c_int, C.c_void_p, C.POINTER(C.c_void_p))),
        ("accept", _F(C.c_int, C.c_void_p, C.POINTER(C.c_void_p))),
        ("regMr", _F(C.c_int, C.c_void_p, C.c_void_p, C.c_int, C.c_int,
                     C.POINTER(C.c_void_p))),
        ("regMrDmaBuf", _F(C.c_int, C.c_void_p, C.c_void_p, C.c_size_t,
                           C.c_int, C.c_uint64, C.c_int,
                           C.POINTER(C.c_void_p))),
        ("deregMr", _F(C.c_int, C.c_void_p, C.c_void_p)),
        ("isend", _F(C.c_int, C.c_void_p, C.c_void_p, C.c_int, C.c_int,
                     C.c_void_p, C.POINTER(C.c_void_p))),
        ("irecv", _F(C.c_int, C.c_void_p, C.c_int, C.POINTER(C.c_void_p),
                     C.POINTER(C.c_int), C.POINTER(C.c_int),
                     C.POINTER(C.c_void_p), C.POINTER(C.c_void_p))),
        ("iflush", _F(C.c_int, C.c_void_p, C.c_int, C.POINTER(C.c_void_p),
                      C.POINTER(C.c_int), C.POINTER(C.c_void_p),
                      C.POINTER(C.c_void_p))),
        ("test", _F(C.c_int, C.c_void_p, C.POINTER(C.c_int),
                    C.POINTER(C.c_int))),
        ("closeSend", _F(C.c_int, C.c_void_p)),
        ("closeRecv", _F(C.c_int, C.c_void_p)),
        ("closeListen", _F(C.c_int, C.c_void_p)),
    ]


def _vtable_struct(abi: int, props_struct) -> type:
    """Build the ncclNet_v7/v8 ctypes struct (v7+: device-handle out-params
    on connect/accept, trailing getDeviceMr/irecvConsumed; v8: size_t
    regMr)."""
    return type(
        f"NcclNetV{abi}",
        (C.Structure,),
        {"_fields_": [
            ("name", C.c_char_p),
            ("init", _F(C.c_int, C.c_void_p)),
            ("devices", _F(C.c_int, C.POINTER(C.c_int))),
            ("getProperties", _F(C.c_int, C.c_int, C.POINTER(props_struct))),
            ("listen", _F(C.c_int, C.c_int, C.c_void_p,
                          C.POINTER(C.c_void_p))),
            ("connect", _F(C.c_int, C.c_int, C.c_void_p,
                           C.POINTER(C.c_void_p), C.POINTER(C.c_void_p))),
            ("accept", _F(C.c_int, C.c_void_p, C.POINTER(C.c_void_p),
                          C.POINTER(C.c_void_p))),
            ("regMr", _F(C.c_int, C.c_void_p, C.c_void_p,
                         C.c_size_t if abi >= 8 else C.c_int, C.c_int,
                         C.POINTER(C.c_void_p))),
            ("regMrDmaBuf", _F(C.c_int, C.c_void_p, C.c_void_p, C.c_size_t,
                               C.c_int, C.c_uint64, C.c_int,
                               C.POINTER(C.c_void_p))),
            ("deregMr", _F(C.c_int, C.c_void_p, C.c_void_p)),
            ("isend", _F(C.c_int, C.c_void_p, C.c_void_p, C.c_int, C.c_int,
                         C.c_void_p, C.POINTER(C.c_void_p))),
            ("irecv", _F(C.c_int, C.c_void_p, C.c_int, C.POINTER(C.c_void_p),
                         C.POINTER(C.c_int), C.POINTER(C.c_int),
                         C.POINTER(C.c_void_p), C.POINTER(C.c_void_p))),
            ("iflush", _F(C.c_int, C.c_void_p, C.c_int, C.POINTER(C.c_void_p),
                          C.POINTER(C.c_int), C.POINTER(C.c_void_p),
                          C.POINTER(C.c_void_p))),
            ("test", _F(C.c_int, C.c_void_p, C.POINTER(C.c_int),
                        C.POINTER(C.c_int))),
            ("closeSend", _F(C.c_int, C.c_void_p)),
            ("closeRecv", _F(C.c_int, C.c_void_p)),
            ("closeListen", _F(C.c_int, C.c_void_p)),
            ("getDeviceMr", _F(C.c_int, C.c_void_p, C.c_void_p,
                               C.POINTER(C.c_void_p))),
            ("irecvConsumed", _F(C.c_int, C.c_void_p, C.c_int, C.c_void_p)),
        ]},
    )


NcclNetV7 = _vtable_struct(7, NetPropertiesV7)
NcclNetV8 = _vtable_struct(8, NetPropertiesV8)


class Plugin:
    """Thin pythonic wrapper over the vtable (raises on non-zero results).

    ``abi`` selects which exported vtable to drive (6, 7 or 8) — the same
    structs RCCL dlsyms, newest-first.
    """

    def __init__(self, path: Optional[str] = None, abi: int = 6):
        self.lib = C.CDLL(str(path or PLUGIN_PATH), mode=C.RTLD_GLOBAL)
        self.abi = abi
        struct = {6: NcclNetV6, 7: NcclNetV7, 8: NcclNetV8}[abi]
        self.vt = struct.in_dll(self.lib, f"ncclNetPlugin_v{abi}")
        self._props_struct = {6: NetProperties, 7: NetPropertiesV7,
                              8: NetPropertiesV8}[abi]
        self._check(self.vt.init(None), "init")

    @staticmethod
    def _check(rc: int, what: str) -> None:
        if rc != ncclSuccess:
            raise RuntimeError(f"{what} failed with ncclResult {rc}")

    @property
    def name(self) -> str:
        return self.vt.name.decode()

    def ndev(self) -> int:
        n = C.c_int(0)
        self._check(self.vt.devices(C.byref(n)), "devices")
        return n.value

    def properties(self, dev: int) -> dict:
        p = self._props_struct()
        self._check(self.vt.getProperties(dev, C.byref(p)), "getProperties")
        out = {
            "name": p.name.decode() if p.name else "",
            "pciPath": p.pciPath.decode() if p.pciPath else "",
            "guid": p.guid,
            "ptrSupport": p.ptrSupport,
            "speed": p.speed,
            "port": p.port,
            "latency": p.latency,
            "maxComms": p.maxComms,
            "maxRecvs": p.maxRecvs,
        }
        for extra in ("regIsGlobal", "netDeviceType", "netDeviceVersion"):
            if hasattr(p, extra):
                out[extra] = getattr(p, extra)
        return out

    def listen(self, dev: int):
        handle = (C.c_char * NCCL_NET_HANDLE_MAXSIZE)()
        lcomm = C.c_void_p(None)
        self._check(self.vt.listen(dev, handle, C.byref(lcomm)), "listen")
        return handle, lcomm

    def connect(self, dev: int, handle) -> Optional[C.c_void_p]:
        scomm = C.c_void_p(None)
        if self.abi >= 7:
            dh = C.c_void_p(None)  # device handle out-param: host plugin
            self._check(self.vt.connect(dev, handle, C.byref(scomm),
                                        C.byref(dh)), "connect")
            assert dh.value is None, "host plugin must not set a dev handle"
        else:
            self._check(self.vt.connect(dev, handle, C.byref(scomm)),
                        "connect")
        return scomm if scomm.value else None

    def accept(self, lcomm) -> Optional[C.c_void_p]:
        rcomm = C.c_void_p(None)
        if self.abi >= 7:
            dh = C.c_void_p(None)
            self._check(self.vt.accept(lcomm, C.byref(rcomm), C.byref(dh)),
                        "accept")
            assert dh.value is None, "host plugin must not set a dev handle"
        else:
            self._check(self.vt.accept(lcomm, C.byref(rcomm)), "accept")
        return rcomm if rcomm.value else None

    def reg_mr(self, comm, data, size: int, ptr_type: int = NCCL_PTR_HOST):
        mh = C.c_void_p(None)
        self._check(
            self.vt.regMr(comm, data, size, ptr_type, C.byref(mh)), "regMr"
        )
        return mh

    def isend(self, scomm, data, size: int, mhandle, tag: int = 0):
        req = C.c_void_p(None)
        self._check(
            self.vt.isend(scomm, data, size, tag, mhandle, C.byref(req)),
            "isend",
        )
        return req if req.value else None

    def irecv(self, rcomm, data, size: int, mhandle, tag: int = 0):
        datav = (C.c_void_p * 1)(C.cast(data, C.c_void_p))
        sizev = (C.c_int * 1)(size)
        tagv = (C.c_int * 1)(tag)
        mhv = (C.c_void_p * 1)(mhandle.value if mhandle else None)
        req = C.c_void_p(None)
        self._check(
            self.vt.irecv(rcomm, 1, datav, sizev, tagv, mhv, C.byref(req)),
            "irecv",
        )
        return req if req.value else None

    def irecv_n(self, rcomm, bufs, sizes, mhandle, tags=None):
        """Grouped receive: post len(bufs) buffers as ONE request
        (maxRecvs).  Returns (request, n) — test() fills n sizes."""
        n = len(bufs)
        datav = (C.c_void_p * n)(*[C.cast(b, C.c_void_p) for b in bufs])
        sizev = (C.c_int * n)(*sizes)
        tagv = (C.c_int * n)(*(tags or [0] * n))
        mhv = (C.c_void_p * n)(
            *[mhandle.value if mhandle else None] * n)
        req = C.c_void_p(None)
        self._check(
            self.vt.irecv(rcomm, n, datav, sizev, tagv, mhv, C.byref(req)),
            "irecv",
        )
        return (req if req.value else None), n

    def test_n(self, req, n):
        done = C.c_int(0)
        sizes = (C.c_int * n)(*([-1] * n))
        self._check(self.vt.test(req, C.byref(done), sizes), "test")
        return bool(done.value), list(sizes)

    def iflush(self, rcomm, data, size: int, mhandle):
        datav = (C.c_void_p * 1)(C.cast(data, C.c_void_p))
        sizev = (C.c_int * 1)(size)
        mhv = (C.c_void_p * 1)(mhandle.value if mhandle else None)
        req = C.c_void_p(None)
        self._check(
            self.vt.iflush(rcomm, 1, datav, sizev, mhv, C.byref(req)),
            "iflush",
        )
        return req if req.value else None

    def test(self, req):
        done = C.c_int(0)
        size = C.c_int(-1)
        self._check(self.vt.test(req, C.byref(done), C.byref(size)), "test")
        return bool(done.value), size.value

    def wait(self, req, timeout_s: float = 30.0):
        import time

        t0 = time.monotonic()
        while True:
            done, size = self.test(req)
            if done:
                return size
            if time.monotonic() - t0 > timeout_s:
                raise TimeoutError("request did not complete")

    def close_send(self, scomm):
        self._check(self.vt.closeSend(scomm), "closeSend")

    def close_recv(self, rcomm):
        self._check(self.vt.closeRecv(rcomm), "closeRecv")

    def close_listen(self, lcomm):
        self._check(self.vt.closeListen(lcomm), "closeListen")


_preloaded = {}


def preload(name: str = "bagua") -> None:
    """dlopen the plugin with its SONAME registered, so RCCL's later
    dlopen("librccl-net-<name>.so") resolves to the already-loaded
    library.  ``name="bagua6"`` preloads the v6-only escape hatch.

    Needed because mutating LD_LIBRARY_PATH inside a running process does
    not affect dlopen search paths (glibc snapshots it at startup) — the
    torchrun-launched bench sets env in-process.
    """
    if name not in _preloaded:
        _preloaded[name] = C.CDLL(str(PLUGIN_DIR / f"librccl-net-{name}.so"),
                                  mode=C.RTLD_GLOBAL)


def rccl_env(
    env: Optional[dict] = None,
    ifname: Optional[str] = None,
    force_net: bool = False,
) -> dict:
    """Env vars that make RCCL load this plugin.

    ``force_net=True`` additionally disables RCCL's P2P/SHM transports so
    even intra-node traffic is routed through the plugin (for A/B
    benchmarking of the transport itself; normal runs keep xGMI intra-node).
    """
    e = dict(env if env is not None else os.environ)
    # RCCL resolves the plugin as librccl-net-<name>.so (observed on the
    # MI355X box: "NET/Plugin: Could not find: librccl-net-bagua.so");
    # the build dir ships librccl-net-bagua.so + libnccl-net-bagua.so.
    e["NCCL_NET_PLUGIN"] = "bagua"
    e["LD_LIBRARY_PATH"] = (
        f"{PLUGIN_DIR}:{e.get('LD_LIBRARY_PATH', '')}".rstrip(":")
    )
    if ifname:
        e["NCCL_SOCKET_IFNAME"] = ifname
    if force_net:
        e["NCCL_P2P_DISABLE"] = "1"
        e["NCCL_SHM_DISABLE"] = "1"
    return e
