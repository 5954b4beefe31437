#!/usr/bin/env python3
"""NCCL_PTR_DMABUF decision probe (GPU box).

Question: should the plugin import dmabuf fds and let
socket writers read GPU memory directly from a host mapping, skipping the
D2H staging copy?  Answer requires one number: CPU read bandwidth from a
CPU mapping of device memory (what sendmsg would achieve from a dmabuf
import) vs the SDMA copy-engine D2H bandwidth the staging path uses.

Exports the device allocation as a dmabuf via
hipMemGetHandleForAddressRange (ROCm's dmabuf export), mmaps it, and
times CPU reads; times hipMemcpyAsync D2H into pinned memory for the
same buffer.  Writes JSON to gpurun_out/dmabuf_probe.json.
"""

import ctypes as C
import json
import mmap
import os
import time

OUT = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "gpurun_out", "dmabuf_probe.json")

hip = C.CDLL("libamdhip64.so")

hipMemRangeHandleTypeDmaBufFd = 0x1
SIZE = 64 << 20  # 64 MiB


def check(rc, what):
    if rc != 0:
        raise RuntimeError(f"{what} failed: hipError {rc}")


def main():
    res = {"size_bytes": SIZE}
    check(hip.hipInit(0), "hipInit")
    dptr = C.c_void_p()
    check(hip.hipMalloc(C.byref(dptr), C.c_size_t(SIZE)), "hipMalloc")
    check(hip.hipMemset(dptr, 0x5A, C.c_size_t(SIZE)), "hipMemset")
    check(hip.hipDeviceSynchronize(), "sync")

    # --- dmabuf export + CPU mmap read ------------------------------------
    fd = C.c_int(-1)
    rc = hip.hipMemGetHandleForAddressRange(
        C.byref(fd), dptr, C.c_size_t(SIZE),
        C.c_int(hipMemRangeHandleTypeDmaBufFd), C.c_ulonglong(0))
    if rc != 0 or fd.value < 0:
        res["dmabuf_export"] = f"unavailable (hipError {rc})"
    else:
        res["dmabuf_export"] = "ok"
        try:
            m = mmap.mmap(fd.value, SIZE, prot=mmap.PROT_READ)
            # warm one page, then time a full sequential read
            _ = m[0]
            sink = bytearray(1 << 20)
            t0 = time.perf_counter()
            view = memoryview(m)
            total = 0
            for off in range(0, SIZE, 1 << 20):
                sink[:] = view[off:off + (1 << 20)]
                total += 1 << 20
            dt = time.perf_counter() - t0
            res["dmabuf_cpu_read_GBps"] = round(total / dt / 1e9, 3)
            view.release()
            m.close()
        except (OSError, ValueError) as e:
            res["dmabuf_mmap"] = f"failed: {e}"
        os.close(fd.value)

    # --- SDMA D2H into pinned (the staging path) ---------------------------
    hptr = C.c_void_p()
    check(hip.hipHostMalloc(C.byref(hptr), C.c_size_t(SIZE), 0), "hostMalloc")
    stream = C.c_void_p()
    check(hip.hipStreamCreate(C.byref(stream)), "streamCreate")
    # warmup
    check(hip.hipMemcpyAsync(hptr, dptr, C.c_size_t(SIZE), 2, stream),
          "memcpyAsync")  # 2 = hipMemcpyDeviceToHost
    check(hip.hipStreamSynchronize(stream), "streamSync")
    iters = 10
    t0 = time.perf_counter()
    for _ in range(iters):
        check(hip.hipMemcpyAsync(hptr, dptr, C.c_size_t(SIZE), 2, stream),
              "memcpyAsync")
    check(hip.hipStreamSynchronize(stream), "streamSync")
    dt = time.perf_counter() - t0
    res["sdma_d2h_GBps"] = round(iters * SIZE / dt / 1e9, 3)

    if "dmabuf_cpu_read_GBps" in res and res["dmabuf_cpu_read_GBps"] > 0:
        res["sdma_advantage_x"] = round(
            res["sdma_d2h_GBps"] / res["dmabuf_cpu_read_GBps"], 1)

    os.makedirs(os.path.dirname(OUT), exist_ok=True)
    with open(OUT, "w") as f:
        json.dump(res, f, indent=1)
    print(json.dumps(res))


if __name__ == "__main__":
    main()
