#!/usr/bin/env python3
"""Staging-path microbenchmarks on the GPU box (results -> profiles/):

1. D2H / H2D pinned copies: SDMA (hipMemcpyAsync via torch copy_) vs the
   hand-written gfx950 pack kernel (baguanet.ops.copy_bytes).
2. multi_pack fused bucket packing vs torch.cat.
3. Plugin GPU->GPU loopback TCP throughput (full staging pipeline:
   D2H pinned ring -> striped sockets -> pinned -> H2D).
"""

from __future__ import annotations

import argparse
import ctypes as C
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def time_gpu(fn, iters=20, warmup=5):
    import torch

    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_copies():
    import torch

    from baguanet import ops

    out = []
    for mb in (1, 4, 16, 64, 256):
        n = mb * 1024 * 1024 // 4
        gpu = torch.randn(n, device="cuda")
        gpu2 = torch.empty_like(gpu)
        pinned = torch.empty(n, pin_memory=True)
        r = {"MB": mb}
        dt = time_gpu(lambda: pinned.copy_(gpu, non_blocking=True))
        r["d2h_sdma_GBps"] = round(n * 4 / dt / 1e9, 2)
        dt = time_gpu(lambda: ops.copy_bytes(pinned, gpu))
        r["d2h_kernel_GBps"] = round(n * 4 / dt / 1e9, 2)
        dt = time_gpu(lambda: gpu.copy_(pinned, non_blocking=True))
        r["h2d_sdma_GBps"] = round(n * 4 / dt / 1e9, 2)
        dt = time_gpu(lambda: ops.copy_bytes(gpu, pinned))
        r["h2d_kernel_GBps"] = round(n * 4 / dt / 1e9, 2)
        dt = time_gpu(lambda: gpu2.copy_(gpu, non_blocking=True))
        r["d2d_sdma_GBps"] = round(2 * n * 4 / dt / 1e9, 2)
        dt = time_gpu(lambda: ops.copy_bytes(gpu2, gpu))
        r["d2d_kernel_GBps"] = round(2 * n * 4 / dt / 1e9, 2)
        out.append(r)
    return out


def bench_multipack():
    import torch

    from baguanet import ops

    out = []
    for ntensors, numel in ((32, 1 << 18), (128, 1 << 16), (64, 1 << 20)):
        ts = [torch.randn(numel, device="cuda") for _ in range(ntensors)]
        flat = torch.empty(ntensors * numel, device="cuda")
        dt_k = time_gpu(lambda: ops.multi_pack(flat, ts))
        dt_cat = time_gpu(lambda: torch.cat(ts, out=flat))
        bytes_moved = 2 * ntensors * numel * 4
        out.append(
            {
                "ntensors": ntensors,
                "numel": numel,
                "multi_pack_GBps": round(bytes_moved / dt_k / 1e9, 2),
                "torch_cat_GBps": round(bytes_moved / dt_cat / 1e9, 2),
            }
        )
    return out


def bench_plugin_gpu_loopback():
    """GPU->GPU over TCP loopback through the plugin staging path."""
    import torch

    sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "tests"))
    from baguanet.plugin import Plugin
    from test_plugin_loopback import establish

    os.environ.setdefault("NCCL_SOCKET_IFNAME", "lo")
    p = Plugin()
    if not (p.properties(0)["ptrSupport"] & 0x2):
        return []
    lcomm, scomm, rcomm = establish(p)
    out = []
    DEPTH = 8
    for mb in (1, 4, 16):
        size = mb * 1024 * 1024
        n_msgs = max(8, (1 << 30) // size // 4)
        src = torch.randn(size // 4, device="cuda")
        dsts = [torch.empty_like(src) for _ in range(DEPTH)]
        torch.cuda.synchronize()  # buffers ready before posting (ABI)
        smh = p.reg_mr(scomm, C.c_void_p(src.data_ptr()), size, 0x2)
        rmh = p.reg_mr(rcomm, C.c_void_p(dsts[0].data_ptr()), size, 0x2)
        t0 = time.perf_counter()
        sdone = rdone = sposted = rposted = 0
        sreqs, rreqs = [], []
        while rdone < n_msgs:
            while rposted < n_msgs and len(rreqs) < DEPTH:
                r = p.irecv(
                    rcomm,
                    C.c_void_p(dsts[rposted % DEPTH].data_ptr()),
                    size,
                    rmh,
                )
                if r is None:
                    break
                rreqs.append(r)
                rposted += 1
            while sposted < n_msgs and len(sreqs) < DEPTH:
                r = p.isend(scomm, C.c_void_p(src.data_ptr()), size, smh)
                if r is None:
                    break
                sreqs.append(r)
                sposted += 1
            if sreqs and p.test(sreqs[0])[0]:
                sreqs.pop(0)
                sdone += 1
            if rreqs and p.test(rreqs[0])[0]:
                rreqs.pop(0)
                rdone += 1
        dt = time.perf_counter() - t0
        out.append(
            {
                "MB": mb,
                "msgs": n_msgs,
                "gpu_loopback_GBps": round(n_msgs * size / dt / 1e9, 3),
            }
        )
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default=None)
    args = ap.parse_args()
    res = {
        "copies": bench_copies(),
        "multipack": bench_multipack(),
        "plugin_gpu_loopback": bench_plugin_gpu_loopback(),
    }
    s = json.dumps(res, indent=1)
    print(s)
    if args.out:
        with open(args.out, "w") as f:
            f.write(s)


if __name__ == "__main__":
    main()
