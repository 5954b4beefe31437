#!/usr/bin/env python3
"""Hot-kernel microbench for rocprofv3 runs (multi_pack / multi_unpack /
fused SGD / copy kernel), sized like the VGG16 bucket workload bench.py
drives.  Run under `rocprofv3 --stats` for per-kernel time and under
`--pmc FETCH_SIZE WRITE_SIZE` to validate HBM traffic; achieved GB/s =
bytes_moved / kernel_time vs the ~8 TB/s HBM3E peak.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from baguanet import ops  # noqa: E402


def timed(name, fn, bytes_moved, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"{name}: {bytes_moved / dt / 1e9:.1f} GB/s moved "
          f"({dt * 1e6:.1f} us/iter)")


def main():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    # VGG16-ish bucket: ~50 MB across mixed-size tensors
    sizes = [25088 * 512, 512 * 512 * 9, 4096 * 4096 // 4, 1000 * 40,
             64 * 3 * 9, 9 * 10 ** 6]
    ts = [torch.randn(n, device=dev) for n in sizes]
    numel = sum(sizes)
    flat = torch.empty(numel, device=dev)
    nbytes = numel * 4

    timed("multi_pack", lambda: ops.multi_pack(flat, ts), 2 * nbytes)
    timed("multi_unpack", lambda: ops.multi_unpack(flat, ts), 2 * nbytes)

    grads = [torch.randn_like(t) for t in ts]
    moms = [torch.zeros_like(t) for t in ts]
    timed("fused_sgd",
          lambda: ops.fused_sgd(ts, grads, moms, 0.01, 0.9, 1e-4, False),
          5 * nbytes)  # p,g,m reads + p,m writes

    src = torch.randn(16 << 20, device=dev)  # 64 MiB
    dst = torch.empty_like(src)
    timed("copy_bytes_d2d",
          lambda: ops.copy_bytes(dst, src), 2 * src.nbytes)


if __name__ == "__main__":
    main()
