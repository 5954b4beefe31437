#!/usr/bin/env python3
"""allreduce_perf — rccl-tests `all_reduce_perf -b 8 -e 128M -f 2` equivalent
on torch.distributed (RCCL on GPU, gloo on CPU), per BASELINE configs 2-3.

busbw = algbw * 2*(n-1)/n  (ring all-reduce bytes-on-wire factor, matching
nccl-tests' definition).

Launch (driver-style):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 benchmarks/allreduce_perf.py [--force-net]

Modes:
    default      — RCCL picks transports (xGMI intra-node; plugin inter-node)
    --force-net  — disable P2P/SHM so ALL traffic runs through the net
                   plugin (A/B of the transport itself on one node)
    --no-plugin  — don't register the plugin (stock RCCL transports)
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--min-bytes", type=int, default=8)
    ap.add_argument("--max-bytes", type=int, default=128 * 1024 * 1024)
    ap.add_argument("--factor", type=int, default=2)
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--force-net", action="store_true")
    ap.add_argument("--no-plugin", action="store_true")
    ap.add_argument("--backend", default=None)
    ap.add_argument("--out", default=None, help="write JSON results here")
    args = ap.parse_args()

    if not args.no_plugin:
        from baguanet.plugin import preload, rccl_env

        try:
            preload()
        except OSError:
            pass

        for k, v in rccl_env(env={}, force_net=args.force_net).items():
            if k == "LD_LIBRARY_PATH":
                os.environ[k] = f"{v}:{os.environ.get(k, '')}".rstrip(":")
            else:
                os.environ.setdefault(k, v)
        if args.force_net:
            # pin OUR net by name: if the plugin failed to load, RCCL must
            # error out rather than silently falling back to its internal
            # socket transport (which would fake a 1.0 plugin/stock ratio)
            os.environ.setdefault("NCCL_NET", "BaguaNetAMD")
    elif args.force_net:
        os.environ["NCCL_P2P_DISABLE"] = "1"
        os.environ["NCCL_SHM_DISABLE"] = "1"
        # pin the internal TCP socket transport: "stock RCCL TCP" is the
        # comparison the reference published (README.md:48-50), and on an
        # RDMA-equipped box RCCL would otherwise pick IB verbs
        os.environ.setdefault("NCCL_NET", "Socket")

    import torch
    import torch.distributed as dist

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available()
    backend = args.backend or ("nccl" if use_cuda else "gloo")
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    dist.init_process_group(backend, rank=rank, world_size=world)

    sizes = []
    b = args.min_bytes
    while b <= args.max_bytes:
        sizes.append(b)
        b *= args.factor

    results = []
    for size in sizes:
        n = max(size // 4, 1)  # float32 elements
        t = torch.ones(n, device=device)
        for _ in range(args.warmup):
            dist.all_reduce(t)
        if use_cuda:
            torch.cuda.synchronize()
        dist.barrier()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            dist.all_reduce(t)
        if use_cuda:
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        # max over ranks
        tt = torch.tensor([dt], device=device if use_cuda else "cpu")
        dist.all_reduce(tt, op=dist.ReduceOp.MAX)
        dt = float(tt.item())
        algbw = n * 4 / dt / 1e9
        busbw = algbw * (2 * (world - 1) / world if world > 1 else 1.0)
        results.append(
            {
                "bytes": n * 4,
                "us": round(dt * 1e6, 1),
                "algbw_GBps": round(algbw, 3),
                "busbw_GBps": round(busbw, 3),
            }
        )
        if rank == 0:
            r = results[-1]
            print(
                f"{r['bytes']:>12} B  {r['us']:>10.1f} us  "
                f"alg {r['algbw_GBps']:>8.3f}  bus {r['busbw_GBps']:>8.3f} GB/s",
                flush=True,
            )

    if rank == 0:
        summary = {
            "bench": "all_reduce_perf",
            "world": world,
            "backend": backend,
            "device": str(device.type),
            "force_net": args.force_net,
            "plugin": not args.no_plugin,
            "results": results,
        }
        if args.out:
            with open(args.out, "w") as f:
                json.dump(summary, f, indent=1)
        print(json.dumps({k: v for k, v in summary.items() if k != "results"}))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
