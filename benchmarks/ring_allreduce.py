#!/usr/bin/env python3
"""Ring all-reduce implemented DIRECTLY on the plugin's ncclNet vtable —
P processes, each with a send comm to the next rank and a recv comm from
the previous one, running the textbook reduce-scatter + all-gather ring
(exactly the byte pattern RCCL's ring all_reduce drives through a net
transport).  CPU float32, numpy reduction, every result verified.

This is the closest all_reduce_perf stand-in producible without multiple
GPUs: the transport carries the full 2*(n-1)/n bytes-on-wire of a real
all-reduce, with the same send/recv dependency chain.  busbw uses the
nccl-tests definition: algbw * 2*(n-1)/n.

    python benchmarks/ring_allreduce.py --ranks 4 --sizes 4194304 67108864
"""

from __future__ import annotations

import argparse
import ctypes as C
import json
import multiprocessing as mp
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402


def _rank(rank, nranks, prev_conn, next_conn, args, out_q, ns_conn=None):
    """prev_conn talks to rank-1 (we receive from it), next_conn to
    rank+1 (we send to it)."""
    if ns_conn is not None:  # --shaped, rank 1 of 2: own netns behind TBF
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        import netns_rig as rig

        rig.unshare_newnet()
        ns_conn.send(os.getpid())
        assert ns_conn.recv() == "veth-moved"
        rig.child_setup(args.shaped, 10.0)
        os.environ["NCCL_SOCKET_IFNAME"] = rig.CHILD_IF
    elif args.shaped is not None:  # rank 0 stays in the parent ns
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        import netns_rig as rig

        os.environ["NCCL_SOCKET_IFNAME"] = rig.PARENT_IF
    os.environ.setdefault("NCCL_SOCKET_IFNAME", "lo")
    os.environ.setdefault("BNET_MIN_CHUNKSIZE", "131072")
    from baguanet.plugin import Plugin

    p = Plugin()
    # I listen; my handle goes to the PREVIOUS rank, which connects to me.
    handle, lcomm = p.listen(0)
    prev_conn.send(bytes(handle))
    peer_handle_bytes = next_conn.recv()
    peer_handle = (C.c_char * len(peer_handle_bytes)).from_buffer_copy(
        peer_handle_bytes)
    scomm = rcomm = None
    while scomm is None or rcomm is None:
        if scomm is None:
            scomm = p.connect(0, peer_handle)
        if rcomm is None:
            rcomm = p.accept(lcomm)
    mh = p.reg_mr(scomm, None, 0)

    results = []
    for size in args.sizes:
        n = max(size // 4, nranks)  # fp32 elements, >= one per chunk
        n -= n % nranks             # chunk-aligned for simplicity
        chunk = n // nranks
        data = (np.arange(n, dtype=np.float32) % 97) + rank
        expect = (np.arange(n, dtype=np.float32) % 97) * nranks \
            + sum(range(nranks))
        buf = data.copy()
        tmp = np.empty(chunk, dtype=np.float32)

        def ptr(a, off_elems=0):
            return C.c_void_p(a.ctypes.data + off_elems * 4)

        # Slice pipelining within each ring step (the same trick as
        # NCCL's slices): the hop is cut into up to 8 slices so slice
        # k's reduction overlaps slice k+1's transfer.
        slice_elems = max(chunk // 8, 65536 // 4)
        nslices = (chunk + slice_elems - 1) // slice_elems

        def slices_of(ci):
            out = []
            for si in range(nslices):
                lo = ci * chunk + si * slice_elems
                n_el = min(slice_elems, (ci + 1) * chunk - lo)
                out.append((lo, n_el))
            return out

        def step_xfer(s_ci, r_ci, tagbase, reduce_into):
            """One ring step: send chunk s_ci, receive chunk r_ci
            (into tmp and += if reduce_into, else straight into buf)."""
            sl_s = slices_of(s_ci)
            sl_r = slices_of(r_ci)
            rreqs = []
            for si, (lo, n_el) in enumerate(sl_r):
                dst_off = si * slice_elems if reduce_into else lo
                a = tmp if reduce_into else buf
                req = None
                while req is None:
                    req = p.irecv(rcomm, ptr(a, dst_off), n_el * 4, mh,
                                  tag=tagbase + si)
                    if req is None:
                        os.sched_yield()
                rreqs.append((req, si, n_el))
            sreqs = []
            for si, (lo, n_el) in enumerate(sl_s):
                req = None
                while req is None:
                    req = p.isend(scomm, ptr(buf, lo), n_el * 4, mh,
                                  tag=tagbase + si)
                    if req is None:
                        os.sched_yield()
                sreqs.append(req)
            # recvs complete FIFO: reduce each slice as it lands, while
            # later slices are still on the wire
            for req, si, n_el in rreqs:
                while not p.test(req)[0]:
                    for sr in sreqs:
                        p.test(sr)
                    os.sched_yield()
                if reduce_into:
                    lo = r_ci * chunk + si * slice_elems
                    buf[lo:lo + n_el] += tmp[si * slice_elems:
                                             si * slice_elems + n_el]
            for sr in sreqs:
                while not p.test(sr)[0]:
                    os.sched_yield()

        iters = args.iters if size >= (1 << 20) else args.iters * 4
        t_total = 0.0
        for it in range(args.warmup + iters):
            buf[:] = data
            t0 = time.perf_counter()
            # reduce-scatter: after P-1 steps, rank r owns the full sum of
            # chunk (r+1) % P
            for step in range(nranks - 1):
                s_ci = (rank - step) % nranks
                r_ci = (rank - step - 1) % nranks
                step_xfer(s_ci, r_ci, tagbase=step * 16, reduce_into=True)
            # all-gather: circulate the reduced chunks
            for step in range(nranks - 1):
                s_ci = (rank - step + 1) % nranks
                r_ci = (rank - step) % nranks
                step_xfer(s_ci, r_ci, tagbase=1024 + step * 16,
                          reduce_into=False)
            dt = time.perf_counter() - t0
            if it >= args.warmup:
                t_total += dt
            if it == args.warmup:  # verify once per size, after warmup
                assert np.array_equal(buf, expect), \
                    f"rank {rank}: all-reduce result mismatch at {size} B"
        avg = t_total / iters
        nbytes = n * 4
        algbw = nbytes / avg / 1e9
        busbw = algbw * 2 * (nranks - 1) / nranks
        results.append({"bytes": nbytes, "us": round(avg * 1e6, 1),
                        "algbw_GBps": round(algbw, 3),
                        "busbw_GBps": round(busbw, 3)})

    # NCCL-lifecycle fence: a send comm may only close after the DOWNSTREAM
    # rank completed all its receives (isend completion means bytes are in
    # the kernel, not delivered; closing early would FIN idle stripe
    # sockets while the peer's recv is still pending — which the transport
    # rightly reports as a premature-EOF comm error).  Ranks finish at
    # different times, so confirm over the control pipes: tell the
    # upstream rank our recvs are done, wait for the downstream's word.
    prev_conn.send("alldone")
    assert next_conn.recv() == "alldone"
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    out_q.put((rank, results))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--ranks", type=int, default=4)
    ap.add_argument("--sizes", type=int, nargs="*",
                    default=[65536, 1 << 20, 16 << 20, 128 << 20])
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--json", action="store_true")
    ap.add_argument("--shaped", type=float, default=None,
                    help="2 ranks only: rank 1 runs in its own netns "
                         "behind a veth pair TBF-shaped to this many "
                         "Gbit/s — a two-node-over-limited-TCP analogue "
                         "(BASELINE config 2).  Needs CAP_NET_ADMIN")
    args = ap.parse_args()
    if args.shaped is not None and args.ranks != 2:
        ap.error("--shaped supports exactly 2 ranks")

    ctx = mp.get_context("spawn")
    # pipe ring: conn[r] connects rank r (as next_conn) with r+1 (as prev)
    pipes = [ctx.Pipe() for _ in range(args.ranks)]
    q = ctx.Queue()
    ns_parent = ns_child = None
    if args.shaped is not None:
        ns_parent, ns_child = ctx.Pipe()
    procs = []
    for r in range(args.ranks):
        prev_conn = pipes[(r - 1) % args.ranks][1]  # to rank r-1
        next_conn = pipes[r][0]                     # to rank r+1
        procs.append(ctx.Process(
            target=_rank, args=(r, args.ranks, prev_conn, next_conn,
                                args, q,
                                ns_child if r == 1 else None)))
    procs[1 if args.shaped is not None else 0].start()
    if args.shaped is not None:
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        import netns_rig as rig

        child_pid = ns_parent.recv()
        rig.parent_setup(child_pid)
        if args.shaped > 0:
            rig.parent_shape(args.shaped, 10.0)
        ns_parent.send("veth-moved")
        procs[0].start()
    else:
        for pr in procs[1:]:
            pr.start()
    outs = {}
    for _ in range(args.ranks):
        r, res = q.get(timeout=900)
        outs[r] = res
    for pr in procs:
        pr.join(30)
    if args.shaped is not None:
        import netns_rig as rig

        rig.parent_teardown()
    # max-over-ranks per size (nccl-tests reports the slowest rank)
    merged = []
    for i, size_res in enumerate(outs[0]):
        worst = max((outs[r][i] for r in outs), key=lambda x: x["us"])
        merged.append(worst)
    header = {"bench": "plugin ring all_reduce", "ranks": args.ranks,
              "verified": True}
    if args.json:
        print(json.dumps({**header, "results": merged}))
    else:
        print(header)
        for r in merged:
            print(f"  {r['bytes']:>11} B  {r['us']:>10.1f} us  "
                  f"alg {r['algbw_GBps']:>7.3f}  bus {r['busbw_GBps']:>7.3f}"
                  f" GB/s")


if __name__ == "__main__":
    main()
