#!/usr/bin/env python3
"""Raw plugin point-to-point throughput: two processes, one-way stream of
pipelined messages through the ncclNetPlugin_v6 vtable over TCP.

This isolates the transport itself — the reference's +50% claim is about
exactly this layer (multi-stream TCP vs single-stream).  Compare:

    python benchmarks/p2p_perf.py --nstreams 1   # single-stream baseline
    python benchmarks/p2p_perf.py --nstreams 4   # striped (default)

Over loopback the kernel memcpy is the bottleneck rather than a NIC, so
single-stream saturates one core (~5-8 GB/s) while striping scales with
streams — same mechanics that win on a real 100GbE NIC.
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

DEPTH = 16  # outstanding messages (NCCL proxy pipelines similarly)


def _enter_child_ns(ns_conn, args):
    """Receiver side of --shaped: unshare a netns, let the parent move the
    veth peer in, configure + shape it, and use it as the NIC."""
    import netns_rig as rig

    rig.unshare_newnet()
    ns_conn.send(os.getpid())
    assert ns_conn.recv() == "veth-moved"
    rig.child_setup(args.shaped, args.delay_us)
    args.ifname = rig.CHILD_IF


def _set_env(args, role=None):
    # per-role engine override for A/B isolation (BNET_IMPL_SENDER/RECEIVER)
    if role:
        impl = os.environ.get(f"BNET_IMPL_{role.upper()}")
        if impl:
            os.environ["BNET_IMPLEMENT"] = impl
    os.environ["NCCL_SOCKET_IFNAME"] = args.ifname
    os.environ["BNET_NSTREAMS"] = str(args.nstreams)
    os.environ["BNET_MIN_CHUNKSIZE"] = str(args.min_chunk)
    os.environ["BNET_IO_THREADS"] = str(args.io_threads)


def _pattern(i: int, size: int):
    """Deterministic per-message payload (numpy-vectorized for soak rates)."""
    import numpy as np

    return ((np.arange(size, dtype=np.uint32) * 31 + i * 131) & 0xFF).astype(
        np.uint8).tobytes()


def _receiver(conn, args, out_q, ns_conn=None):
    if ns_conn is not None:
        _enter_child_ns(ns_conn, args)
    _set_env(args, "receiver")
    from baguanet.plugin import Plugin

    p = Plugin()
    handle, lcomm = p.listen(0)
    conn.send(bytes(handle))
    rcomm = None
    while rcomm is None:
        rcomm = p.accept(lcomm)
    mh = p.reg_mr(rcomm, None, 0)
    verified = 0
    for size in args.sizes:
        n_msgs = max(4, min(args.max_msgs, args.bytes_per_size // size))
        bufs = [C.create_string_buffer(size) for _ in range(DEPTH)]
        conn.send(("ready", size))  # buffers allocated — sender may start
        done = 0
        posted = 0
        reqs = []  # (req, msg_idx)
        # completion can be out of order across the window, so a slot is
        # reusable only when ITS request finished — not when any DEPTH
        # requests have (verify mode checks content per slot)
        slot_free = [True] * DEPTH
        while done < n_msgs:
            while posted < n_msgs and len(reqs) < DEPTH and \
                    slot_free[posted % DEPTH]:
                r = p.irecv(rcomm, bufs[posted % DEPTH], size, mh)
                if r is None:
                    break
                slot_free[posted % DEPTH] = False
                reqs.append((r, posted))
                posted += 1
            for item in list(reqs):
                ok, _ = p.test(item[0])
                if ok:
                    if args.verify:
                        i = item[1]
                        want = _pattern(i, size)
                        got = bufs[i % DEPTH].raw[:size]
                        if got != want:
                            match = [j for j in range(n_msgs)
                                     if _pattern(j, size) == got]
                            raise AssertionError(
                                f"payload corrupted: size={size} msg={i} "
                                f"slot={i % DEPTH} content-matches-msgs="
                                f"{match[:5]} outstanding="
                                f"{[x[1] for x in reqs]}")
                        verified += 1
                    slot_free[item[1] % DEPTH] = True
                    reqs.remove(item)
                    done += 1
        conn.send(("size-done", size))
    conn.recv()
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    out_q.put(f"recv-ok verified={verified}" if args.verify else "recv-ok")


def _sender(conn, args, out_q):
    _set_env(args, "sender")
    from baguanet.plugin import Plugin

    p = Plugin()
    hb = conn.recv()
    handle = (C.c_char * len(hb)).from_buffer_copy(hb)
    scomm = None
    while scomm is None:
        scomm = p.connect(0, handle)
    mh = p.reg_mr(scomm, None, 0)
    results = []
    for size in args.sizes:
        n_msgs = max(4, min(args.max_msgs, args.bytes_per_size // size))
        if args.verify:
            bufs = [C.create_string_buffer(size) for _ in range(DEPTH)]
        else:
            buf = C.create_string_buffer(os.urandom(size), size)
        tag, s = conn.recv()
        assert tag == "ready" and s == size
        t0 = time.perf_counter()
        done = 0
        posted = 0
        reqs = []  # (req, msg_idx)
        slot_free = [True] * DEPTH  # see receiver: per-slot gating
        while done < n_msgs:
            while posted < n_msgs and len(reqs) < DEPTH:
                if args.verify:
                    if not slot_free[posted % DEPTH]:
                        break
                    sb = bufs[posted % DEPTH]
                    sb.raw = _pattern(posted, size)
                    r = p.isend(scomm, sb, size, mh)
                else:
                    r = p.isend(scomm, buf, size, mh)
                if r is None:
                    break
                if args.verify:
                    slot_free[posted % DEPTH] = False
                reqs.append((r, posted))
                posted += 1
            for item in list(reqs):
                ok, _ = p.test(item[0])
                if ok:
                    if args.verify:
                        slot_free[item[1] % DEPTH] = True
                    reqs.remove(item)
                    done += 1
        # wait for receiver to fully drain this size
        tag, s = conn.recv()
        assert tag == "size-done" and s == size
        dt = time.perf_counter() - t0
        gbps = n_msgs * size / dt / 1e9
        results.append(
            {"size": size, "msgs": n_msgs, "secs": round(dt, 4),
             "GBps": round(gbps, 3)}
        )
    conn.send("done")
    p.close_send(scomm)
    out_q.put(results)


def _receiver_pairs(conn, args, out_q, ns_conn=None):
    """--pairs N: N independent comms between the same two processes, all
    receiving concurrently (models N channels/peers sharing one NIC — the
    fairness regime the reference optimized for)."""
    if ns_conn is not None:
        _enter_child_ns(ns_conn, args)
    _set_env(args, "receiver")
    from baguanet.plugin import Plugin

    p = Plugin()
    handle, lcomm = p.listen(0)
    conn.send(bytes(handle))
    rcomms = []
    while len(rcomms) < args.pairs:
        rc = p.accept(lcomm)
        if rc is not None:
            rcomms.append(rc)
    mh = p.reg_mr(rcomms[0], None, 0)
    size = args.sizes[0]
    n_msgs = max(4, min(args.max_msgs, args.bytes_per_size // size))
    bufs = [[C.create_string_buffer(size) for _ in range(DEPTH)]
            for _ in rcomms]
    conn.send(("ready", size))
    done = [0] * args.pairs
    posted = [0] * args.pairs
    reqs = [[] for _ in rcomms]
    while any(d < n_msgs for d in done):
        for ci, rc in enumerate(rcomms):
            while posted[ci] < n_msgs and len(reqs[ci]) < DEPTH:
                r = p.irecv(rc, bufs[ci][posted[ci] % DEPTH], size, mh)
                if r is None:
                    break
                reqs[ci].append(r)
                posted[ci] += 1
            for r in list(reqs[ci]):
                ok, _ = p.test(r)
                if ok:
                    reqs[ci].remove(r)
                    done[ci] += 1
    conn.send(("size-done", size))
    conn.recv()
    for rc in rcomms:
        p.close_recv(rc)
    p.close_listen(lcomm)
    out_q.put("recv-ok")


def _sender_pairs(conn, args, out_q):
    _set_env(args, "sender")
    from baguanet.plugin import Plugin

    p = Plugin()
    hb = conn.recv()
    handle_t = C.c_char * len(hb)
    scomms = []
    handles = [handle_t.from_buffer_copy(hb) for _ in range(args.pairs)]
    pending = list(range(args.pairs))
    while pending:
        for i in list(pending):
            sc = p.connect(0, handles[i])
            if sc is not None:
                scomms.append(sc)
                pending.remove(i)
    mh = p.reg_mr(scomms[0], None, 0)
    size = args.sizes[0]
    n_msgs = max(4, min(args.max_msgs, args.bytes_per_size // size))
    buf = C.create_string_buffer(os.urandom(size), size)
    tag, s = conn.recv()
    assert tag == "ready" and s == size
    t0 = time.perf_counter()
    done = [0] * args.pairs
    posted = [0] * args.pairs
    reqs = [[] for _ in scomms]
    t_done = [None] * args.pairs
    while any(d < n_msgs for d in done):
        for ci, sc in enumerate(scomms):
            while posted[ci] < n_msgs and len(reqs[ci]) < DEPTH:
                r = p.isend(sc, buf, size, mh)
                if r is None:
                    break
                reqs[ci].append(r)
                posted[ci] += 1
            for r in list(reqs[ci]):
                ok, _ = p.test(r)
                if ok:
                    reqs[ci].remove(r)
                    done[ci] += 1
                    if done[ci] == n_msgs:
                        t_done[ci] = time.perf_counter() - t0
    tag, s = conn.recv()
    assert tag == "size-done" and s == size
    total_dt = time.perf_counter() - t0
    per_comm = [round(n_msgs * size / dt / 1e9, 3) for dt in t_done]
    results = [{
        "size": size, "msgs_per_comm": n_msgs, "pairs": args.pairs,
        "secs": round(total_dt, 4),
        "GBps_aggregate": round(args.pairs * n_msgs * size / total_dt / 1e9,
                                3),
        "GBps_per_comm": per_comm,
        # min/max completion-rate ratio: 1.0 = perfectly fair service
        "fairness": round(min(per_comm) / max(per_comm), 3),
    }]
    conn.send("done")
    for sc in scomms:
        p.close_send(sc)
    out_q.put(results)


def _duplex_worker(conn, args, out_q, is_a, ns_conn=None):
    """Each process sends AND receives simultaneously (ring-edge pattern)."""
    if ns_conn is not None:
        _enter_child_ns(ns_conn, args)
    _set_env(args)
    from baguanet.plugin import Plugin

    p = Plugin()
    handle, lcomm = p.listen(0)
    conn.send(bytes(handle))
    peer_hb = conn.recv()
    peer_handle = (C.c_char * len(peer_hb)).from_buffer_copy(peer_hb)
    scomm = rcomm = None
    while scomm is None or rcomm is None:
        if scomm is None:
            scomm = p.connect(0, peer_handle)
        if rcomm is None:
            rcomm = p.accept(lcomm)
    smh = p.reg_mr(scomm, None, 0)
    rmh = p.reg_mr(rcomm, None, 0)
    results = []
    for size in args.sizes:
        n_msgs = max(4, min(args.max_msgs, args.bytes_per_size // size))
        sbuf = C.create_string_buffer(os.urandom(size), size)
        rbufs = [C.create_string_buffer(size) for _ in range(DEPTH)]
        conn.send(("ready", size))
        assert conn.recv() == ("ready", size)
        t0 = time.perf_counter()
        sdone = rdone = sposted = rposted = 0
        sreqs, rreqs = [], []
        while sdone < n_msgs or rdone < n_msgs:
            while rposted < n_msgs and len(rreqs) < DEPTH:
                r = p.irecv(rcomm, rbufs[rposted % DEPTH], size, rmh)
                if r is None:
                    break
                rreqs.append(r)
                rposted += 1
            while sposted < n_msgs and len(sreqs) < DEPTH:
                r = p.isend(scomm, sbuf, size, smh)
                if r is None:
                    break
                sreqs.append(r)
                sposted += 1
            for r in list(sreqs):
                if p.test(r)[0]:
                    sreqs.remove(r)
                    sdone += 1
            for r in list(rreqs):
                if p.test(r)[0]:
                    rreqs.remove(r)
                    rdone += 1
        conn.send(("size-done", size))
        assert conn.recv() == ("size-done", size)
        dt = time.perf_counter() - t0
        results.append(
            {"size": size, "msgs": n_msgs, "secs": round(dt, 4),
             "GBps_each_way": round(n_msgs * size / dt / 1e9, 3),
             "GBps_aggregate": round(2 * n_msgs * size / dt / 1e9, 3)}
        )
    conn.send("done")
    conn.recv()
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    out_q.put(results if is_a else "b-ok")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nstreams", type=int, default=4)
    ap.add_argument("--io-threads", type=int, default=4)
    ap.add_argument("--min-chunk", type=int, default=131072)
    ap.add_argument("--ifname", default="lo")
    ap.add_argument("--sizes", type=int, nargs="*",
                    default=[65536, 1 << 20, 4 << 20, 16 << 20, 64 << 20])
    ap.add_argument("--bytes-per-size", type=int, default=2 << 30)
    ap.add_argument("--max-msgs", type=int, default=2000)
    ap.add_argument("--json", action="store_true")
    ap.add_argument("--duplex", action="store_true",
                    help="both directions simultaneously (ring-edge pattern)")
    ap.add_argument("--shaped", type=float, default=None,
                    help="run the receiver behind a veth pair in its own "
                         "netns, egress-shaped to this many Gbit/s on both "
                         "devices (0 = veth, unshaped).  Needs "
                         "CAP_NET_ADMIN; see netns_rig.py")
    ap.add_argument("--delay-us", type=float, default=10.0,
                    help="one-way netem delay (ignored on kernels without "
                         "sch_netem, where TBF provides the rate ceiling)")
    ap.add_argument("--verify", action="store_true",
                    help="content-verify every message (deterministic "
                         "per-message patterns; soak mode, not peak rate)")
    ap.add_argument("--pairs", type=int, default=1,
                    help="N concurrent comms between the two processes "
                         "(fairness measurement; uses the FIRST size only)")
    args = ap.parse_args()
    if args.verify and (args.duplex or args.pairs > 1):
        ap.error("--verify supports the plain one-way mode")

    ctx = mp.get_context("spawn")
    a, b = ctx.Pipe()
    q = ctx.Queue()
    rig = None
    ns_parent = ns_child = None
    sender_args = args
    if args.shaped is not None:
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        import netns_rig as rig
        import copy

        ns_parent, ns_child = ctx.Pipe()
        args.ifname = rig.CHILD_IF  # receiver side (child netns)
        sender_args = copy.copy(args)
        sender_args.ifname = rig.PARENT_IF
    if args.duplex:
        pr = ctx.Process(target=_duplex_worker,
                         args=(a, args, q, True, ns_child))
        ps = ctx.Process(target=_duplex_worker,
                         args=(b, sender_args, q, False))
    elif args.pairs > 1:
        pr = ctx.Process(target=_receiver_pairs, args=(a, args, q, ns_child))
        ps = ctx.Process(target=_sender_pairs, args=(b, sender_args, q))
    else:
        pr = ctx.Process(target=_receiver, args=(a, args, q, ns_child))
        ps = ctx.Process(target=_sender, args=(b, sender_args, q))
    pr.start()
    if rig is not None:
        child_pid = ns_parent.recv()
        rig.parent_setup(child_pid)
        kind = None
        if args.shaped > 0:
            kind = rig.parent_shape(args.shaped, args.delay_us)
        ns_parent.send("veth-moved")
    ps.start()
    outs = [q.get(timeout=600), q.get(timeout=600)]
    pr.join(30)
    ps.join(30)
    if rig is not None:
        rig.parent_teardown()
    results = next(o for o in outs if isinstance(o, list))
    header = {
        "bench": "plugin p2p duplex" if args.duplex else "plugin p2p one-way",
        "nstreams": args.nstreams,
        "io_threads": args.io_threads,
        "ifname": args.ifname,
    }
    if args.shaped is not None:
        header["shaped_gbit"] = args.shaped
        header["qdisc"] = kind if args.shaped > 0 else "none"
    if args.json:
        print(json.dumps({**header, "results": results}))
    else:
        print(header)
        for r in results:
            bw = r.get("GBps", r.get("GBps_aggregate"))
            print(f"  {r['size']:>10} B x {r['msgs']:>5} msgs: "
                  f"{bw:8.3f} GB/s"
                  + (" (aggregate both ways)" if "GBps_aggregate" in r
                     else ""))


if __name__ == "__main__":
    main()
