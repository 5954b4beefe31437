#!/usr/bin/env python3
"""Soak test: sustained random traffic (mixed host/CUDA buffers, random
sizes, pipelined) through the plugin for --seconds, verifying every
payload.  Exercises slot reuse, staging ring churn, and parking under
load for much longer than the unit tests."""

from __future__ import annotations

import argparse
import ctypes as C
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tests"))


def _dump_stall(inflight, rreq, sreq, plugin=None, rcomm=None, scomm=None):
    """A refused post with depth < 32 means some message stalled — dump
    each in-flight entry's completion state before failing."""
    import ctypes as C

    print(f"POST REFUSED: rreq={rreq} sreq={sreq}; "
          f"{len(inflight)} in flight:")
    for i, it in enumerate(inflight):
        print(f"  [{i}] nbytes={it['nbytes']} sdone={it['sdone']} "
              f"rdone={it['rdone']}")
    if plugin is not None and rcomm is not None:
        buf = C.create_string_buffer(8192)
        plugin.lib.bnet_dump_recv_state(rcomm, buf, 8192)
        print("recv comm state:", buf.value.decode())
    if plugin is not None and scomm is not None:
        buf = C.create_string_buffer(8192)
        plugin.lib.bnet_dump_send_state(scomm, buf, 8192)
        print("send comm state:", buf.value.decode())
    raise AssertionError("post refused — stalled message (state above)")


_SIZES = [0, 64, 5000, 65536, 300_000, 1 << 20, 3 << 20]
if os.environ.get("BNET_SOAK_SIZES"):
    _SIZES = [int(x) for x in os.environ["BNET_SOAK_SIZES"].split(",")]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=30.0)
    ap.add_argument("--gpu", action="store_true")
    ap.add_argument("--seed", type=int, default=7)
    args = ap.parse_args()

    os.environ.setdefault("NCCL_SOCKET_IFNAME", "lo")
    os.environ.setdefault("BNET_MIN_CHUNKSIZE", "32768")
    from baguanet.plugin import Plugin
    from test_plugin_loopback import establish

    torch = None
    if args.gpu:
        import torch  # noqa: F811

    rng = random.Random(args.seed)
    p = Plugin()
    lcomm, scomm, rcomm = establish(p)
    smh_h = p.reg_mr(scomm, None, 0)
    rmh_h = p.reg_mr(rcomm, None, 0)

    # entries: dict(sreq, rreq, sdone, rdone, verify, nbytes) — a request is
    # tested ONLY until it reports done (test() frees the slot at done;
    # re-testing a freed handle is outside the ABI contract)
    inflight = []
    sent = 0
    bytes_total = 0
    t0 = time.monotonic()
    while time.monotonic() - t0 < args.seconds or inflight:
        while (len(inflight) < 12
               and time.monotonic() - t0 < args.seconds):
            size = rng.choice(_SIZES)
            use_gpu = args.gpu and torch is not None and rng.random() < 0.5
            if use_gpu and size > 0:
                n = max(size // 4, 1)
                src = torch.randn(n, device="cuda")
                dst = torch.zeros_like(src)
                # ncclNet semantics: buffers must be ready at isend/irecv
                # time (NCCL's proxy guarantees this before calling the
                # plugin) — the producer kernels run async, so sync first
                torch.cuda.synchronize()
                nbytes = n * 4
                smh = p.reg_mr(scomm, C.c_void_p(src.data_ptr()), nbytes,
                               0x2)
                rmh = p.reg_mr(rcomm, C.c_void_p(dst.data_ptr()), nbytes,
                               0x2)
                # depth 12 < 32 slots: posts can never be refused
                rreq = p.irecv(rcomm, C.c_void_p(dst.data_ptr()), nbytes,
                               rmh)
                sreq = p.isend(scomm, C.c_void_p(src.data_ptr()), nbytes,
                               smh)
                if rreq is None or sreq is None:
                    _dump_stall(inflight, rreq, sreq, p, rcomm, scomm)

                def verify(src=src, dst=dst):
                    torch.cuda.synchronize()
                    assert torch.equal(src, dst), "GPU payload corrupt"

                inflight.append(dict(sreq=sreq, rreq=rreq, sdone=False,
                                     rdone=False, verify=verify,
                                     nbytes=nbytes))
            else:
                payload = rng.randbytes(size) if size else b""
                sbuf = C.create_string_buffer(payload, max(size, 1))
                rbuf = C.create_string_buffer(size + 1)
                rreq = p.irecv(rcomm, rbuf, size, rmh_h)
                sreq = p.isend(scomm, sbuf, size, smh_h)
                if rreq is None or sreq is None:
                    _dump_stall(inflight, rreq, sreq, p, rcomm, scomm)

                def verify(payload=payload, rbuf=rbuf, sbuf=sbuf, size=size):
                    assert rbuf.raw[:size] == payload, "payload corrupt"

                inflight.append(dict(sreq=sreq, rreq=rreq, sdone=False,
                                     rdone=False, verify=verify,
                                     nbytes=size))
            sent += 1
        done_any = False
        for item in list(inflight):
            if not item["sdone"]:
                item["sdone"], _ = p.test(item["sreq"])
            if not item["rdone"]:
                item["rdone"], _ = p.test(item["rreq"])
            if item["sdone"] and item["rdone"]:
                item["verify"]()
                bytes_total += item["nbytes"]
                inflight.remove(item)
                done_any = True
        if not done_any:
            time.sleep(0)
    dt = time.monotonic() - t0
    print(f"soak ok: {sent} messages, {bytes_total/1e9:.2f} GB in "
          f"{dt:.1f}s ({bytes_total/dt/1e9:.2f} GB/s), all verified")
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)


if __name__ == "__main__":
    main()
