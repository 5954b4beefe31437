#!/usr/bin/env python3
"""Soak test: sustained random traffic (mixed host/CUDA buffers, random
sizes, pipelined) through the plugin for --seconds, verifying every
payload.  Exercises slot reuse, staging ring churn, and parking under
load for much longer than the unit tests.

NCCL-faithful driver: refused posts (request=NULL) are backpressure and
are retried while completions are polled; only a >30 s stall is a
failure (it then dumps transport state via bnet_dump_*).  This harness
found two real transport bugs in round 1 (the claim-cursor ABA race and
a lost-wakeup in the epoll task drain) — keep it brutal.
"""

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

SIZES = [0, 64, 5000, 65536, 300_000, 1 << 20, 3 << 20]
if os.environ.get("BNET_SOAK_SIZES"):
    SIZES = [int(x) for x in os.environ["BNET_SOAK_SIZES"].split(",")]

DEPTH = 12


def dump_state(p, scomm, rcomm, live):
    print(f"STALL: {len(live)} in flight:")
    for i, m in enumerate(live):
        print(f"  [{i}] size={m['size']} gpu={m['gpu']} "
              f"sdone={m['sdone']} rdone={m['rdone']}")
    buf = C.create_string_buffer(8192)
    p.lib.bnet_dump_send_state(scomm, buf, 8192)
    print("send:", buf.value.decode())
    p.lib.bnet_dump_recv_state(rcomm, buf, 8192)
    print("recv:", buf.value.decode())
    raise AssertionError("transport stalled >30s (state above)")


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

        if not torch.cuda.is_available():
            sys.exit("--gpu requires a visible GPU "
                     "(torch.cuda.is_available() is False)")

    rng = random.Random(args.seed)
    p = Plugin()
    lcomm, scomm, rcomm = establish(p)
    smh = p.reg_mr(scomm, None, 0)
    rmh = p.reg_mr(rcomm, None, 0)

    live = []
    completed = 0
    bytes_total = 0
    t_start = time.monotonic()
    deadline = t_start + args.seconds
    last_progress = time.monotonic()

    def poll():
        nonlocal completed, bytes_total, last_progress
        for m in list(live):
            if m.get("group"):
                if not m["sdone"]:
                    m["sdone"] = all(p.test(s)[0] for s in m["sreqs"])
                if not m["rdone"]:
                    m["rdone"], szs = p.test_n(m["rreq"], m["n"])
                    if m["rdone"]:
                        assert szs == m["gsizes"], (szs, m["gsizes"])
                if m["sdone"] and m["rdone"]:
                    m["verify"]()
                    live.remove(m)
                    completed += m["n"]
                    bytes_total += m["size"]
                    last_progress = time.monotonic()
                continue
            if not m["sdone"]:
                m["sdone"], _ = p.test(m["sreq"])
            if not m["rdone"]:
                m["rdone"], sz = p.test(m["rreq"])
                if m["rdone"]:
                    assert sz == m["size"], (sz, m["size"])
            if m["sdone"] and m["rdone"]:
                m["verify"]()
                live.remove(m)
                completed += 1
                bytes_total += m["size"]
                last_progress = time.monotonic()
        if live and time.monotonic() - last_progress > 30:
            dump_state(p, scomm, rcomm, live)

    while time.monotonic() < deadline or live:
        while len(live) < DEPTH and time.monotonic() < deadline:
            size = rng.choice(SIZES)
            use_gpu = args.gpu and torch is not None and size > 0 \
                and rng.random() < 0.5
            if use_gpu:
                n = max(size // 4, 1)
                src = torch.randn(n, device="cuda")
                dst = torch.zeros_like(src)
                # ncclNet contract: buffers ready at post time
                torch.cuda.synchronize()
                nbytes = n * 4
                gsmh = p.reg_mr(scomm, C.c_void_p(src.data_ptr()), nbytes,
                                0x2)
                grmh = p.reg_mr(rcomm, C.c_void_p(dst.data_ptr()), nbytes,
                                0x2)
                rreq = p.irecv(rcomm, C.c_void_p(dst.data_ptr()), nbytes,
                               grmh)
                if rreq is None:  # bounce-pool backpressure
                    poll()
                    break
                sreq = p.isend(scomm, C.c_void_p(src.data_ptr()), nbytes,
                               gsmh)
                t0 = time.monotonic()
                while sreq is None:  # staging-pool backpressure
                    poll()
                    sreq = p.isend(scomm, C.c_void_p(src.data_ptr()),
                                   nbytes, gsmh)
                    if time.monotonic() - t0 > 30:
                        dump_state(p, scomm, rcomm, live)

                def verify(src=src, dst=dst):
                    torch.cuda.synchronize()
                    assert torch.equal(src, dst), "GPU payload corrupt"

                live.append(dict(sreq=sreq, rreq=rreq, sdone=False,
                                 rdone=False, verify=verify, size=nbytes,
                                 gpu=True))
            elif rng.random() < 0.2:
                # grouped receive: 2-4 messages posted as ONE request
                n = rng.randint(2, 4)
                gsizes = [rng.choice(SIZES) for _ in range(n)]
                payloads = [(bytes([rng.getrandbits(8)]) * s) for s in
                            gsizes]
                sbufs = [C.create_string_buffer(pl, max(s, 1))
                         for pl, s in zip(payloads, gsizes)]
                rbufs = [C.create_string_buffer(s + 1) for s in gsizes]
                tags = [rng.randrange(1 << 16) for _ in range(n)]
                greq, _ = p.irecv_n(rcomm, rbufs, gsizes, rmh, tags)
                if greq is None:  # slot backpressure
                    poll()
                    break
                sreqs = []
                t0 = time.monotonic()
                for sb, s, t in zip(sbufs, gsizes, tags):
                    sr = p.isend(scomm, sb, s, smh, tag=t)
                    while sr is None:
                        poll()
                        sr = p.isend(scomm, sb, s, smh, tag=t)
                        if time.monotonic() - t0 > 30:
                            dump_state(p, scomm, rcomm, live)
                    sreqs.append(sr)

                def verify(payloads=payloads, rbufs=rbufs, gsizes=gsizes):
                    for pl, rb, s in zip(payloads, rbufs, gsizes):
                        assert rb.raw[:s] == pl, "grouped payload corrupt"

                live.append(dict(group=True, sreqs=sreqs, rreq=greq, n=n,
                                 gsizes=gsizes, verify=verify,
                                 size=sum(gsizes), sdone=False,
                                 rdone=False, gpu=False,
                                 _bufs=(sbufs, rbufs)))  # keep alive
            else:
                payload = (bytes([rng.getrandbits(8)]) * size) if size \
                    else b""
                sbuf = C.create_string_buffer(payload, max(size, 1))
                rbuf = C.create_string_buffer(size + 1)
                rreq = p.irecv(rcomm, rbuf, size, rmh)
                if rreq is None:
                    poll()
                    break
                sreq = p.isend(scomm, sbuf, size, smh)
                t0 = time.monotonic()
                while sreq is None:
                    poll()
                    sreq = p.isend(scomm, sbuf, size, smh)
                    if time.monotonic() - t0 > 30:
                        dump_state(p, scomm, rcomm, live)

                def verify(payload=payload, rbuf=rbuf, sbuf=sbuf,
                           size=size):
                    assert rbuf.raw[:size] == payload, "payload corrupt"

                live.append(dict(sreq=sreq, rreq=rreq, sdone=False,
                                 rdone=False, verify=verify, size=size,
                                 gpu=False))
        poll()

    dt = time.monotonic() - t_start
    print(f"soak ok: {completed} messages, {bytes_total/1e9:.2f} GB in "
          f"{dt:.1f}s ({bytes_total/dt/1e9:.2f} GB/s), all verified")
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)


if __name__ == "__main__":
    main()
