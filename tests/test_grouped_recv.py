"""Grouped receives (irecv n>1, properties.maxRecvs=4).

The RCCL proxy aggregates small receives into one irecv when the plugin
advertises maxRecvs>1: one request posts n buffers/tags, the n next sends
land in them in order (per-comm FIFO), and test() reports n sizes at once.
The seq-slot design extends naturally: a group is n consecutive slots and
the echoed wire tag verifies member alignment.
"""

import ctypes as C
import multiprocessing as mp
import os

import pytest


def _grouped(env, q):
    for k, v in env.items():
        os.environ[k] = v
    import time

    from baguanet.plugin import Plugin

    p = Plugin()
    assert p.properties(0)["maxRecvs"] == 4
    handle, lcomm = p.listen(0)
    scomm = rcomm = None
    t0 = time.monotonic()
    while scomm is None or rcomm is None:
        assert time.monotonic() - t0 < 30
        if scomm is None:
            scomm = p.connect(0, handle)
        if rcomm is None:
            rcomm = p.accept(lcomm)
    mh = p.reg_mr(scomm, None, 0)

    # 1) group of 3 (one zero-size member), distinct tags
    payloads = [b"alpha-payload", b"", os.urandom(70000)]
    tags = [11, 22, 33]
    bufs = [C.create_string_buffer(max(len(x), 1)) for x in payloads]
    req, n = p.irecv_n(rcomm, bufs, [len(x) for x in payloads], mh, tags)
    assert req is not None
    for x, t in zip(payloads, tags):
        sb = C.create_string_buffer(x, max(len(x), 1))
        sreq = p.isend(scomm, sb, len(x), mh, tag=t)
        assert p.wait(sreq, 30) == len(x)
    t0 = time.monotonic()
    while True:
        done, sizes = p.test_n(req, n)
        if done:
            break
        assert time.monotonic() - t0 < 30
    assert sizes == [len(x) for x in payloads]
    for b, x in zip(bufs, payloads):
        assert b.raw[: len(x)] == x

    # 2) back-to-back groups interleaved with singles keep FIFO order
    msgs = [os.urandom(2000 + 137 * i) for i in range(6)]
    b2 = [C.create_string_buffer(len(x)) for x in msgs]
    g1, n1 = p.irecv_n(rcomm, b2[0:2], [len(x) for x in msgs[0:2]], mh,
                       [1, 2])
    single = p.irecv(rcomm, b2[2], len(msgs[2]), mh, tag=3)
    g2, n2 = p.irecv_n(rcomm, b2[3:6], [len(x) for x in msgs[3:6]], mh,
                       [4, 5, 6])
    assert g1 is not None and single is not None and g2 is not None
    for i, x in enumerate(msgs):
        sb = C.create_string_buffer(x, len(x))
        sreq = p.isend(scomm, sb, len(x), mh, tag=i + 1)
        assert p.wait(sreq, 30) == len(x)
    for req_, nn in ((g1, n1), (single, 1), (g2, n2)):
        t0 = time.monotonic()
        while True:
            done, _ = p.test_n(req_, nn)
            if done:
                break
            assert time.monotonic() - t0 < 30
    for b, x in zip(b2, msgs):
        assert b.raw[: len(x)] == x

    # 3) a tag mismatch INSIDE a group is a loud comm error
    gb = [C.create_string_buffer(64) for _ in range(2)]
    req3, _ = p.irecv_n(rcomm, gb, [64, 64], mh, [7, 8])
    sb = C.create_string_buffer(b"x" * 64, 64)
    p.isend(scomm, sb, 64, mh, tag=7)
    p.isend(scomm, sb, 64, mh, tag=999)  # mismatches member tag 8
    t0 = time.monotonic()
    while time.monotonic() - t0 < 30:
        try:
            done, _ = p.test_n(req3, 2)
        except RuntimeError:
            q.put("ok")
            return
        if done:
            q.put("unexpected-done")
            return
    q.put("timeout")


def test_grouped_recv():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(
        target=_grouped,
        args=({"NCCL_SOCKET_IFNAME": "lo", "BNET_MIN_CHUNKSIZE": "8192"}, q),
    )
    proc.start()
    assert q.get(timeout=120) == "ok"
    proc.join(30)
    assert proc.exitcode == 0


@pytest.mark.parametrize("engine", ["EPOLL", "URING"])
def test_grouped_recv_both_engines(engine):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(
        target=_grouped,
        args=({"NCCL_SOCKET_IFNAME": "lo", "BNET_IMPLEMENT": engine}, q),
    )
    proc.start()
    assert q.get(timeout=120) == "ok"
    proc.join(30)
    assert proc.exitcode == 0
