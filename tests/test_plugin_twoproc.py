"""Two-process plugin test: receiver listens, ships the 128-byte NCCL handle
through a pipe (standing in for NCCL's bootstrap), sender connects; payloads
flow over real TCP sockets between distinct address spaces.
"""

import hashlib
import multiprocessing as mp
import os

SIZES = [0, 5, 4096, 1 << 16, (1 << 20) + 7, 1 << 22]
SEED = 1234


def _payload(i, size):
    import random

    rng = random.Random(SEED + i)
    block = bytes(rng.getrandbits(8) for _ in range(min(size, 65536)))
    if size > len(block):
        block = (block * (size // max(len(block), 1) + 1))[:size]
    return block


def _set_env():
    os.environ["NCCL_SOCKET_IFNAME"] = os.environ.get("BNET_TEST_IFNAME",
                                                      "lo")
    os.environ["BNET_MIN_CHUNKSIZE"] = "8192"
    os.environ["BNET_NSTREAMS"] = "3"


def _receiver(conn, result, engine=None):
    _set_env()
    if engine:
        os.environ["BNET_IMPLEMENT"] = engine
    import ctypes as C

    from baguanet.plugin import Plugin

    p = Plugin()
    handle, lcomm = p.listen(0)
    conn.send(bytes(handle))
    rcomm = None
    while rcomm is None:
        rcomm = p.accept(lcomm)
    mh = p.reg_mr(rcomm, None, 0)
    digests = []
    for i, size in enumerate(SIZES):
        buf = C.create_string_buffer(size + 1)
        req = None
        while req is None:
            req = p.irecv(rcomm, buf, size, mh)
        got = p.wait(req, timeout_s=60)
        assert got == size, f"msg {i}: got {got} != {size}"
        digests.append(hashlib.sha256(buf.raw[:size]).hexdigest())
    result.put(digests)
    conn.recv()  # wait for sender done before teardown
    p.close_recv(rcomm)
    p.close_listen(lcomm)


def _sender(conn, result, engine=None):
    _set_env()
    if engine:
        os.environ["BNET_IMPLEMENT"] = engine
    import ctypes as C

    from baguanet.plugin import Plugin

    p = Plugin()
    handle_bytes = conn.recv()
    handle = (C.c_char * len(handle_bytes)).from_buffer_copy(handle_bytes)
    scomm = None
    while scomm is None:
        scomm = p.connect(0, handle)
    mh = p.reg_mr(scomm, None, 0)
    for i, size in enumerate(SIZES):
        payload = _payload(i, size)
        buf = C.create_string_buffer(payload, max(size, 1))
        req = None
        while req is None:
            req = p.isend(scomm, buf, size, mh)
        sent = p.wait(req, timeout_s=60)
        assert sent == size
    result.put("sender-ok")
    conn.send("done")
    p.close_send(scomm)


def _run_pair(send_engine=None, recv_engine=None):
    ctx = mp.get_context("spawn")
    a, b = ctx.Pipe()
    res = ctx.Queue()
    pr = ctx.Process(target=_receiver, args=(a, res, recv_engine))
    ps = ctx.Process(target=_sender, args=(b, res, send_engine))
    pr.start()
    ps.start()
    outs = [res.get(timeout=120), res.get(timeout=120)]
    pr.join(30)
    ps.join(30)
    assert pr.exitcode == 0, "receiver failed"
    assert ps.exitcode == 0, "sender failed"
    digests = next(o for o in outs if isinstance(o, list))
    expect = [
        hashlib.sha256(_payload(i, s)).hexdigest() for i, s in enumerate(SIZES)
    ]
    assert digests == expect


def test_two_process_transfer():
    _run_pair()


def test_two_process_transfer_real_nic():
    """Same, over the default (non-loopback) interface — exercises the NIC
    addressing/bind path used inter-node."""
    import pytest

    try:
        ifaces = os.listdir("/sys/class/net")
    except OSError:
        ifaces = []
    if not [i for i in ifaces if i != "lo" and not i.startswith("docker")]:
        pytest.skip("no non-loopback interface")
    os.environ["BNET_TEST_IFNAME"] = "^docker,lo"
    try:
        _run_pair()
    finally:
        del os.environ["BNET_TEST_IFNAME"]


def test_cross_engine_interop():
    """The wire format is engine-independent: epoll sender <-> io_uring
    receiver and vice versa must interoperate byte-perfectly (the
    reference's BASIC and TOKIO backends could NOT interoperate — their
    length frames differed)."""
    _run_pair(send_engine="EPOLL", recv_engine="URING")
    _run_pair(send_engine="URING", recv_engine="EPOLL")


def test_ring_allreduce_cpu():
    """Verified ring all-reduce over the plugin at 3 ranks (odd count
    exercises non-power-of-two chunking)."""
    import json
    import subprocess
    import sys

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

    res = subprocess.run(
        [sys.executable,
         os.path.join(REPO, "benchmarks", "ring_allreduce.py"),
         "--ranks", "3", "--sizes", "999936", "--iters", "2",
         "--warmup", "1", "--json"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
        env={**os.environ, "BNET_IO_THREADS": "2"},
    )
    assert res.returncode == 0, res.stderr[-2000:]
    out = json.loads([l for l in res.stdout.splitlines()
                      if l.startswith("{")][-1])
    assert out["verified"] and out["ranks"] == 3
    assert out["results"][0]["busbw_GBps"] > 0
