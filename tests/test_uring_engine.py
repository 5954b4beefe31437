"""io_uring engine (BNET_IMPLEMENT=URING): the full wire protocol must
behave identically to the epoll engine.  Subprocess-isolated (engine is
chosen once per process); skips where io_uring is seccomp-blocked."""

import multiprocessing as mp
import os

import pytest


def _uring_ok():
    import ctypes
    import ctypes.util

    libc = ctypes.CDLL(None, use_errno=True)
    # __NR_io_uring_setup == 425 on x86-64
    fd = libc.syscall(425, 4, ctypes.create_string_buffer(120))
    if fd >= 0:
        os.close(fd)
        return True
    return False


def _run(q):
    os.environ["BNET_IMPLEMENT"] = "URING"
    os.environ["NCCL_SOCKET_IFNAME"] = "lo"
    os.environ["BNET_MIN_CHUNKSIZE"] = "8192"
    os.environ["BNET_NSTREAMS"] = "4"
    import sys

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from baguanet.plugin import Plugin
    from test_plugin_loopback import establish, xfer

    p = Plugin()
    lcomm, scomm, rcomm = establish(p)
    for size in (0, 1, 100, 8192, 65536, 1 << 20, (1 << 22) + 13):
        payload = os.urandom(min(size, 1 << 16))
        if size > len(payload):
            payload = (payload * (size // max(len(payload), 1) + 1))[:size]
        assert xfer(p, scomm, rcomm, payload) == payload, size
    # pipelined burst
    import ctypes as C

    msgs = [os.urandom(5000 + 37 * i) for i in range(50)]
    smh = p.reg_mr(scomm, None, 0)
    rmh = p.reg_mr(rcomm, None, 0)
    rbufs = [C.create_string_buffer(len(m) + 1) for m in msgs]
    sbufs = [C.create_string_buffer(m, len(m)) for m in msgs]
    rreqs = [None] * len(msgs)
    sreqs = [None] * len(msgs)
    import time

    t0 = time.monotonic()
    ri = si = 0
    pending = set()
    while ri < len(msgs) or si < len(msgs) or pending:
        assert time.monotonic() - t0 < 60
        if ri < len(msgs):
            r = p.irecv(rcomm, rbufs[ri], len(msgs[ri]), rmh)
            if r is not None:
                rreqs[ri] = r
                pending.add(("r", ri))
                ri += 1
        if si < len(msgs):
            r = p.isend(scomm, sbufs[si], len(msgs[si]), smh)
            if r is not None:
                sreqs[si] = r
                pending.add(("s", si))
                si += 1
        for kind, i in list(pending):
            req = rreqs[i] if kind == "r" else sreqs[i]
            done, sz = p.test(req)
            if done:
                assert sz == len(msgs[i])
                pending.discard((kind, i))
    for i, m in enumerate(msgs):
        assert rbufs[i].raw[: len(m)] == m, f"msg {i} corrupt"
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


def test_uring_engine_wire_protocol():
    if not _uring_ok():
        pytest.skip("io_uring unavailable (seccomp)")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(target=_run, args=(q,))
    proc.start()
    assert q.get(timeout=180) == "ok"
    proc.join(30)
    assert proc.exitcode == 0


def test_uring_unavailable_falls_back(tmp_path):
    """BNET_IMPLEMENT=URING on a host without io_uring (seccomp) must
    fall back to the epoll engine and still move data (BNET_FORCE_NO_URING
    exercises the probe-failure path)."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code = (
        "import os, sys, ctypes as C;"
        f"sys.path.insert(0, {repo!r});"
        f"sys.path.insert(0, {os.path.join(repo, 'tests')!r});"
        "from baguanet.plugin import Plugin;"
        "from test_plugin_loopback import establish, xfer;"
        "p = Plugin(); l, s, r = establish(p);"
        "assert xfer(p, s, r, b'f' * 70000) == b'f' * 70000;"
        "buf = C.create_string_buffer(1024);"
        "p.lib.bnet_config_json(buf, 1024);"
        "print(buf.value.decode());"
        "p.close_send(s); p.close_recv(r); p.close_listen(l)"
    )
    env = dict(os.environ, NCCL_SOCKET_IFNAME="lo",
               BNET_IMPLEMENT="URING", BNET_FORCE_NO_URING="1",
               BNET_LOG="info")
    res = subprocess.run([sys.executable, "-c", code], env=env,
                         capture_output=True, text=True, timeout=120)
    assert res.returncode == 0, res.stderr[-1500:]
    # the engine must have REPORTED epoll (fallback), though config still
    # says URING was requested
    assert "engine: EPOLL" in res.stderr or "falling back" in res.stderr
