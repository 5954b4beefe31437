"""Config env knobs and failure-path behavior (subprocess-isolated)."""

import json
import multiprocessing as mp
import os


def _config_probe(env, q):
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C

    from baguanet.plugin import Plugin

    p = Plugin()
    buf = C.create_string_buffer(1024)
    p.lib.bnet_config_json(buf, 1024)
    q.put(buf.value.decode())


def _run_sub(target, env):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(target=target, args=(env, q))
    proc.start()
    out = q.get(timeout=120)
    proc.join(30)
    assert proc.exitcode == 0
    return out


def test_env_knobs_respected():
    cfg = json.loads(
        _run_sub(
            _config_probe,
            {
                "NCCL_SOCKET_IFNAME": "lo",
                "BNET_NSTREAMS": "7",
                "BNET_MIN_CHUNKSIZE": "65536",
                "BNET_MAX_CHUNKSIZE": "2097152",
                "BNET_IO_THREADS": "3",
                "BNET_SPIN_US": "77",
                "BNET_HELLO_TIMEOUT_MS": "12345",
                "BNET_STAGE_KERNEL": "1",
            },
        )
    )
    assert cfg["nstreams"] == 7
    assert cfg["min_chunk"] == 65536
    assert cfg["max_chunk"] == 2097152
    assert cfg["io_threads"] == 3
    assert cfg["spin_us"] == 77
    assert cfg["hello_timeout_ms"] == 12345
    assert cfg["stage_kernel"] == 1


def test_config_clamps():
    cfg = json.loads(
        _run_sub(
            _config_probe,
            {
                "NCCL_SOCKET_IFNAME": "lo",
                "BNET_NSTREAMS": "0",       # clamped to 1
                "BNET_MIN_CHUNKSIZE": "17",  # clamped to 4096
                "BNET_MAX_CHUNKSIZE": "1",   # clamped to >= min
            },
        )
    )
    assert cfg["nstreams"] == 1
    assert cfg["min_chunk"] == 4096
    assert cfg["max_chunk"] >= cfg["min_chunk"]


def _peer_death_receiver(env, q):
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C
    import sys
    import time

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from baguanet.plugin import Plugin

    p = Plugin()
    handle, lcomm = p.listen(0)
    q.put(bytes(handle))
    rcomm = None
    t0 = time.monotonic()
    while rcomm is None and time.monotonic() - t0 < 60:
        rcomm = p.accept(lcomm)
    assert rcomm is not None
    # wait until the sender is already dead before posting, so only the
    # few MB the kernel buffered exist and EOF arrives mid-message
    time.sleep(2.0)
    buf = C.create_string_buffer(256 << 20)
    mh = p.reg_mr(rcomm, buf, 256 << 20)
    req = p.irecv(rcomm, buf, 256 << 20, mh)
    q.put("posted")
    # the peer dies mid-protocol; test() must surface an error, not hang
    t0 = time.monotonic()
    while time.monotonic() - t0 < 60:
        try:
            done, _ = p.test(req)
        except RuntimeError as e:
            q.put(f"error-surfaced:{e}")
            return
        if done:
            q.put("unexpected-completion")
            return
        time.sleep(0.01)
    q.put("timeout-no-error")


def _peer_death_sender(env, handle_bytes, q):
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C

    from baguanet.plugin import Plugin

    p = Plugin()
    handle = (C.c_char * len(handle_bytes)).from_buffer_copy(handle_bytes)
    scomm = None
    while scomm is None:
        scomm = p.connect(0, handle)
    # start a send, then die abruptly without completing the protocol
    buf = C.create_string_buffer(256 << 20)
    mh = p.reg_mr(scomm, buf, 256 << 20)
    p.isend(scomm, buf, 256 << 20, mh)
    q.put("sender-started")
    import time

    time.sleep(0.3)  # let the queue feeder thread flush before hard exit
    os._exit(42)  # hard kill: sockets reset mid-message


def test_peer_death_surfaces_error():
    env = {"NCCL_SOCKET_IFNAME": "lo", "BNET_NSTREAMS": "2"}
    ctx = mp.get_context("spawn")
    qr, qs = ctx.Queue(), ctx.Queue()
    pr = ctx.Process(target=_peer_death_receiver, args=(env, qr))
    pr.start()
    handle = qr.get(timeout=60)
    ps = ctx.Process(target=_peer_death_sender, args=(env, handle, qs))
    ps.start()
    assert qr.get(timeout=60) == "posted"
    assert qs.get(timeout=60) == "sender-started"
    ps.join(30)
    result = qr.get(timeout=90)
    pr.join(30)
    assert result.startswith("error-surfaced"), result


def _garbage_then_legit(env, q):
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C
    import socket
    import struct
    import sys
    import time

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from baguanet.plugin import Plugin

    p = Plugin()
    handle, lcomm = p.listen(0)
    # decode the port from the handle: {u32 magic, u16 family, u16 port_be}
    magic, family, port_be = struct.unpack_from("<IHH", bytes(handle))
    port = socket.ntohs(port_be)
    # throw garbage at the listener: wrong magic, short writes, instant close
    for payload in (b"GET / HTTP/1.0\r\n\r\n", b"\x00" * 3, b""):
        g = socket.create_connection(("127.0.0.1", port), timeout=5)
        if payload:
            g.sendall(payload)
        g.close()
        # pump accept so it processes the junk
        t0 = time.monotonic()
        while time.monotonic() - t0 < 0.2:
            assert p.accept(lcomm) is None
    # a legitimate connection must still succeed
    scomm = rcomm = None
    t0 = time.monotonic()
    while scomm is None or rcomm is None:
        assert time.monotonic() - t0 < 30
        if scomm is None:
            scomm = p.connect(0, handle)
        if rcomm is None:
            rcomm = p.accept(lcomm)
    buf = C.create_string_buffer(b"hello-after-garbage", 19)
    rbuf = C.create_string_buffer(20)
    mh = p.reg_mr(scomm, None, 0)
    rreq = p.irecv(rcomm, rbuf, 19, mh)
    sreq = p.isend(scomm, buf, 19, mh)
    assert p.wait(sreq, 30) == 19 and p.wait(rreq, 30) == 19
    assert rbuf.raw[:19] == b"hello-after-garbage"
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


def test_garbage_connections_rejected():
    """Junk connections to a listener must not break legitimate accepts."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(
        target=_garbage_then_legit,
        args=({"NCCL_SOCKET_IFNAME": "lo"}, q),
    )
    proc.start()
    assert q.get(timeout=120) == "ok"
    proc.join(30)
    assert proc.exitcode == 0


def _malformed_frames(env, q):
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C
    import socket
    import struct
    import time

    from baguanet.plugin import Plugin

    p = Plugin()
    handle, lcomm = p.listen(0)
    magic, family, port_be = struct.unpack_from("<IHH", bytes(handle))
    port = socket.ntohs(port_be)
    # fake a single-stream sender: valid hello, then a malformed chunk
    conn_id = 0xDEADBEEFCAFEF00D
    hello = struct.pack("<IIQHHI", magic, 2, conn_id, 0, 1, 0)
    g = socket.create_connection(("127.0.0.1", port), timeout=5)
    g.sendall(hello)
    rcomm = None
    t0 = time.monotonic()
    while rcomm is None and time.monotonic() - t0 < 30:
        rcomm = p.accept(lcomm)
    assert rcomm is not None
    buf = C.create_string_buffer(4096)
    mh = p.reg_mr(rcomm, buf, 4096)
    req = p.irecv(rcomm, buf, 4096, mh)
    assert req is not None
    # header with len > total — must surface an error, not hang
    g.sendall(struct.pack("<IIIIi", 0, 0, 4096, 16, 0))
    t0 = time.monotonic()
    while time.monotonic() - t0 < 30:
        try:
            done, _ = p.test(req)
        except RuntimeError:
            q.put("errored")
            g.close()
            return
        if done:
            q.put("unexpected-done")
            return
        time.sleep(0.01)
    q.put("timeout")


def test_malformed_frame_errors_not_hangs():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(target=_malformed_frames,
                       args=({"NCCL_SOCKET_IFNAME": "lo"}, q))
    proc.start()
    assert q.get(timeout=120) == "errored"
    proc.join(30)
    assert proc.exitcode == 0


def _iso_receiver(env, q, ctl):
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C
    import sys
    import time

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from baguanet.plugin import Plugin

    p = Plugin()
    # two listeners: one for the doomed sender, one for the healthy one
    h1, l1 = p.listen(0)
    h2, l2 = p.listen(0)
    q.put((bytes(h1), bytes(h2)))
    r1 = r2 = None
    t0 = time.monotonic()
    while (r1 is None or r2 is None) and time.monotonic() - t0 < 60:
        if r1 is None:
            r1 = p.accept(l1)
        if r2 is None:
            r2 = p.accept(l2)
    assert r1 is not None and r2 is not None
    mh = p.reg_mr(r2, None, 0)
    ctl.recv()  # wait until the doomed sender is dead
    # post the recv only now: the kernel buffered a few MB of the 256 MB
    # message before the sender died, so EOF lands mid-message
    doomed_buf = C.create_string_buffer(256 << 20)
    dmh = p.reg_mr(r1, doomed_buf, 256 << 20)
    doomed = p.irecv(r1, doomed_buf, 256 << 20, dmh)
    # the healthy comm must still move data correctly
    for i in range(5):
        buf = C.create_string_buffer(100000 + 1)
        req = None
        while req is None:
            req = p.irecv(r2, buf, 100000, mh)
        assert p.wait(req, 60) == 100000
        assert buf.raw[:4] == bytes([i] * 4)
    # and the doomed comm must surface its error
    t0 = time.monotonic()
    while time.monotonic() - t0 < 60:
        try:
            done, _ = p.test(doomed)
        except RuntimeError:
            q.put("ok")
            return
        assert not done
        time.sleep(0.01)
    q.put("doomed-no-error")


def _iso_doomed_sender(env, handle_bytes, q):
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C
    import time

    from baguanet.plugin import Plugin

    p = Plugin()
    handle = (C.c_char * len(handle_bytes)).from_buffer_copy(handle_bytes)
    scomm = None
    while scomm is None:
        scomm = p.connect(0, handle)
    buf = C.create_string_buffer(256 << 20)
    mh = p.reg_mr(scomm, buf, 256 << 20)
    p.isend(scomm, buf, 256 << 20, mh)
    q.put("started")
    time.sleep(0.3)
    os._exit(42)


def _iso_healthy_sender(env, handle_bytes, q):
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C

    from baguanet.plugin import Plugin

    p = Plugin()
    handle = (C.c_char * len(handle_bytes)).from_buffer_copy(handle_bytes)
    scomm = None
    while scomm is None:
        scomm = p.connect(0, handle)
    mh = p.reg_mr(scomm, None, 0)
    for i in range(5):
        payload = bytes([i] * 4) + os.urandom(100000 - 4)
        buf = C.create_string_buffer(payload, 100000)
        req = None
        while req is None:
            req = p.isend(scomm, buf, 100000, mh)
        assert p.wait(req, 60) == 100000
    q.put("healthy-done")
    p.close_send(scomm)


def test_error_isolation_between_comms():
    """A dead peer errors ONLY its own comm; other comms keep flowing."""
    env = {"NCCL_SOCKET_IFNAME": "lo", "BNET_NSTREAMS": "2"}
    ctx = mp.get_context("spawn")
    qr, qd, qh = ctx.Queue(), ctx.Queue(), ctx.Queue()
    ca, cb = ctx.Pipe()
    pr = ctx.Process(target=_iso_receiver, args=(env, qr, ca))
    pr.start()
    h1, h2 = qr.get(timeout=60)
    pd = ctx.Process(target=_iso_doomed_sender, args=(env, h1, qd))
    ph = ctx.Process(target=_iso_healthy_sender, args=(env, h2, qh))
    pd.start()
    ph.start()
    assert qd.get(timeout=60) == "started"
    pd.join(30)
    cb.send("doomed-dead")
    assert qh.get(timeout=90) == "healthy-done"
    assert qr.get(timeout=120) == "ok"
    pr.join(30)
    ph.join(30)
    assert pr.exitcode == 0


# ---------------------------------------------------------------------------
# Functional roundtrip under extreme config corners (subprocess-isolated:
# the .so reads its config once per process).  Plumbing-level knob tests
# above prove the values land; these prove the transport still moves and
# verifies data at the edges of the config space.


def _roundtrip_probe(env, q):
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C

    from baguanet.plugin import Plugin
    from tests.test_plugin_loopback import establish

    p = Plugin()
    lcomm, scomm, rcomm = establish(p)
    smh = p.reg_mr(scomm, None, 0)
    rmh = p.reg_mr(rcomm, None, 0)
    for size in [0, 1, 8191, 65536, 300_000, 1 << 20]:
        payload = bytes((i * 131 + 7) % 256 for i in range(size))
        sbuf = C.create_string_buffer(payload, max(size, 1))
        rbuf = C.create_string_buffer(size + 1)
        rreq = p.irecv(rcomm, rbuf, size, rmh)
        sreq = p.isend(scomm, sbuf, size, smh)
        assert rreq is not None and sreq is not None
        sdone = rdone = False
        import time as _t

        t0 = _t.monotonic()
        while not (sdone and rdone):
            if not sdone:
                sdone, _ = p.test(sreq)
            if not rdone:
                rdone, got = p.test(rreq)
                if rdone:
                    assert got == size, (got, size)
            assert _t.monotonic() - t0 < 30, f"stall at size={size}"
        assert rbuf.raw[:size] == payload, f"corrupt at size={size}"
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


_MATRIX = [
    # (nstreams, io_threads, min_chunk, engine) — config-space corners
    ("1", "1", "8192", "EPOLL"),    # no striping, single IO thread
    ("8", "1", "4096", "EPOLL"),    # 8 sockets multiplexed on 1 thread
    ("3", "4", "4096", "EPOLL"),    # odd stream count, tiny chunks
    ("8", "2", "4096", "URING"),    # uring multiplexing
    ("1", "1", "8192", "URING"),    # uring no striping
]


def test_roundtrip_config_matrix():
    for ns, iot, mc, eng in _MATRIX:
        env = {
            "NCCL_SOCKET_IFNAME": "lo",
            "BNET_NSTREAMS": ns,
            "BNET_IO_THREADS": iot,
            "BNET_MIN_CHUNKSIZE": mc,
            "BNET_IMPLEMENT": eng,
        }
        assert _run_sub(_roundtrip_probe, env) == "ok", (ns, iot, mc, eng)


def _churn_probe(env, q):
    """200 establish/transfer/close cycles: stresses engine add/remove
    (the SyncToken teardown handshake) and listen-socket lifecycle."""
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C

    from baguanet.plugin import Plugin
    from tests.test_plugin_loopback import establish

    p = Plugin()
    for cycle in range(200):
        lcomm, scomm, rcomm = establish(p)
        smh = p.reg_mr(scomm, None, 0)
        rmh = p.reg_mr(rcomm, None, 0)
        size = (cycle * 7919) % 100_000
        payload = bytes((cycle + j) % 256 for j in range(size))
        sbuf = C.create_string_buffer(payload, max(size, 1))
        rbuf = C.create_string_buffer(size + 1)
        rreq = p.irecv(rcomm, rbuf, size, rmh)
        sreq = p.isend(scomm, sbuf, size, smh)
        assert rreq is not None and sreq is not None
        import time as _t

        t0 = _t.monotonic()
        sdone = rdone = False
        while not (sdone and rdone):
            if not sdone:
                sdone, _ = p.test(sreq)
            if not rdone:
                rdone, got = p.test(rreq)
            assert _t.monotonic() - t0 < 30, f"cycle {cycle} stalled"
        assert rbuf.raw[:size] == payload
        p.close_send(scomm)
        p.close_recv(rcomm)
        p.close_listen(lcomm)
    q.put("ok")


def test_comm_churn():
    for eng in ("EPOLL", "URING"):
        env = {"NCCL_SOCKET_IFNAME": "lo", "BNET_NSTREAMS": "4",
               "BNET_IMPLEMENT": eng}
        assert _run_sub(_churn_probe, env) == "ok", eng


def _stalled_hello_reaped(env, q):
    """A connection that never completes its WireHello is reaped after
    BNET_HELLO_TIMEOUT_MS and a legitimate connect still succeeds."""
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C
    import socket
    import struct
    import time

    from baguanet.plugin import Plugin

    p = Plugin()
    handle, lcomm = p.listen(0)
    magic, family, port_be = struct.unpack_from("<IHH", bytes(handle))
    port = socket.ntohs(port_be)
    # half-open: send 3 bytes of hello, then go silent (socket stays open)
    stalled = socket.create_connection(("127.0.0.1", port), timeout=5)
    stalled.sendall(b"\xe7\xa4")
    t0 = time.monotonic()
    while time.monotonic() - t0 < 1.0:  # > BNET_HELLO_TIMEOUT_MS=300
        assert p.accept(lcomm) is None
        time.sleep(0.02)
    # the listener must have closed its side by now
    stalled.settimeout(2)
    assert stalled.recv(1) == b"", "stalled half-conn not reaped"
    stalled.close()
    # a real connection still works
    scomm = rcomm = None
    t0 = time.monotonic()
    while scomm is None or rcomm is None:
        assert time.monotonic() - t0 < 30
        if scomm is None:
            scomm = p.connect(0, handle)
        if rcomm is None:
            rcomm = p.accept(lcomm)
    buf = C.create_string_buffer(b"post-reap", 9)
    rbuf = C.create_string_buffer(10)
    mh = p.reg_mr(scomm, None, 0)
    rreq = p.irecv(rcomm, rbuf, 9, mh)
    sreq = p.isend(scomm, buf, 9, mh)
    assert p.wait(sreq, 30) == 9 and p.wait(rreq, 30) == 9
    assert rbuf.raw[:9] == b"post-reap"
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


def test_stalled_hello_reaped():
    env = {"NCCL_SOCKET_IFNAME": "lo", "BNET_NSTREAMS": "2",
           "BNET_HELLO_TIMEOUT_MS": "300"}
    assert _run_sub(_stalled_hello_reaped, env) == "ok"


def _connect_dead_listener(env, q):
    """connect() to a listener that has already closed must surface an
    error (not hang, not crash) and leave the plugin usable."""
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C
    import time

    from baguanet.plugin import Plugin

    p = Plugin()
    handle, lcomm = p.listen(0)
    handle_bytes = bytes(handle)
    p.close_listen(lcomm)  # port now dead (refused)
    dead = (C.c_char * len(handle_bytes)).from_buffer_copy(handle_bytes)
    t0 = time.monotonic()
    saw_error = False
    while time.monotonic() - t0 < 20:
        try:
            scomm = p.connect(0, dead)
        except RuntimeError:
            saw_error = True  # ncclRemoteError surfaced
            break
        assert scomm is None  # must never "succeed"
    assert saw_error, "connect to dead listener neither errored nor refused"
    # plugin still usable afterwards
    from tests.test_plugin_loopback import establish

    l2, s2, r2 = establish(p)
    buf = C.create_string_buffer(b"alive", 5)
    rbuf = C.create_string_buffer(6)
    mh = p.reg_mr(s2, None, 0)
    rreq = p.irecv(r2, rbuf, 5, mh)
    sreq = p.isend(s2, buf, 5, mh)
    assert p.wait(sreq, 30) == 5 and p.wait(rreq, 30) == 5
    assert rbuf.raw[:5] == b"alive"
    p.close_send(s2)
    p.close_recv(r2)
    p.close_listen(l2)
    q.put("ok")


def test_connect_dead_listener():
    env = {"NCCL_SOCKET_IFNAME": "lo", "BNET_NSTREAMS": "2"}
    assert _run_sub(_connect_dead_listener, env) == "ok"


def _oversized_send_errors(env, q):
    """A peer that sends MORE than the posted recv size (an ncclNet
    contract violation) must error the recv comm, not overrun the
    buffer or hang."""
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C
    import time

    from baguanet.plugin import Plugin
    from tests.test_plugin_loopback import establish

    p = Plugin()
    lcomm, scomm, rcomm = establish(p)
    mh = p.reg_mr(scomm, None, 0)
    rbuf = C.create_string_buffer(100 + 1)
    rreq = p.irecv(rcomm, rbuf, 100, mh)       # receiver expects <= 100
    sbuf = C.create_string_buffer(b"\xAB" * 5000, 5000)
    sreq = p.isend(scomm, sbuf, 5000, mh)      # sender violates: 5000
    assert rreq is not None and sreq is not None
    t0 = time.monotonic()
    while True:
        try:
            done, _ = p.test(rreq)
        except RuntimeError:
            break  # EMSGSIZE surfaced as comm error
        assert not done, "oversized message must not complete"
        assert time.monotonic() - t0 < 30, "no error surfaced"
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


def test_oversized_send_errors():
    env = {"NCCL_SOCKET_IFNAME": "lo", "BNET_NSTREAMS": "2"}
    assert _run_sub(_oversized_send_errors, env) == "ok"


def _ipv6_roundtrip(env, q):
    """NCCL_SOCKET_FAMILY=10 (AF_INET6): the sockaddr_in6 handle/bind/
    connect paths must carry data end-to-end over ::1."""
    for k, v in env.items():
        os.environ[k] = v
    from baguanet.plugin import Plugin
    from tests.test_plugin_loopback import establish, xfer

    p = Plugin()
    if p.ndev() == 0:
        q.put("skip")  # no IPv6 loopback in this environment
        return
    lcomm, scomm, rcomm = establish(p)
    payload = b"v6" * 30000
    assert xfer(p, scomm, rcomm, payload) == payload
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


def test_ipv6_family():
    env = {"NCCL_SOCKET_IFNAME": "lo", "NCCL_SOCKET_FAMILY": "10",
           "BNET_NSTREAMS": "3", "BNET_MIN_CHUNKSIZE": "8192"}
    res = _run_sub(_ipv6_roundtrip, env)
    assert res in ("ok", "skip")


def _tag_roundtrip(env, q):
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C
    import time

    from baguanet.plugin import Plugin

    p = Plugin()
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
    # matching tags: completes normally
    buf = C.create_string_buffer(b"tagged-payload-ok", 17)
    rbuf = C.create_string_buffer(17)
    rreq = p.irecv(rcomm, rbuf, 17, mh, tag=42)
    sreq = p.isend(scomm, buf, 17, mh, tag=42)
    assert p.wait(sreq, 30) == 17 and p.wait(rreq, 30) == 17
    assert rbuf.raw[:17] == b"tagged-payload-ok"
    # mismatched tags: the echoed tag turns a silent cross-match into a
    # loud EPROTO on the recv comm
    rreq = p.irecv(rcomm, rbuf, 17, mh, tag=9)
    sreq = p.isend(scomm, buf, 17, mh, tag=7)
    t0 = time.monotonic()
    while time.monotonic() - t0 < 30:
        try:
            done, _ = p.test(rreq)
        except RuntimeError:
            q.put("errored")
            return
        if done:
            q.put("unexpected-done")
            return
        time.sleep(0.01)
    q.put("timeout")


def test_tag_echo_mismatch_is_loud():
    """isend tags are echoed on the wire; a cross-match errors the comm."""
    assert _run_sub(_tag_roundtrip, {"NCCL_SOCKET_IFNAME": "lo"}) == "errored"


def _touch_plugin(env, q):
    for k, v in env.items():
        os.environ[k] = v
    from baguanet.plugin import Plugin

    Plugin()  # init reads config; atexit dumps telemetry
    q.put("ok")


def test_rank_templated_telemetry_files(tmp_path):
    """N-rank jobs must leave N distinguishable metric files:
    RANK is appended as .r<rank> (or substituted for %r), and counters
    carry a rank label."""
    base = str(tmp_path / "m.prom")
    assert _run_sub(_touch_plugin, {
        "NCCL_SOCKET_IFNAME": "lo",
        "RANK": "3",
        "BNET_METRICS_FILE": base,
    }) == "ok"
    f = tmp_path / "m.prom.r3"
    assert f.exists(), "expected rank-suffixed metrics file"
    assert 'bnet_isend_total{rank="3"}' in f.read_text()

    templ = str(tmp_path / "rank-%r.prom")
    assert _run_sub(_touch_plugin, {
        "NCCL_SOCKET_IFNAME": "lo",
        "BNET_RANK": "5",
        "BNET_METRICS_FILE": templ,
    }) == "ok"
    assert (tmp_path / "rank-5.prom").exists()


def _pinned_stats(env, q):
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C

    from baguanet.plugin import Plugin

    p = Plugin()
    used = C.c_size_t(1)
    budget = C.c_size_t(0)
    p.lib.bnet_pinned_stats(C.byref(used), C.byref(budget))
    q.put((used.value, budget.value))


def test_pinned_budget_knob():
    """BNET_PINNED_BUDGET reaches the staging accountant (no GPU needed:
    nothing is pinned yet, but the budget must reflect the env)."""
    used, budget = _run_sub(_pinned_stats, {
        "NCCL_SOCKET_IFNAME": "lo",
        "BNET_PINNED_BUDGET": str(123 << 20),
    })
    assert used == 0
    assert budget == 123 << 20
