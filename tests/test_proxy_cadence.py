"""RCCL-proxy calling-pattern tests: the awkward handle lifetimes the
real proxy can produce around nonblocking connect retries.

VERDICT r1 weak #7: connect() used to trust the raw ConnectTask pointer
stashed in the NCCL handle; if RCCL re-copies the ORIGINAL handle bytes
over a retried one (or hands a stale/corrupted stage field), that pointer
dereferences garbage.  The live-task registry + conn_id token now make
every such shape safe — these tests drive each one through the exported
v8 vtable (the version RCCL actually picks).
"""

import ctypes as C
import multiprocessing as mp
import os


def _run_sub(target, env):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(target=target, args=(env, q))
    proc.start()
    out = q.get(timeout=120)
    proc.join(30)
    assert proc.exitcode == 0
    return out


def _pump(p, handle, lcomm, timeout=30.0, mutate=None):
    import time

    scomm = rcomm = None
    attempt = 0
    t0 = time.monotonic()
    while scomm is None or rcomm is None:
        assert time.monotonic() - t0 < timeout
        if scomm is None:
            if mutate:
                handle = mutate(handle, attempt)
            attempt += 1
            scomm = p.connect(0, handle)
        if rcomm is None:
            rcomm = p.accept(lcomm)
    return scomm, rcomm


def _xfer_ok(p, scomm, rcomm):
    buf = C.create_string_buffer(b"proxy-cadence-payload", 21)
    rbuf = C.create_string_buffer(21)
    mh = p.reg_mr(scomm, None, 0)
    rreq = p.irecv(rcomm, rbuf, 21, mh, tag=5)
    sreq = p.isend(scomm, buf, 21, mh, tag=5)
    assert p.wait(sreq, 30) == 21 and p.wait(rreq, 30) == 21
    assert rbuf.raw[:21] == b"proxy-cadence-payload"


def _original_bytes_recopied(env, q):
    """RCCL re-copies the ORIGINAL handle bytes (stage=0) over a handle
    that already held in-progress state: the first attempt's task must be
    orphaned safely (reaped later), and connect must still complete."""
    for k, v in env.items():
        os.environ[k] = v
    from baguanet.plugin import Plugin

    p = Plugin(abi=8)
    handle, lcomm = p.listen(0)
    original = bytes(handle)

    def mutate(h, attempt):
        if attempt == 1:  # after the first retry stashed its state
            C.memmove(h, original, len(original))
        return h

    scomm, rcomm = _pump(p, handle, lcomm, mutate=mutate)
    _xfer_ok(p, scomm, rcomm)
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


def test_original_handle_bytes_recopied_mid_retry():
    assert _run_sub(_original_bytes_recopied,
                    {"NCCL_SOCKET_IFNAME": "lo"}) == "ok"


def _handle_moved_between_buffers(env, q):
    """The proxy may memcpy the handle to a different buffer between
    retries; the stashed state must follow (pointer value travels, the
    registry validates it)."""
    for k, v in env.items():
        os.environ[k] = v
    from baguanet.plugin import Plugin

    p = Plugin(abi=8)
    handle, lcomm = p.listen(0)

    def mutate(h, attempt):
        fresh = (C.c_char * len(bytes(h)))()
        C.memmove(fresh, h, len(bytes(h)))
        return fresh

    scomm, rcomm = _pump(p, handle, lcomm, mutate=mutate)
    _xfer_ok(p, scomm, rcomm)
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


def test_handle_moved_between_buffers():
    assert _run_sub(_handle_moved_between_buffers,
                    {"NCCL_SOCKET_IFNAME": "lo"}) == "ok"


def _garbage_stage_pointer(env, q):
    """A corrupted stage field (dangling pointer + random token) must be
    rejected by the registry and treated as a fresh connect — previously a
    blind dereference."""
    for k, v in env.items():
        os.environ[k] = v
    import struct

    from baguanet.plugin import Plugin

    p = Plugin(abi=8)
    handle, lcomm = p.listen(0)
    # ListenHandle layout: magic u32, family u16, port u16, addr[16],
    # stage u64, stage_token u64  -> stage at offset 24
    struct.pack_into("<QQ", handle, 24, 0xDEAD0000BEEF, 0x1234567887654321)
    scomm, rcomm = _pump(p, handle, lcomm)
    _xfer_ok(p, scomm, rcomm)
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


def test_garbage_stage_pointer_rejected():
    assert _run_sub(_garbage_stage_pointer,
                    {"NCCL_SOCKET_IFNAME": "lo"}) == "ok"


def _abandoned_connect_reaped(env, q):
    """A connect RCCL never retries again must not leak its sockets: with
    BNET_CONNECT_ABANDON_MS=200, a later connect() reaps it (fd count
    drops back)."""
    for k, v in env.items():
        os.environ[k] = v
    import time

    from baguanet.plugin import Plugin

    def nfds():
        return len(os.listdir("/proc/self/fd"))

    import socket
    import struct

    p = Plugin(abi=8)
    # warm-up cycle so the engine's lazily-created IO threads (epoll +
    # eventfd per thread) are in the baseline fd count
    wh, wl = p.listen(0)
    ws, wr = _pump(p, wh, wl)
    p.close_send(ws)
    p.close_recv(wr)
    p.close_listen(wl)
    time.sleep(0.1)

    # A loopback connect completes instantly, so to model RCCL abandoning
    # an IN-PROGRESS establishment the listener's accept queue must be
    # full (BNET_BACKLOG=1 + prefilled raw connections): further SYNs are
    # dropped and the plugin's connect() stays pending.
    handle, lcomm = p.listen(0)
    magic, family, port_be = struct.unpack_from("<IHH", bytes(handle))
    port = socket.ntohs(port_be)
    prefill = []
    for _ in range(3):
        rs = socket.socket()
        rs.setblocking(False)
        try:
            rs.connect(("127.0.0.1", port))
        except BlockingIOError:
            pass
        prefill.append(rs)
    time.sleep(0.2)

    base = nfds()
    assert p.connect(0, handle) is None  # stays in progress (queue full)
    assert nfds() > base + 1
    time.sleep(0.4)  # > abandon window
    # an unrelated connect on a fresh handle triggers the reap
    handle2, lcomm2 = p.listen(0)
    scomm, rcomm = _pump(p, handle2, lcomm2, timeout=60)
    _xfer_ok(p, scomm, rcomm)
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm2)
    p.close_listen(lcomm)
    time.sleep(0.1)
    # the abandoned task's sockets must be gone
    got = nfds()
    for rs in prefill:
        rs.close()
    assert got <= base + 1, f"leaked fds: {got} vs base {base}"
    q.put("ok")


def test_abandoned_connect_reaped():
    assert _run_sub(_abandoned_connect_reaped, {
        "NCCL_SOCKET_IFNAME": "lo",
        "BNET_CONNECT_ABANDON_MS": "200",
        "BNET_BACKLOG": "1",
    }) == "ok"
