"""Loopback coverage of every exported plugin ABI version (v6, v7, v8).

RCCL probes ncclNetPlugin_v10..v6 newest-first and uses the first symbol it
finds; the reference shipped floor+current (v3+v4, cc/v3/nccl_net_v3.cc:210,
cc/v4/nccl_net_v4.cc:210).  We export v6+v7+v8 from the main .so and a
v6-only escape-hatch .so — each vtable must carry real traffic, not just
dlsym.
"""

import ctypes as C
import multiprocessing as mp
import os

import pytest


def _abi_roundtrip(env, abi, q):
    for k, v in env.items():
        os.environ[k] = v
    import time

    from baguanet.plugin import Plugin

    p = Plugin(abi=abi)
    assert p.name == "BaguaNetAMD"
    props = p.properties(0)
    assert props["name"] == "lo"
    if abi >= 7:
        assert props["netDeviceType"] == 0  # NCCL_NET_DEVICE_HOST
        assert props["netDeviceVersion"] == 0
    if abi >= 8:
        assert props["regIsGlobal"] == 0

    handle, lcomm = p.listen(0)
    scomm = rcomm = None
    t0 = time.monotonic()
    while scomm is None or rcomm is None:
        assert time.monotonic() - t0 < 30
        if scomm is None:
            scomm = p.connect(0, handle)  # v7/v8 path checks devComm==NULL
        if rcomm is None:
            rcomm = p.accept(lcomm)

    for size in (0, 1, 17, 8192, 1 << 20):
        payload = bytes((i * 31 + size) & 0xFF for i in range(size))
        sbuf = C.create_string_buffer(payload, max(size, 1))
        rbuf = C.create_string_buffer(max(size, 1))
        mh = p.reg_mr(scomm, sbuf, size)
        rreq = p.irecv(rcomm, rbuf, size, mh, tag=abi)
        sreq = p.isend(scomm, sbuf, size, mh, tag=abi)
        assert p.wait(sreq, 30) == size
        assert p.wait(rreq, 30) == size
        assert rbuf.raw[:size] == payload

    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


@pytest.mark.parametrize("abi", [6, 7, 8])
def test_abi_version_roundtrip(abi):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(
        target=_abi_roundtrip,
        args=({"NCCL_SOCKET_IFNAME": "lo", "BNET_MIN_CHUNKSIZE": "8192"},
              abi, q),
    )
    proc.start()
    assert q.get(timeout=120) == "ok"
    proc.join(30)
    assert proc.exitcode == 0


def test_exported_symbols():
    """Main .so exports v6+v7+v8; the escape-hatch .so exports v6 only."""
    from baguanet import PLUGIN_DIR

    main = C.CDLL(str(PLUGIN_DIR / "libnccl-net-bagua.so"))
    for v in (6, 7, 8):
        assert C.c_void_p.in_dll(main, f"ncclNetPlugin_v{v}")
    v6only = C.CDLL(str(PLUGIN_DIR / "libnccl-net-bagua6.so"))
    assert C.c_void_p.in_dll(v6only, "ncclNetPlugin_v6")
    with pytest.raises(ValueError):
        C.c_void_p.in_dll(v6only, "ncclNetPlugin_v8")
