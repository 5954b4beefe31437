"""GPU staging with BNET_STAGE_KERNEL=1: the pack kernels (not SDMA) move
bytes between HBM and the pinned ring.  Subprocess-isolated (config env is
read once per process)."""

import multiprocessing as mp
import os

import pytest

pytestmark = pytest.mark.gpu


def _run(q):
    os.environ["NCCL_SOCKET_IFNAME"] = "lo"
    os.environ["BNET_STAGE_KERNEL"] = "1"
    os.environ["BNET_MIN_CHUNKSIZE"] = "8192"
    import ctypes as C
    import sys

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import torch

    from baguanet.plugin import Plugin
    from test_plugin_loopback import establish

    p = Plugin()
    assert p.properties(0)["ptrSupport"] & 0x2
    lcomm, scomm, rcomm = establish(p)
    torch.manual_seed(0)
    for n in (1000, 1 << 18, (1 << 21) + 3):
        src = torch.randn(n, device="cuda")
        dst = torch.zeros_like(src)
        size = n * 4
        smh = p.reg_mr(scomm, C.c_void_p(src.data_ptr()), size, 0x2)
        rmh = p.reg_mr(rcomm, C.c_void_p(dst.data_ptr()), size, 0x2)
        rreq = None
        while rreq is None:
            rreq = p.irecv(rcomm, C.c_void_p(dst.data_ptr()), size, rmh)
        sreq = None
        while sreq is None:
            sreq = p.isend(scomm, C.c_void_p(src.data_ptr()), size, smh)
        assert p.wait(sreq, 60) == size
        assert p.wait(rreq, 60) == size
        torch.cuda.synchronize()
        assert torch.equal(src, dst), f"kernel-staged payload corrupt n={n}"
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


def test_kernel_staging_roundtrip():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(target=_run, args=(q,))
    proc.start()
    assert q.get(timeout=240) == "ok"
    proc.join(30)
    assert proc.exitcode == 0


def _oversize(q):
    os.environ["NCCL_SOCKET_IFNAME"] = "lo"
    os.environ["BNET_STAGE_POOL"] = str(8 << 20)  # 8 MiB pool
    import ctypes as C
    import sys

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import time

    import torch

    from baguanet.plugin import Plugin
    from test_plugin_loopback import establish

    p = Plugin()
    lcomm, scomm, rcomm = establish(p)
    big = torch.randn(16 << 18, device="cuda")  # 16 MiB > pool
    out = torch.zeros_like(big)
    torch.cuda.synchronize()
    size = big.numel() * 4
    smh = p.reg_mr(scomm, C.c_void_p(big.data_ptr()), size, 0x2)
    rmh = p.reg_mr(rcomm, C.c_void_p(out.data_ptr()), size, 0x2)
    rreq = sreq = None
    t0 = time.monotonic()
    while rreq is None:
        rreq = p.irecv(rcomm, C.c_void_p(out.data_ptr()), size, rmh)
        assert time.monotonic() - t0 < 60
    while sreq is None:
        sreq = p.isend(scomm, C.c_void_p(big.data_ptr()), size, smh)
        assert time.monotonic() - t0 < 60
    assert p.wait(sreq, 120) == size
    assert p.wait(rreq, 120) == size
    torch.cuda.synchronize()
    assert torch.equal(big, out)
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


def test_message_larger_than_pool_transfers():
    """A message larger than the staging pool streams through a dedicated
    budget-accounted pinned allocation (it used to hard-fail with
    ncclInternalError)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(target=_oversize, args=(q,))
    proc.start()
    assert q.get(timeout=240) == "ok"
    proc.join(30)
    assert proc.exitcode == 0
