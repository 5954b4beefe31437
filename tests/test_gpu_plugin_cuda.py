"""Plugin loopback with NCCL_PTR_CUDA: GPU-to-GPU transfers over TCP through
the pinned-ring staging path (D2H copy pipeline -> sockets -> H2D copies).
"""

import ctypes as C
import time

import pytest
import torch

from tests.test_plugin_loopback import establish

pytestmark = pytest.mark.gpu

NCCL_PTR_CUDA = 0x2


@pytest.fixture(scope="module")
def gconn(plugin):
    props = plugin.properties(0)
    if not (props["ptrSupport"] & NCCL_PTR_CUDA):
        pytest.skip("plugin built without CUDA staging on this box")
    lcomm, scomm, rcomm = establish(plugin)
    yield plugin, scomm, rcomm
    plugin.close_send(scomm)
    plugin.close_recv(rcomm)
    plugin.close_listen(lcomm)


def gpu_xfer(plugin, scomm, rcomm, src: torch.Tensor, dst: torch.Tensor):
    # ncclNet contract: buffers are ready when isend/irecv is called
    torch.cuda.synchronize()
    size = src.numel() * src.element_size()
    smh = plugin.reg_mr(scomm, C.c_void_p(src.data_ptr()), size,
                        NCCL_PTR_CUDA)
    rmh = plugin.reg_mr(rcomm, C.c_void_p(dst.data_ptr()), size,
                        NCCL_PTR_CUDA)
    rreq = sreq = None
    t0 = time.monotonic()
    while rreq is None:
        rreq = plugin.irecv(rcomm, C.c_void_p(dst.data_ptr()), size, rmh)
        assert time.monotonic() - t0 < 30
    while sreq is None:
        sreq = plugin.isend(scomm, C.c_void_p(src.data_ptr()), size, smh)
        assert time.monotonic() - t0 < 30
    assert plugin.wait(sreq, 60) == size
    assert plugin.wait(rreq, 60) == size
    # iflush must be a no-op (data already in HBM at completion)
    fl = plugin.iflush(rcomm, C.c_void_p(dst.data_ptr()), size, rmh)
    assert fl is None


@pytest.mark.parametrize(
    "nbytes", [16, 4096, 65536, 1 << 20, (1 << 23) + 52]
)
def test_gpu_roundtrip(gconn, nbytes):
    plugin, scomm, rcomm = gconn
    n = nbytes // 4
    src = torch.randn(n, device="cuda")
    dst = torch.zeros(n, device="cuda")
    gpu_xfer(plugin, scomm, rcomm, src, dst)
    torch.cuda.synchronize()
    assert torch.equal(src, dst)


def test_gpu_many_messages(gconn):
    plugin, scomm, rcomm = gconn
    torch.manual_seed(3)
    msgs = [torch.randn(10000 + 77 * i, device="cuda") for i in range(20)]
    outs = [torch.zeros_like(m) for m in msgs]
    for m, o in zip(msgs, outs):
        gpu_xfer(plugin, scomm, rcomm, m, o)
    torch.cuda.synchronize()
    for i, (m, o) in enumerate(zip(msgs, outs)):
        assert torch.equal(m, o), f"message {i} corrupted"


def test_gpu_mixed_host_and_cuda(gconn):
    """Host-buffer messages interleaved with staged GPU messages."""
    import os

    plugin, scomm, rcomm = gconn
    payload = os.urandom(100000)
    sbuf = C.create_string_buffer(payload, len(payload))
    rbuf = C.create_string_buffer(len(payload) + 1)
    smh = plugin.reg_mr(scomm, sbuf, len(payload))
    rmh = plugin.reg_mr(rcomm, rbuf, len(payload))
    rreq = plugin.irecv(rcomm, rbuf, len(payload), rmh)
    sreq = plugin.isend(scomm, sbuf, len(payload), smh)
    assert rreq is not None and sreq is not None
    plugin.wait(sreq, 60)
    plugin.wait(rreq, 60)
    assert rbuf.raw[: len(payload)] == payload

    src = torch.arange(123456, device="cuda", dtype=torch.float32)
    dst = torch.zeros_like(src)
    gpu_xfer(plugin, scomm, rcomm, src, dst)
    torch.cuda.synchronize()
    assert torch.equal(src, dst)


def test_gpu_grouped_recv_staged(gconn):
    """Grouped irecv (maxRecvs) with NCCL_PTR_CUDA buffers: three staged
    GPU messages land in one grouped request."""
    plugin, scomm, rcomm = gconn
    torch.cuda.synchronize()
    srcs = [torch.randn(5000 + 999 * i, device="cuda") for i in range(3)]
    dsts = [torch.zeros_like(s) for s in srcs]
    sizes = [s.numel() * 4 for s in srcs]
    torch.cuda.synchronize()
    smh = plugin.reg_mr(scomm, C.c_void_p(srcs[0].data_ptr()), sizes[0],
                        NCCL_PTR_CUDA)
    rmh = plugin.reg_mr(rcomm, C.c_void_p(dsts[0].data_ptr()), sizes[0],
                        NCCL_PTR_CUDA)
    req = None
    t0 = time.monotonic()
    while req is None:
        req, n = plugin.irecv_n(
            rcomm, [C.c_void_p(d.data_ptr()) for d in dsts], sizes, rmh,
            tags=[1, 2, 3])
        assert time.monotonic() - t0 < 30
    for s, size, tag in zip(srcs, sizes, (1, 2, 3)):
        sreq = None
        while sreq is None:
            sreq = plugin.isend(scomm, C.c_void_p(s.data_ptr()), size, smh,
                                tag=tag)
            assert time.monotonic() - t0 < 60
        assert plugin.wait(sreq, 60) == size
    while True:
        done, got = plugin.test_n(req, 3)
        if done:
            break
        assert time.monotonic() - t0 < 60
    assert got == sizes
    torch.cuda.synchronize()
    for i, (s, d) in enumerate(zip(srcs, dsts)):
        assert torch.equal(s, d), f"grouped member {i} corrupted"
