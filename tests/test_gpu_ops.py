"""GPU numerics tests for the HIP kernels (vs plain torch fp32 reference)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ops():
    from baguanet import ops as O

    O.build_extension()
    return O


@pytest.mark.parametrize("numel", [1, 63, 4096, 1 << 20, (1 << 22) + 5])
def test_copy_bytes_d2d(ops, numel):
    src = torch.randn(numel, device="cuda")
    dst = torch.empty_like(src)
    ops.copy_bytes(dst, src)
    torch.cuda.synchronize()
    assert torch.equal(dst, src)


def test_copy_bytes_pinned_roundtrip(ops):
    n = 1 << 20
    src = torch.randn(n, device="cuda")
    pinned = torch.empty(n, pin_memory=True)
    ops.copy_bytes(pinned, src)  # D2H via kernel
    torch.cuda.synchronize()
    assert torch.equal(pinned, src.cpu())
    back = torch.empty(n, device="cuda")
    ops.copy_bytes(back, pinned)  # H2D via kernel
    torch.cuda.synchronize()
    assert torch.equal(back, src)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16,
                                   torch.float16])
def test_multi_pack_matches_cat(ops, dtype):
    torch.manual_seed(0)
    ts = [
        torch.randn(s, device="cuda").to(dtype)
        for s in [(1000,), (3, 37), (64, 64), (1,), (2, 3, 5, 7)]
    ]
    flat = torch.empty(sum(t.numel() for t in ts), device="cuda", dtype=dtype)
    ops.multi_pack(flat, ts)
    torch.cuda.synchronize()
    ref = torch.cat([t.flatten() for t in ts])
    assert torch.equal(flat, ref)


def test_multi_unpack_roundtrip(ops):
    torch.manual_seed(1)
    shapes = [(4096,), (123,), (256, 512), (7,)]
    ts = [torch.randn(s, device="cuda") for s in shapes]
    flat = torch.empty(sum(t.numel() for t in ts), device="cuda")
    ops.multi_pack(flat, ts)
    outs = [torch.zeros_like(t) for t in ts]
    ops.multi_unpack(flat, outs)
    torch.cuda.synchronize()
    for t, o in zip(ts, outs):
        assert torch.equal(t, o)


def test_multi_pack_large_bucket(ops):
    # ~200 MB bucket, many tensors — fills the chip (grid-stride)
    torch.manual_seed(2)
    ts = [torch.randn(1 << 20, device="cuda") for _ in range(50)]
    flat = torch.empty(sum(t.numel() for t in ts), device="cuda")
    ops.multi_pack(flat, ts)
    torch.cuda.synchronize()
    ref = torch.cat(ts)
    assert torch.equal(flat, ref)
