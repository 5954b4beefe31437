"""Fused SGD kernel numerics vs torch.optim.SGD on the GPU."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("momentum,wd,nesterov", [
    (0.0, 0.0, False),
    (0.9, 0.0, False),
    (0.9, 1e-4, False),
    (0.9, 1e-4, True),
])
def test_fused_kernel_matches_torch(momentum, wd, nesterov):
    from baguanet.optim import FusedSGD

    torch.manual_seed(1)
    shapes = [(1000,), (64, 128), (3, 3, 32, 32), (7,), (1 << 20,)]
    p1 = [torch.randn(s, device="cuda") for s in shapes]
    p2 = [t.clone() for t in p1]
    g = [torch.randn(s, device="cuda") for s in shapes]
    params1 = [torch.nn.Parameter(t) for t in p1]
    params2 = [torch.nn.Parameter(t) for t in p2]
    o1 = FusedSGD(params1, lr=0.01, momentum=momentum, weight_decay=wd,
                  nesterov=nesterov)
    o2 = torch.optim.SGD(params2, lr=0.01, momentum=momentum,
                         weight_decay=wd, nesterov=nesterov)
    for step in range(4):
        for pp, oo in ((params1, o1), (params2, o2)):
            for t, gr in zip(pp, g):
                t.grad = (gr * (step + 1)).clone()
            oo.step()
    torch.cuda.synchronize()
    for a, b in zip(params1, params2):
        torch.testing.assert_close(a, b, rtol=1e-6, atol=1e-6)


def test_fused_sgd_speed_smoke():
    """One fused launch updates a VGG16-sized parameter set."""
    from baguanet import ops

    torch.manual_seed(2)
    sizes = [1 << 20] * 100  # ~400 MB of fp32 params
    ps = [torch.randn(n, device="cuda") for n in sizes]
    gs = [torch.randn(n, device="cuda") for n in sizes]
    ms = [torch.zeros(n, device="cuda") for n in sizes]
    ops.fused_sgd(ps, gs, ms, 0.01, 0.9, 0.0, False)
    torch.cuda.synchronize()
    # first step: m == g, p == p0 - lr*g
    torch.testing.assert_close(ms[0], gs[0])
