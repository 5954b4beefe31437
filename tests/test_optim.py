"""FusedSGD vs torch.optim.SGD numerics (eager path on CPU; the fused
kernel path is covered in tests/test_gpu_fused_sgd.py)."""

import torch

from baguanet.optim import FusedSGD


def _compare(momentum, weight_decay, nesterov, steps=5):
    torch.manual_seed(0)
    m1 = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.Tanh(),
                             torch.nn.Linear(32, 4))
    m2 = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.Tanh(),
                             torch.nn.Linear(32, 4))
    m2.load_state_dict(m1.state_dict())
    o1 = FusedSGD(m1.parameters(), lr=0.05, momentum=momentum,
                  weight_decay=weight_decay, nesterov=nesterov)
    o2 = torch.optim.SGD(m2.parameters(), lr=0.05, momentum=momentum,
                         weight_decay=weight_decay, nesterov=nesterov)
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)
    for _ in range(steps):
        for m, o in ((m1, o1), (m2, o2)):
            o.zero_grad()
            torch.nn.functional.mse_loss(m(x), y).backward()
            o.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        torch.testing.assert_close(p1, p2, rtol=1e-6, atol=1e-7)


def test_plain_sgd():
    _compare(0.0, 0.0, False)


def test_momentum():
    _compare(0.9, 0.0, False)


def test_momentum_wd():
    _compare(0.9, 1e-4, False)


def test_nesterov():
    _compare(0.9, 1e-4, True)
