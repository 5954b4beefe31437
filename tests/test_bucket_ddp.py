"""BucketedDDP numerics: 2-rank gloo gradients must equal single-process
gradients on the combined batch (DDP averaging semantics)."""

import multiprocessing as mp
import os
import socket

import pytest
import torch
from torch import nn


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(
        nn.Linear(32, 64),
        nn.ReLU(),
        nn.Linear(64, 64),
        nn.ReLU(),
        nn.Linear(64, 10),
    )


def _data(seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(8, 32, generator=g)
    y = torch.randint(0, 10, (8,), generator=g)
    return x, y


def _worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    from baguanet.parallel import BucketedDDP

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        model = BucketedDDP(_model(seed=7), bucket_cap_mb=0.01)
        x, y = _data(100 + rank)
        for step in range(3):
            model.zero_grad()
            loss = nn.functional.cross_entropy(model(x), y)
            loss.backward()
            model.finish_backward()
            with torch.no_grad():
                for p in model.module.parameters():
                    p -= 0.1 * p.grad
        # pickle by value (numpy), not torch shared memory: the child may
        # exit before the parent drains the queue
        grads = [p.grad.numpy().copy() for p in model.module.parameters()]
        params = [
            p.detach().numpy().copy() for p in model.module.parameters()
        ]
        if rank == 0:
            q.put(("ok", grads, params))
    finally:
        dist.destroy_process_group()


def _reference(world=2):
    """Single-process equivalent: average of the ranks' gradients."""
    model = _model(seed=7)
    for step in range(3):
        for p in model.parameters():
            p.grad = None
        gs = []
        for rank in range(world):
            x, y = _data(100 + rank)
            loss = nn.functional.cross_entropy(model(x), y)
            g = torch.autograd.grad(loss, list(model.parameters()))
            gs.append(g)
        avg = [sum(t) / world for t in zip(*gs)]
        with torch.no_grad():
            for p, g in zip(model.parameters(), avg):
                p -= 0.1 * g
    return avg, [p.detach().clone() for p in model.parameters()]


@pytest.mark.parametrize("world", [2, 4])
def test_bucket_ddp_matches_reference(world):
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_worker, args=(r, world, port, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    tag, grads, params = q.get(timeout=180)
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    assert tag == "ok"
    ref_grads, ref_params = _reference(world)
    for g, rg in zip(grads, ref_grads):
        torch.testing.assert_close(torch.from_numpy(g), rg, rtol=1e-5,
                                   atol=1e-6)
    for p, rp in zip(params, ref_params):
        torch.testing.assert_close(torch.from_numpy(p), rp, rtol=1e-5,
                                   atol=1e-6)


def test_no_sync_and_world1():
    # world_size-less usage: BucketedDDP without init still works (world=1)
    from baguanet.parallel import BucketedDDP

    model = BucketedDDP(_model(seed=3), broadcast_params=False)
    x, y = _data(5)
    model.zero_grad()
    loss = nn.functional.cross_entropy(model(x), y)
    loss.backward()
    model.finish_backward()
    ref = _model(seed=3)
    loss2 = nn.functional.cross_entropy(ref(x), y)
    loss2.backward()
    for p, rp in zip(model.module.parameters(), ref.parameters()):
        torch.testing.assert_close(p.grad, rp.grad, rtol=1e-6, atol=1e-7)
    # grad accumulation under no_sync accumulates into the same views
    with model.no_sync():
        loss3 = nn.functional.cross_entropy(model(x), y)
        loss3.backward()
    for p, rp in zip(model.module.parameters(), ref.parameters()):
        torch.testing.assert_close(p.grad, 2 * rp.grad, rtol=1e-5, atol=1e-6)


def test_mixed_dtype_buckets():
    """Buckets must split on dtype boundaries (flat buffer is one dtype)."""
    import torch
    from baguanet.parallel import BucketedDDP

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.a = torch.nn.Linear(8, 8)
            self.b = torch.nn.Linear(8, 8).to(torch.bfloat16)
            self.c = torch.nn.Linear(8, 8)

    m = BucketedDDP(M(), bucket_cap_mb=100, broadcast_params=False)
    for bkt in m.buckets:
        dts = {p.dtype for p in bkt.params}
        assert len(dts) == 1
        assert bkt.flat.dtype == next(iter(dts))


def test_unused_parameter_raises():
    import pytest

    from baguanet.parallel import BucketedDDP

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.used = torch.nn.Linear(4, 4)
            self.unused = torch.nn.Linear(4, 4)

        def forward(self, x):
            return self.used(x)

    m = BucketedDDP(M(), bucket_cap_mb=100, broadcast_params=False)
    m.zero_grad()
    m(torch.randn(2, 4)).sum().backward()
    with pytest.raises(RuntimeError, match="received no gradient"):
        m.finish_backward()


def test_bucket_ddp_world4_gloo(tmp_path):
    """World-4 data-parallel numerics (the driver's 8-GPU shape, scaled
    down): BucketedDDP gradients equal the single-process reference."""
    import subprocess
    import sys

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

    code = r'''
import os, torch, torch.distributed as dist, sys
sys.path.insert(0, os.environ["BNET_REPO"])
from baguanet.models import resnet50
from baguanet.parallel import BucketedDDP

dist.init_process_group("gloo")
rank, world = dist.get_rank(), dist.get_world_size()
torch.manual_seed(1234)
model = resnet50(num_classes=10)
ref = resnet50(num_classes=10)
ref.load_state_dict(model.state_dict())
ddp = BucketedDDP(model, bucket_cap_mb=5.0)

torch.manual_seed(99)
xs = [torch.randn(2, 3, 64, 64) for _ in range(world)]
ys = [torch.randint(0, 10, (2,)) for _ in range(world)]

ddp.zero_grad()
loss = torch.nn.functional.cross_entropy(ddp(xs[rank]), ys[rank])
loss.backward()
ddp.finish_backward()

# single-process reference: mean gradient over all shards
ref.zero_grad()
for x, y in zip(xs, ys):
    torch.nn.functional.cross_entropy(ref(x), y).div(world).backward()

for (n, p), (_, rp) in zip(ddp.module.named_parameters(),
                           ref.named_parameters()):
    assert torch.allclose(p.grad, rp.grad, atol=2e-5), n
if rank == 0:
    print("WORLD4_OK")
'''
    script = tmp_path / "world4_ddp.py"
    script.write_text(code)
    env = dict(os.environ)
    env["BNET_REPO"] = REPO
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", "29571", str(script)],
        capture_output=True, text=True, timeout=600, env=env,
    )
    assert res.returncode == 0, res.stderr[-3000:]
    assert "WORLD4_OK" in res.stdout
