"""RCCL integration: torch.distributed (nccl==RCCL) world=1 on the GPU with
the baguanet plugin on NCCL_NET_PLUGIN — verifies RCCL dlopens the plugin,
accepts the ncclNetPlugin_v6 ABI, and collectives still work."""

import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

SCRIPT = r"""
import os, torch, torch.distributed as dist
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29611")
os.environ.setdefault("RANK", "0")
os.environ.setdefault("WORLD_SIZE", "1")
dist.init_process_group("nccl")
t = torch.ones(1024, device="cuda") * (dist.get_rank() + 1)
dist.all_reduce(t)
torch.cuda.synchronize()
assert torch.equal(t, torch.ones(1024, device="cuda"))
dist.destroy_process_group()
print("RCCL_OK")
"""


def test_rccl_loads_plugin(tmp_path):
    sys.path.insert(0, os.path.dirname(os.path.dirname(__file__)))
    from baguanet.plugin import rccl_env

    env = rccl_env()
    env["NCCL_DEBUG"] = "INFO"
    env["NCCL_DEBUG_SUBSYS"] = "INIT,NET"
    res = subprocess.run(
        [sys.executable, "-c", SCRIPT],
        env=env,
        capture_output=True,
        text=True,
        timeout=300,
    )
    out = res.stdout + res.stderr
    assert res.returncode == 0, f"RCCL run failed:\n{out[-4000:]}"
    assert "RCCL_OK" in out
    # RCCL must have loaded our plugin (v6) — not fallen back silently
    assert "Loaded net plugin" in out or "BaguaNetAMD" in out, (
        "plugin not loaded by RCCL:\n" + out[-4000:]
    )
