"""RCCL integration: torch.distributed (nccl==RCCL) on the GPU with the
baguanet plugin on NCCL_NET_PLUGIN.

With world_size=1 RCCL performs no transport setup, so the net plugin is
not dlopened (lazy net init) — what this test CAN verify on a 1-GPU box:
  * collectives work with the plugin env set (LD_LIBRARY_PATH, NCCL_NET_PLUGIN)
  * RCCL did NOT report a plugin load/ABI failure
  * if RCCL did probe the plugin, our ncclNetPlugin_v6 was found
The real multi-rank net path is covered by the plugin-API loopback tests
(same vtable RCCL calls) and by the driver's multi-GPU bench runs.
"""

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


def test_rccl_with_plugin_env(tmp_path):
    sys.path.insert(0, os.path.dirname(os.path.dirname(__file__)))
    from baguanet.plugin import rccl_env

    env = rccl_env()
    env["NCCL_DEBUG"] = "INFO"
    env["NCCL_DEBUG_SUBSYS"] = "INIT,NET,ENV"
    res = subprocess.run(
        [sys.executable, "-c", SCRIPT],
        env=env,
        capture_output=True,
        text=True,
        timeout=300,
    )
    out = res.stdout + res.stderr
    # keep the full log for offline inspection (merged back by gpurun)
    logdir = os.environ.get("BNET_TEST_LOG_DIR")
    if logdir:
        os.makedirs(logdir, exist_ok=True)
        with open(os.path.join(logdir, "rccl_world1_debug.log"), "w") as f:
            f.write(out)
    assert res.returncode == 0, f"RCCL run failed:\n{out[-4000:]}"
    assert "RCCL_OK" in out
    # none of our exported symbols may be reported missing
    for v in (6, 7, 8):
        assert f"Failed to find ncclNetPlugin_v{v}" not in out
    # RCCL probes v10..v6 newest-first and must pick our v8 vtable (net
    # init runs even at world=1: topology probing enumerates net devices)
    loaded = [ln for ln in out.splitlines() if "Loaded net plugin" in ln]
    assert loaded, "RCCL did not load the plugin:\n" + out[-3000:]
    assert any("(v8)" in ln for ln in loaded), loaded
    # and select it — garbage v8 properties (wrong struct layout) would
    # break network selection
    assert any("Using network BaguaNetAMD" in ln
               for ln in out.splitlines()), out[-3000:]
