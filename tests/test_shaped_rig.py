"""Shaped-link rig regression: plugin traffic through a veth pair with the
receiver in its own network namespace, TBF rate ceiling (netns_rig.py).

Skipped where CAP_NET_ADMIN is unavailable (e.g. unprivileged CI).
"""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "benchmarks"))


def _has_net_admin() -> bool:
    try:
        caps = [l for l in open("/proc/self/status")
                if l.startswith("CapEff")][0].split()[1]
        return bool(int(caps, 16) & (1 << 12))
    except (OSError, IndexError, ValueError):
        return False


pytestmark = pytest.mark.skipif(not _has_net_admin(),
                                reason="needs CAP_NET_ADMIN")


def test_shaped_p2p_saturates_and_respects_ceiling():
    """Data through the shaped rig is correct (p2p_perf verifies protocol
    completion) and throughput lands at the 10 Gb TBF line rate:
    >= 80% of it (saturation) and <= 120% (the shaper actually binds —
    i.e. traffic really crossed the veths, not loopback)."""
    res = subprocess.run(
        [sys.executable, os.path.join(REPO, "benchmarks", "p2p_perf.py"),
         "--json", "--shaped", "10", "--sizes", "4194304",
         "--bytes-per-size", str(256 << 20)],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert res.returncode == 0, res.stderr[-2000:]
    line = [l for l in res.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["qdisc"] == "tbf"
    bw = out["results"][0]["GBps"]
    line_rate = 10 / 8  # GB/s
    assert bw >= 0.8 * line_rate, f"only {bw} GB/s on a 10 Gb shape"
    assert bw <= 1.2 * line_rate, f"{bw} GB/s exceeds the 10 Gb ceiling"


def test_shaped_ring_allreduce_two_nodes():
    """The 2-'node' ring all-reduce analogue of BASELINE config 2: result
    verified and busbw bounded by the 5 Gb shape."""
    res = subprocess.run(
        [sys.executable,
         os.path.join(REPO, "benchmarks", "ring_allreduce.py"),
         "--ranks", "2", "--shaped", "5", "--sizes", "4194304",
         "--iters", "2", "--warmup", "1", "--json"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
        env={**os.environ, "BNET_IO_THREADS": "2"},
    )
    assert res.returncode == 0, res.stderr[-2000:]
    out = json.loads([l for l in res.stdout.splitlines()
                      if l.startswith("{")][-1])
    assert out["verified"]
    bw = out["results"][0]["busbw_GBps"]
    assert 0 < bw <= 5 / 8 * 1.2, bw  # under the shaped ceiling
