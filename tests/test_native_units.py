"""Native (C++) unit tests for transport internals, built + run via make."""

import subprocess
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_freelist_native():
    res = subprocess.run(
        ["make", "native-test"], cwd=REPO / "csrc",
        capture_output=True, text=True, timeout=300,
    )
    assert res.returncode == 0, res.stdout + res.stderr
    assert "freelist tests ok" in res.stdout
