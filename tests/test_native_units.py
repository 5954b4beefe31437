"""Native (C++) unit tests for transport internals, built + run via make.

`make native-test` runs: the freelist shadow-model test, the 3M-message
claim/slot race stress, and the full-transport in-process loopback soak
under BOTH engines.  `make tsan-test` repeats the claim stress and the
loopback soak under ThreadSanitizer (the race-detection pass the
reference never had — SURVEY.md §5)."""

import subprocess
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_native_suite():
    res = subprocess.run(
        ["make", "native-test"], cwd=REPO / "csrc",
        capture_output=True, text=True, timeout=600,
    )
    assert res.returncode == 0, res.stdout + res.stderr
    assert "freelist tests ok" in res.stdout
    assert "claim stress ok" in res.stdout
    assert res.stdout.count("loopback soak ok") == 2  # epoll + uring


def test_native_tsan():
    res = subprocess.run(
        ["make", "tsan-test"], cwd=REPO / "csrc",
        capture_output=True, text=True, timeout=900,
    )
    assert res.returncode == 0, res.stdout + res.stderr
    assert "claim stress ok" in res.stdout
    assert res.stdout.count("loopback soak ok") == 2
    assert "ThreadSanitizer" not in res.stdout + res.stderr


def test_native_asan():
    res = subprocess.run(
        ["make", "asan-test"], cwd=REPO / "csrc",
        capture_output=True, text=True, timeout=900,
    )
    assert res.returncode == 0, res.stdout + res.stderr
    assert res.stdout.count("loopback soak ok") == 2
    assert "AddressSanitizer" not in res.stdout + res.stderr
    assert "LeakSanitizer" not in res.stdout + res.stderr
