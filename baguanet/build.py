"""Build helpers: compile the native plugin and the torch HIP extension."""

from __future__ import annotations

import os
import subprocess
from pathlib import Path

from . import PLUGIN_PATH, REPO_ROOT


def build_plugin(force: bool = False, quiet: bool = True) -> Path:
    """Compile csrc/ into build/lib/libnccl-net-bagua.so (gfx950)."""
    if PLUGIN_PATH.exists() and not force:
        # rebuild only if sources are newer
        newest = max(
            p.stat().st_mtime
            for p in (REPO_ROOT / "csrc").rglob("*")
            if p.suffix in (".cc", ".h", ".hip") or p.name == "Makefile"
        )
        if newest <= PLUGIN_PATH.stat().st_mtime:
            return PLUGIN_PATH
    cmd = ["make", "-j", str(os.cpu_count() or 8)]
    res = subprocess.run(
        cmd,
        cwd=REPO_ROOT / "csrc",
        capture_output=quiet,
        text=True,
    )
    if res.returncode != 0:
        raise RuntimeError(
            f"plugin build failed:\n{res.stdout or ''}\n{res.stderr or ''}"
        )
    if not PLUGIN_PATH.exists():
        raise RuntimeError(f"build produced no {PLUGIN_PATH}")
    return PLUGIN_PATH
