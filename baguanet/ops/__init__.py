"""baguanet.ops — HIP (gfx950) kernels exposed to torch.

The extension is compiled IN-TREE (build/torch_ext) so the built .so
travels with the repo snapshot.  On a GPU machine a missing/failed
extension raises — the HIP path must never silently fall back to eager.
"""

from __future__ import annotations

import os
from pathlib import Path

import torch

from .. import REPO_ROOT

_C = None
_ERR: Exception | None = None


def build_extension(verbose: bool = False):
    """Compile (if needed) and load the extension; in-tree build dir."""
    global _C, _ERR
    if _C is not None:
        return _C
    from torch.utils import cpp_extension

    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    build_dir = REPO_ROOT / "build" / "torch_ext"
    build_dir.mkdir(parents=True, exist_ok=True)
    src = REPO_ROOT / "csrc"
    try:
        _C = cpp_extension.load(
            name="baguanet_C",
            sources=[
                str(src / "torch" / "ops.cc"),
                str(src / "hip" / "pack_kernels.hip"),
                str(src / "hip" / "multi_tensor.hip"),
            ],
            build_directory=str(build_dir),
            extra_cflags=["-O3"],
            verbose=verbose,
        )
    except Exception as e:  # pragma: no cover
        _ERR = e
        raise
    return _C


def _load():
    global _C
    if _C is None:
        build_extension()
    return _C


def copy_bytes(dst: torch.Tensor, src: torch.Tensor) -> None:
    _load().copy_bytes(dst, src)


def multi_pack(flat: torch.Tensor, tensors: list[torch.Tensor]) -> None:
    _load().multi_pack(flat, tensors)


def multi_unpack(flat: torch.Tensor, tensors: list[torch.Tensor]) -> None:
    _load().multi_unpack(flat, tensors)


def fused_sgd(params, grads, momenta, lr, momentum=0.0, weight_decay=0.0,
              nesterov=False) -> None:
    _load().fused_sgd(params, grads, momenta, lr, momentum, weight_decay,
                      nesterov)
