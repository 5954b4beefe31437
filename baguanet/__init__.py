"""baguanet — MI355X-native RCCL network plugin + data-parallel training stack.

A from-scratch rebuild of the capabilities of BaguaSys/bagua-net (an NCCL
multi-TCP-stream network transport plugin) for AMD MI355X clusters:

* ``csrc/`` — the native plugin (``libnccl-net-bagua.so``): multi-stream TCP
  transport with dynamic stripe scheduling, epoll IO threads, and a
  HIP-pinned-ring GPU staging path (``NCCL_PTR_CUDA``) for RCCL.
* ``baguanet.plugin`` — loader/env helpers + ctypes binding to the plugin's
  ``ncclNetPlugin_v6`` vtable (used by tests and benchmarks).
* ``baguanet.parallel`` — bucketed data-parallel gradient all-reduce driver
  (the Bagua/DDP layer of the reference's benchmark stack).
* ``baguanet.models`` — VGG16 / ResNet-50 for the reference's benchmark
  workloads.
* ``baguanet.ops`` — HIP staging/pack kernels exposed to torch for tests
  and microbenchmarks.
"""

__version__ = "0.1.0"

from pathlib import Path

REPO_ROOT = Path(__file__).resolve().parent.parent
PLUGIN_DIR = REPO_ROOT / "build" / "lib"
PLUGIN_PATH = PLUGIN_DIR / "libnccl-net-bagua.so"
