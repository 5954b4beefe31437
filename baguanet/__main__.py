"""CLI: `python -m baguanet` — build status, devices, config, env help."""

import argparse
import ctypes as C
import json
import sys


def main():
    ap = argparse.ArgumentParser(prog="baguanet")
    ap.add_argument("cmd", nargs="?", default="info",
                    choices=["info", "build", "env"])
    args = ap.parse_args()

    if args.cmd == "build":
        from .build import build_plugin

        print(build_plugin(quiet=False))
        from .ops import build_extension

        build_extension(verbose=True)
        print("torch extension built")
        return

    if args.cmd == "env":
        from .plugin import rccl_env

        for k, v in rccl_env(env={}).items():
            print(f"export {k}={v}")
        return

    # info
    from . import PLUGIN_PATH, __version__
    from .build import build_plugin
    from .plugin import Plugin

    print(f"baguanet {__version__}")
    build_plugin()
    p = Plugin()
    print(f"plugin: {PLUGIN_PATH} ({p.name})")
    buf = C.create_string_buffer(1024)
    p.lib.bnet_config_json(buf, 1024)
    print("config:", buf.value.decode())
    for i in range(p.ndev()):
        props = p.properties(i)
        print(f"dev[{i}]: {props['name']} speed={props['speed']}Mbps "
              f"ptrSupport=0x{props['ptrSupport']:x} pci={props['pciPath']}")
    try:
        import torch

        print(f"torch {torch.__version__}, cuda={torch.cuda.is_available()}")
    except Exception as e:  # pragma: no cover
        print(f"torch unavailable: {e}")


if __name__ == "__main__":
    sys.exit(main())
