"""NIC discovery semantics (NCCL_SOCKET_IFNAME / NCCL_SOCKET_FAMILY),
each case in a subprocess because the device table is built once per
process at plugin init."""

import multiprocessing as mp
import os


def _probe(env, q):
    for k, v in env.items():
        os.environ[k] = v
    from baguanet.plugin import Plugin

    p = Plugin()
    q.put([p.properties(i)["name"] for i in range(p.ndev())])


def _names(env):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(target=_probe, args=(env, q))
    proc.start()
    out = q.get(timeout=120)
    proc.join(30)
    assert proc.exitcode == 0
    return out


def test_default_excludes_loopback_and_docker():
    names = _names({})
    assert "lo" not in names
    assert not any(n.startswith("docker") for n in names)
    assert names, "no devices discovered with default filter"


def test_explicit_lo():
    assert _names({"NCCL_SOCKET_IFNAME": "lo"}) == ["lo"]


def test_exact_match_spec():
    # "=lo" exact-match also admits loopback
    assert _names({"NCCL_SOCKET_IFNAME": "=lo"}) == ["lo"]


def test_prefix_match():
    names = _names({"NCCL_SOCKET_IFNAME": "eth"})
    assert all(n.startswith("eth") for n in names)


def test_exclude_spec():
    names = _names({"NCCL_SOCKET_IFNAME": "^eth"})
    assert not any(n.startswith("eth") for n in names)


def test_family_filter():
    import socket

    names4 = _names({"NCCL_SOCKET_IFNAME": "lo",
                     "NCCL_SOCKET_FAMILY": str(int(socket.AF_INET))})
    assert names4 == ["lo"]
