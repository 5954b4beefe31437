"""Telemetry: counters and spans must reflect actual transfers (run in a
subprocess so env + atexit dumps are isolated)."""

import json
import multiprocessing as mp
import os


def _run(tmpdir, q):
    os.environ["NCCL_SOCKET_IFNAME"] = "lo"
    os.environ["BNET_TRACE_FILE"] = os.path.join(tmpdir, "trace.json")
    os.environ["BNET_METRICS_FILE"] = os.path.join(tmpdir, "metrics.prom")
    import ctypes as C

    from baguanet.plugin import Plugin

    import sys
    sys.path.insert(0, os.path.join(os.path.dirname(__file__)))
    from test_plugin_loopback import establish, xfer

    p = Plugin()
    lcomm, scomm, rcomm = establish(p)
    for size in (100, 5000, 200000):
        payload = os.urandom(size)
        assert xfer(p, scomm, rcomm, payload) == payload
    # on-demand dump through the exported C API
    mfile = os.path.join(tmpdir, "ondemand.prom")
    p.lib.bnet_dump_metrics(mfile.encode())
    p.lib.bnet_dump_trace(
        os.path.join(tmpdir, "ondemand_trace.json").encode()
    )
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


def test_metrics_and_spans(tmp_path):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(target=_run, args=(str(tmp_path), q))
    proc.start()
    assert q.get(timeout=120) == "ok"
    proc.join(30)
    assert proc.exitcode == 0

    metrics = (tmp_path / "ondemand.prom").read_text()
    m = {}
    for line in metrics.splitlines():
        if line.startswith("#") or not line.strip():
            continue
        k, v = line.rsplit(" ", 1)
        m[k] = int(v)
    assert m["bnet_isend_total"] == 3
    assert m["bnet_irecv_total"] == 3
    assert m["bnet_bytes_sent_total"] == 100 + 5000 + 200000
    assert m["bnet_bytes_recv_total"] == 100 + 5000 + 200000
    assert m["bnet_send_comms_total"] == 1
    assert m['bnet_isend_nbytes_bucket{le="+Inf"}'] == 3

    spans = json.loads((tmp_path / "ondemand_trace.json").read_text())
    isends = [s for s in spans if s["name"].startswith("isend")]
    irecvs = [s for s in spans if s["name"].startswith("irecv")]
    assert len(isends) == 3 and len(irecvs) == 3
    assert all(s["dur"] > 0 for s in spans)

    # atexit dumps also fired
    assert (tmp_path / "metrics.prom").exists()
    assert (tmp_path / "trace.json").exists()


def test_debug_exports_smoke(plugin):
    """bnet_force_kick / bnet_dump_send_state / bnet_dump_recv_state are
    the stall-triage API (used by the soak's dump): they must be callable
    on a live comm without crashing and produce non-empty state."""
    import ctypes as C

    from tests.test_plugin_loopback import establish

    lcomm, scomm, rcomm = establish(plugin)
    buf = C.create_string_buffer(8192)
    n = plugin.lib.bnet_dump_send_state(scomm, buf, 8192)
    assert n > 0 and b"oldest" in buf.value
    n = plugin.lib.bnet_dump_recv_state(rcomm, buf, 8192)
    assert n > 0
    plugin.lib.bnet_force_kick(scomm)  # send-comm only (see telemetry.cc)
    # comm still functional after forced kicks
    mh = plugin.reg_mr(scomm, None, 0)
    sbuf = C.create_string_buffer(b"kicked", 6)
    rbuf = C.create_string_buffer(7)
    rreq = plugin.irecv(rcomm, rbuf, 6, mh)
    sreq = plugin.isend(scomm, sbuf, 6, mh)
    assert plugin.wait(sreq, 30) == 6 and plugin.wait(rreq, 30) == 6
    assert rbuf.raw[:6] == b"kicked"
    plugin.close_send(scomm)
    plugin.close_recv(rcomm)
    plugin.close_listen(lcomm)


def test_atexit_file_dumps(tmp_path):
    """BNET_METRICS_FILE / BNET_TRACE_FILE must be written at process
    exit (the atexit hook) without any explicit dump call."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    mfile = tmp_path / "exit_metrics.prom"
    tfile = tmp_path / "exit_trace.json"
    code = (
        "import os, sys, ctypes as C;"
        f"sys.path.insert(0, {repo!r});"
        f"sys.path.insert(0, {os.path.join(repo, 'tests')!r});"
        "from baguanet.plugin import Plugin;"
        "from test_plugin_loopback import establish, xfer;"
        "p = Plugin(); l, s, r = establish(p);"
        "assert xfer(p, s, r, b'x' * 1234) == b'x' * 1234;"
        "p.close_send(s); p.close_recv(r); p.close_listen(l)"
    )
    env = dict(os.environ, NCCL_SOCKET_IFNAME="lo",
               BNET_METRICS_FILE=str(mfile), BNET_TRACE_FILE=str(tfile))
    res = subprocess.run([sys.executable, "-c", code], env=env,
                         capture_output=True, text=True, timeout=120)
    assert res.returncode == 0, res.stderr[-1500:]
    assert "bnet_bytes_sent_total 1234" in mfile.read_text()
    spans = json.loads(tfile.read_text())
    events = spans["traceEvents"] if isinstance(spans, dict) else spans
    assert any(str(ev.get("name", "")).startswith("isend") for ev in events)


def _live_endpoint(env, q):
    for k, v in env.items():
        os.environ[k] = v
    import ctypes as C
    import time
    import urllib.request

    from baguanet.plugin import Plugin

    p = Plugin()
    # one quick loopback message so counters are non-zero
    handle, lcomm = p.listen(0)
    scomm = rcomm = None
    t0 = time.monotonic()
    while scomm is None or rcomm is None:
        assert time.monotonic() - t0 < 30
        scomm = scomm or p.connect(0, handle)
        rcomm = rcomm or p.accept(lcomm)
    buf = C.create_string_buffer(b"live-metrics", 12)
    rbuf = C.create_string_buffer(12)
    mh = p.reg_mr(scomm, None, 0)
    rreq = p.irecv(rcomm, rbuf, 12, mh)
    sreq = p.isend(scomm, buf, 12, mh)
    p.wait(sreq, 30)
    p.wait(rreq, 30)
    port = int(env["BNET_METRICS_PORT"]) + int(env.get("RANK", 0))
    text = None
    for _ in range(50):
        try:
            text = urllib.request.urlopen(
                f"http://127.0.0.1:{port}/metrics", timeout=2
            ).read().decode()
            break
        except OSError:
            time.sleep(0.1)
    assert text is not None, "metrics endpoint never came up"
    assert 'bnet_isend_total{rank="2"} 1' in text, text[:500]
    # scrape again: counters are live, not a one-shot snapshot
    sreq = p.isend(scomm, buf, 12, mh)
    rreq = p.irecv(rcomm, rbuf, 12, mh)
    p.wait(sreq, 30)
    p.wait(rreq, 30)
    text2 = urllib.request.urlopen(
        f"http://127.0.0.1:{port}/metrics", timeout=2).read().decode()
    assert 'bnet_isend_total{rank="2"} 2' in text2, text2[:500]
    p.close_send(scomm)
    p.close_recv(rcomm)
    p.close_listen(lcomm)
    q.put("ok")


def test_live_metrics_endpoint():
    """BNET_METRICS_PORT serves live Prometheus text on 127.0.0.1:(port+rank)
    — the pull-style equivalent of the reference's push-gateway uploader."""
    import multiprocessing as mp
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    proc = ctx.Process(target=_live_endpoint, args=({
        "NCCL_SOCKET_IFNAME": "lo",
        "BNET_METRICS_PORT": str(port - 2),
        "RANK": "2",
    }, q))
    proc.start()
    assert q.get(timeout=120) == "ok"
    proc.join(30)
    assert proc.exitcode == 0
