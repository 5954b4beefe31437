"""Randomized wire-protocol stress: several comms in both directions,
random message sizes (0..2MB), random posting interleave, verifying every
byte.  Catches striping/slot-reuse/parking races the structured tests
might miss."""

import ctypes as C
import hashlib
import random
import time

from tests.test_plugin_loopback import establish


def test_stress_random_traffic(plugin):
    rng = random.Random(99)
    n_pairs = 3
    msgs_per_pair = 120
    pairs = [establish(plugin) for _ in range(n_pairs)]

    class Dir:
        def __init__(self, scomm, rcomm):
            self.scomm, self.rcomm = scomm, rcomm
            self.smh = plugin.reg_mr(scomm, None, 0)
            self.rmh = plugin.reg_mr(rcomm, None, 0)
            self.sizes = [
                rng.choice([0, 1, 17, 1000, 8192, 65536, 500_000, 2 << 20])
                for _ in range(msgs_per_pair)
            ]
            self.sbufs = {}
            self.rbufs = {}
            self.sreqs = {}
            self.rreqs = {}
            self.si = self.ri = 0
            self.sdone = self.rdone = 0

        def step(self):
            if self.ri < msgs_per_pair and len(self.rreqs) < 24:
                sz = self.sizes[self.ri]
                buf = C.create_string_buffer(sz + 1)
                r = plugin.irecv(self.rcomm, buf, sz, self.rmh)
                if r is not None:
                    self.rbufs[self.ri] = buf
                    self.rreqs[self.ri] = r
                    self.ri += 1
            if self.si < msgs_per_pair and len(self.sreqs) < 24:
                sz = self.sizes[self.si]
                payload = bytes([(self.si * 31 + j) % 256
                                 for j in range(min(sz, 997))])
                if sz > len(payload):
                    payload = (payload * (sz // 997 + 2))[:sz]
                buf = C.create_string_buffer(payload, max(sz, 1))
                r = plugin.isend(self.scomm, buf, sz, self.smh)
                if r is not None:
                    self.sbufs[self.si] = (buf, hashlib.sha256(
                        payload).digest())
                    self.sreqs[self.si] = r
                    self.si += 1
            for k in list(self.sreqs):
                done, size = plugin.test(self.sreqs[k])
                if done:
                    assert size == self.sizes[k]
                    del self.sreqs[k]
                    del self.sbufs[k]
                    self.sdone += 1
            for k in list(self.rreqs):
                done, size = plugin.test(self.rreqs[k])
                if done:
                    sz = self.sizes[k]
                    assert size == sz
                    got = self.rbufs[k].raw[:sz]
                    payload = bytes([(k * 31 + j) % 256
                                     for j in range(min(sz, 997))])
                    if sz > len(payload):
                        payload = (payload * (sz // 997 + 2))[:sz]
                    assert hashlib.sha256(got).digest() == hashlib.sha256(
                        payload).digest(), f"msg {k} corrupted"
                    del self.rreqs[k]
                    del self.rbufs[k]
                    self.rdone += 1

        def finished(self):
            return self.sdone == msgs_per_pair and self.rdone == msgs_per_pair

    dirs = []
    for lc, sc, rc in pairs:
        dirs.append(Dir(sc, rc))

    t0 = time.monotonic()
    while not all(d.finished() for d in dirs):
        assert time.monotonic() - t0 < 180, "stress test stalled"
        for d in rng.sample(dirs, len(dirs)):
            d.step()

    for lc, sc, rc in pairs:
        plugin.close_send(sc)
        plugin.close_recv(rc)
        plugin.close_listen(lc)


def test_many_parallel_comms(plugin):
    """32 comm pairs alive at once, one verified message each, closed in
    an interleaved order — stresses conn_id grouping in accept and the
    engines' shared-socket bookkeeping."""
    import ctypes as C
    import time

    pairs = [establish(plugin) for _ in range(32)]
    msgs = []
    for k, (lcomm, scomm, rcomm) in enumerate(pairs):
        smh = plugin.reg_mr(scomm, None, 0)
        rmh = plugin.reg_mr(rcomm, None, 0)
        size = 1000 * (k + 1)
        payload = bytes((k + j) % 256 for j in range(size))
        sbuf = C.create_string_buffer(payload, size)
        rbuf = C.create_string_buffer(size + 1)
        rreq = plugin.irecv(rcomm, rbuf, size, rmh)
        sreq = plugin.isend(scomm, sbuf, size, smh)
        assert rreq is not None and sreq is not None
        msgs.append([sreq, rreq, False, False, size, payload, sbuf, rbuf])
    t0 = time.monotonic()
    while any(not (m[2] and m[3]) for m in msgs):
        for m in msgs:
            if not m[2]:
                m[2], _ = plugin.test(m[0])
            if not m[3]:
                done, got = plugin.test(m[1])
                if done:
                    assert got == m[4]
                    m[3] = True
        assert time.monotonic() - t0 < 30
    for m in msgs:
        assert m[7].raw[:m[4]] == m[5]
    # interleaved close order (even pairs first, then odd)
    for k in list(range(0, 32, 2)) + list(range(1, 32, 2)):
        lcomm, scomm, rcomm = pairs[k]
        plugin.close_send(scomm)
        plugin.close_recv(rcomm)
        plugin.close_listen(lcomm)
