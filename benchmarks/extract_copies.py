#!/usr/bin/env python3
"""Extract memory-copy + kernel timeline rows from a rocprofv3 rocpd DB
into a compact CSV (overlap evidence for the staging pipeline)."""

import glob
import sqlite3
import sys

db_glob, out_csv = sys.argv[1], sys.argv[2]
path = sorted(glob.glob(db_glob))[-1]
db = sqlite3.connect(path)
cur = db.cursor()
tables = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")]


def t(prefix):
    m = [x for x in tables if x.startswith(prefix)]
    return m[0] if m else None


strings = {}
st = t("rocpd_string")
if st:
    cols = [c[1] for c in cur.execute(f"PRAGMA table_info({st})")]
    val = "string" if "string" in cols else cols[-1]
    for i, v in cur.execute(f"SELECT id, {val} FROM {st}"):
        strings[i] = v

rows = []
mc = t("rocpd_memory_copy")
if mc:
    for s, e, nid, size, stream in cur.execute(
            f"SELECT start, end, name_id, size, stream_id FROM {mc} "
            "ORDER BY start"):
        rows.append((s, e, f"copy:{strings.get(nid, nid)}:{size}B:"
                           f"stream{stream}"))
kd = t("rocpd_kernel_dispatch")
ks = t("rocpd_info_kernel_symbol")
if kd and ks:
    for s, e, name in cur.execute(
            f"SELECT d.start, d.end, y.display_name FROM {kd} d "
            f"JOIN {ks} y ON d.kernel_id=y.id ORDER BY d.start"):
        rows.append((s, e, "kernel:" + name[:60]))
rows.sort()
with open(out_csv, "w") as f:
    f.write("start_ns,end_ns,what\n")
    for s, e, name in rows:
        f.write(f"{s},{e},{name}\n")
print(f"{len(rows)} rows -> {out_csv}")
