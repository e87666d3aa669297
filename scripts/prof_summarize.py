"""Summarize rocprofv3 rocpd result DBs into a small text table
(gpurun copies back at most 64 MiB; the raw DBs stay on the box)."""

import glob
import re
import sqlite3
import sys


def summarize(label: str, pattern: str, limit: int = 9) -> None:
    paths = glob.glob(pattern)
    if not paths:
        print(f"--- {label}: no results db at {pattern}")
        return
    db = sqlite3.connect(paths[0])
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    uid = [t for t in tables if t.startswith("rocpd_kernel_dispatch")][0]
    uid = uid.split("dispatch_")[1]
    total = db.execute(
        f"SELECT SUM(end-start)/1e6 FROM rocpd_kernel_dispatch_{uid}"
    ).fetchone()[0]
    print(f"--- {label}: total {total:.1f} ms")
    q = (f"SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6 "
         f"FROM rocpd_kernel_dispatch_{uid} kd "
         f"JOIN rocpd_info_kernel_symbol_{uid} ks ON kd.kernel_id=ks.id "
         f"GROUP BY ks.display_name ORDER BY 3 DESC LIMIT {limit}")
    for name, cnt, ms in db.execute(q):
        short = re.sub(r"[(].*", "", name)[:54]
        print(f"{ms:8.2f} ms {100 * ms / total:5.1f}% n={cnt:4d}  {short}")


if __name__ == "__main__":
    limit = 9
    args = []
    for arg in sys.argv[1:]:
        if arg.startswith("--limit="):
            limit = int(arg.split("=", 1)[1])
        else:
            args.append(arg)
    for arg in args:
        label, pattern = arg.split("=", 1)
        summarize(label, pattern, limit=limit)
