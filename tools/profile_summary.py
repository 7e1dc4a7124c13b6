#!/usr/bin/env python3
"""Distill a rocprofv3 rocpd results database into a small markdown
kernel-time table (run on the GPU box so only the summary travels back).

Usage: python tools/profile_summary.py <results.db-or-dir> <out.md> [title]
"""
import glob
import os
import sqlite3
import sys


def main():
    src, out = sys.argv[1], sys.argv[2]
    title = sys.argv[3] if len(sys.argv) > 3 else os.path.basename(src)
    dbs = [src] if src.endswith(".db") else sorted(
        glob.glob(os.path.join(src, "**", "*.db"), recursive=True))
    assert dbs, f"no rocpd db under {src}"
    c = sqlite3.connect(dbs[0])
    tables = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sfx = disp[len("rocpd_kernel_dispatch_"):]
    rows = list(c.execute(f"""
        SELECT ks.display_name, COUNT(*) n, SUM(k.end-k.start)/1e6 ms,
               AVG(k.end-k.start)/1e3 us
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id
        GROUP BY ks.display_name ORDER BY ms DESC LIMIT 25"""))
    tot, cnt = list(c.execute(
        f"SELECT SUM(end-start)/1e6, COUNT(*) "
        f"FROM rocpd_kernel_dispatch_{sfx}"))[0]
    with open(out, "w") as f:
        f.write(f"# {title}\n\n")
        f.write(f"Total kernel time {tot:.1f} ms over {cnt} dispatches.\n\n")
        f.write("| total ms | calls | avg us | kernel |\n")
        f.write("|---:|---:|---:|---|\n")
        for name, n, ms, us in rows:
            nm = name.replace("|", "/")[:88]
            f.write(f"| {ms:.2f} | {n} | {us:.1f} | `{nm}` |\n")
    print(f"wrote {out}: {tot:.1f} ms / {cnt} dispatches")


if __name__ == "__main__":
    main()
