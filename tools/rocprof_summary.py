"""Summarise a rocprofv3 results.db into a markdown kernel table.

Usage: python tools/rocprof_summary.py gpurun_out/prof/*/NNN_results.db \
           profiles/rNN_name.md "title / context line"
"""
from __future__ import annotations

import glob
import sqlite3
import sys


def summarize(db_path: str, out_path: str, title: str) -> None:
    db = sqlite3.connect(db_path)
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = [t for t in tables if t.startswith("rocpd_kernel_dispatch_")]
    if not disp:
        raise SystemExit(f"no kernel dispatch table in {db_path}")
    sfx = disp[0][len("rocpd_kernel_dispatch_"):]
    rows = list(db.execute(f"""
        SELECT ks.display_name, COUNT(*) n, SUM(k.end-k.start)/1e6 total_ms,
               AVG(k.end-k.start)/1e3 avg_us,
               MAX(ks.arch_vgpr_count), MAX(ks.accum_vgpr_count),
               MAX(ks.sgpr_count), MAX(k.group_segment_size)
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        GROUP BY ks.display_name ORDER BY total_ms DESC LIMIT 40
    """))
    total = sum(r[2] for r in rows)
    with open(out_path, "w") as f:
        f.write(f"# {title}\n\n")
        f.write(f"Source: `{db_path}` (rocprofv3 --kernel-trace --stats, "
                "MI355X/gfx950)\n\n")
        f.write(f"Total kernel time: **{total:.3f} ms**\n\n")
        f.write("| total ms | calls | avg us | VGPR | AGPR | SGPR | LDS B |"
                " kernel |\n|---|---|---|---|---|---|---|---|\n")
        for r in rows:
            name = r[0].split("(")[0][:80]
            f.write(f"| {r[2]:.3f} | {r[1]} | {r[3]:.1f} | {r[4]} | {r[5]} "
                    f"| {r[6]} | {r[7]} | `{name}` |\n")
    print(f"wrote {out_path} ({len(rows)} kernels, {total:.3f} ms total)")


if __name__ == "__main__":
    paths = sorted(glob.glob(sys.argv[1]))
    summarize(paths[-1], sys.argv[2], sys.argv[3] if len(sys.argv) > 3
              else "rocprof kernel summary")
