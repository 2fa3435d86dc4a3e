#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite database (kernel-trace) into per-kernel
aggregate stats: calls, total/mean time, share of GPU time. Used to turn
gpurun rocprof captures into the committed summaries under profiles/."""

import glob
import sqlite3
import sys


def summarize(path: str) -> str:
    db = sqlite3.connect(path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = list(cur.execute(
        f"""SELECT s.display_name, COUNT(*), SUM(d.end - d.start), AVG(d.end - d.start),
                   MIN(d.end - d.start), MAX(d.end - d.start),
                   MAX(s.arch_vgpr_count), MAX(s.accum_vgpr_count), MAX(s.sgpr_count),
                   MAX(d.grid_size_x), MAX(d.workgroup_size_x)
            FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
            GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC"""))
    total = sum(r[2] for r in rows) or 1
    out = [f"# rocpd kernel summary: {path}",
           f"{'kernel':<44} {'calls':>6} {'total_ms':>10} {'mean_us':>10} {'min_us':>8} {'max_us':>9} {'%gpu':>6} {'vgpr':>5} {'agpr':>5} {'grid':>9} {'wg':>5}"]
    for name, calls, tot, avg, mn, mx, vgpr, agpr, sgpr, grid, wg in rows:
        short = name.split("(")[0]
        if len(short) > 43:
            short = short[:40] + "..."
        out.append(f"{short:<44} {calls:>6} {tot/1e6:>10.3f} {avg/1e3:>10.2f} {mn/1e3:>8.2f} {mx/1e3:>9.2f} {100*tot/total:>5.1f}% {vgpr:>5} {agpr:>5} {grid:>9} {wg:>5}")
    out.append(f"{'TOTAL':<44} {sum(r[1] for r in rows):>6} {total/1e6:>10.3f}")
    return "\n".join(out)


if __name__ == "__main__":
    paths = sys.argv[1:] or sorted(glob.glob("gpurun_out/prof*/**/*_results.db", recursive=True))
    for p in paths:
        print(summarize(p))
        print()
