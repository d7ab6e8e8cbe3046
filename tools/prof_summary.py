#!/usr/bin/env python3
"""Summarize rocprofv3 sqlite results into a markdown kernel table.

Usage: python tools/prof_summary.py gpurun_out/prof_native/native_results.db \
           profiles/bench_native_kernels.md ["title"]
"""

import sqlite3
import sys
from pathlib import Path


def summarize(db_path: str, out_path: str, title: str = "") -> None:
    c = sqlite3.connect(db_path)
    names = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sym = next(t for t in names if t.startswith("rocpd_info_kernel"))
    disp = next(t for t in names if t.startswith("rocpd_kernel_dispatch"))
    rows = list(c.execute(f"""
        SELECT s.kernel_name, COUNT(*), SUM(d.end-d.start)/1e6,
               AVG(d.end-d.start)/1e3
        FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
        GROUP BY s.kernel_name ORDER BY SUM(d.end-d.start) DESC"""))
    import subprocess
    dm = subprocess.run(["c++filt"], input="\n".join(r[0] for r in rows),
                        capture_output=True, text=True).stdout.splitlines()
    rows = [(d,) + tuple(r[1:]) for r, d in zip(rows, dm)]
    total = sum(r[2] for r in rows)
    out = Path(out_path)
    out.parent.mkdir(parents=True, exist_ok=True)
    with open(out, "w") as f:
        f.write(f"# {title or out.stem}\n\n")
        f.write(f"source: `{db_path}` — total GPU kernel time "
                f"{total:.1f} ms over {sum(r[1] for r in rows)} dispatches\n\n")
        f.write("| kernel | calls | total ms | avg µs | % |\n|---|---|---|---|---|\n")
        for name, n, ms, avg_us in rows[:40]:
            short = name.replace("(anonymous namespace)::", "")
            short = short.split("(")[0][:110]
            f.write(f"| `{short}` | {n} | {ms:.2f} | {avg_us:.1f} | "
                    f"{100*ms/total:.1f} |\n")
    print(f"wrote {out} ({len(rows)} kernels, {total:.1f} ms)")


if __name__ == "__main__":
    db, out = sys.argv[1], sys.argv[2]
    summarize(db, out, sys.argv[3] if len(sys.argv) > 3 else "")
