#!/usr/bin/env python3
"""Summarize a rocprofv3 PMC sqlite DB into a markdown per-kernel table.

Usage: python tools/pmc_summary.py <results.db> <out.md> ["title"]
"""

import collections
import sqlite3
import subprocess
import sys
from pathlib import Path


def main(db_path: str, out_path: str, title: str = "") -> None:
    c = sqlite3.connect(db_path)
    names = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sym = next(t for t in names if t.startswith("rocpd_info_kernel"))
    disp = next(t for t in names if t.startswith("rocpd_kernel_dispatch"))
    info = next(t for t in names if t.startswith("rocpd_info_pmc"))
    ev = next(t for t in names if t.startswith("rocpd_pmc_event"))
    rows = list(c.execute(f"""
      SELECT s.kernel_name, i.name, SUM(e.value)
      FROM {ev} e JOIN {info} i ON e.pmc_id = i.id
      JOIN {disp} d ON e.event_id = d.id
      JOIN {sym} s ON d.kernel_id = s.id
      GROUP BY s.id, i.id"""))
    agg = collections.defaultdict(dict)
    for kn, cn, v in rows:
        agg[kn][cn] = v
    res = sorted(((d.get("SQ_WAVE_CYCLES", 0), kn, d)
                  for kn, d in agg.items()), reverse=True)
    out = Path(out_path)
    out.parent.mkdir(parents=True, exist_ok=True)
    with open(out, "w") as f:
        f.write(f"# {title or out.stem}\n\nsource: `{db_path}`\n\n")
        f.write("| kernel | wave Gcyc | wait_any % | wait_inst % | "
                "bank_conf M |\n|---|---|---|---|---|\n")
        for wc, kn, d in res[:40]:
            if wc < 1e6:
                continue
            dm = subprocess.run(["c++filt"], input=kn, capture_output=True,
                                text=True).stdout.strip()
            wa = d.get("SQ_WAIT_ANY", 0) / wc * 100
            wi = d.get("SQ_WAIT_INST_ANY", 0) / wc * 100
            bc = d.get("SQ_LDS_BANK_CONFLICT", 0) / 1e6
            f.write(f"| `{dm[:90]}` | {wc/1e9:.2f} | {wa:.1f} | {wi:.1f} | "
                    f"{bc:.1f} |\n")
    print(f"wrote {out}")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2], sys.argv[3] if len(sys.argv) > 3 else "")
