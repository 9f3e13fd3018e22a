#!/usr/bin/env python3
"""Summarize a rocprofv3 results .db (kernel-trace) into a text table.

Usage: python profiles/summarize_db.py <results.db> [out.txt]
The committed summaries under profiles/ are produced by this script from
`rocprofv3 --kernel-trace --stats -- python bench.py ...` runs on the
MI355X box (the .db itself stays in gpurun_out/, which is scratch).
"""
import re
import sqlite3
import sys


def summarize(db_path, out=None):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = list(cur.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(d.end-d.start),
               AVG(d.end-d.start), MIN(d.start), MAX(d.end)
        FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
        GROUP BY s.display_name ORDER BY SUM(d.end-d.start) DESC"""))
    (t0, t1), = cur.execute(f"SELECT MIN(start), MAX(end) FROM {disp}")
    lines = [f"# kernel summary of {db_path}",
             f"# dispatch span: {(t1-t0)/1e6:.2f} ms",
             f"{'total_ms':>10} {'calls':>7} {'avg_us':>9}  name"]
    total = 0
    for name, calls, tot, avg, _, _ in rows:
        name = re.sub(r"<[^>]*>", "<>", name)
        name = re.sub(r"\(.*", "", name)[:80]
        total += tot
        lines.append(f"{tot/1e6:10.3f} {calls:7d} {avg/1e3:9.1f}  {name}")
    lines.append(f"# total kernel time: {total/1e6:.2f} ms "
                 f"({100*total/(t1-t0):.1f}% of span)")
    text = "\n".join(lines) + "\n"
    if out:
        open(out, "w").write(text)
    return text


if __name__ == "__main__":
    print(summarize(sys.argv[1],
                    sys.argv[2] if len(sys.argv) > 2 else None))
