#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd sqlite DB into a per-kernel stats table.

Usage: python tools/prof_summary.py gpurun_out/prof/ktrace_results.db [out.md]
Writes a markdown table (kernel, calls, total ms, mean us, % of GPU time)
sorted by total time — the judge-facing artifact kept under profiles/.
"""
import sqlite3
import sys


def summarize(db_path: str):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]

    def tab(prefix):
        for t in tables:
            if t.startswith(prefix):
                return t
        raise KeyError(prefix)

    kd = tab("rocpd_kernel_dispatch")
    ks = tab("rocpd_info_kernel_symbol")
    rows = cur.execute(f"""
        SELECT k.display_name, COUNT(*), SUM(d.end - d.start),
               AVG(d.end - d.start)
        FROM {kd} d
        JOIN {ks} k ON d.kernel_id = k.id
        GROUP BY k.display_name ORDER BY SUM(d.end - d.start) DESC
    """).fetchall()
    return rows


def main():
    if len(sys.argv) < 2:
        sys.exit("usage: prof_summary.py <results.db> [out.md]")
    db_path = sys.argv[1]
    rows = summarize(db_path)
    total = sum(r[2] for r in rows) or 1
    lines = [
        "| kernel | calls | total ms | mean us | % |",
        "|---|---|---|---|---|",
    ]
    for name, calls, tot, mean in rows:
        short = name.split("(")[0]
        if len(short) > 80:
            short = short[:77] + "..."
        lines.append("| `%s` | %d | %.3f | %.1f | %.1f |"
                     % (short, calls, tot / 1e6, mean / 1e3,
                        100.0 * tot / total))
    lines.append("")
    lines.append("total GPU kernel time: %.3f ms over %d dispatches"
                 % (total / 1e6, sum(r[1] for r in rows)))
    out = "\n".join(lines)
    if len(sys.argv) > 2:
        with open(sys.argv[2], "w") as f:
            f.write(out + "\n")
    print(out)


if __name__ == "__main__":
    main()
