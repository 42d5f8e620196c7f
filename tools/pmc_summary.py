#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc rocpd DB per kernel: wave cycles, stall
fractions, HBM fetch bytes. Usage: pmc_summary.py db [out.md]"""
import sqlite3
import sys


def main():
    if len(sys.argv) < 2:
        sys.exit("usage: pmc_summary.py <results.db> [out.md]")
    db = sqlite3.connect(sys.argv[1])
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]

    def tab(p):
        return [t for t in tabs if t.startswith(p)][0]

    pmc, kd, ks = (tab("rocpd_pmc_event"), tab("rocpd_kernel_dispatch"),
                   tab("rocpd_info_kernel_symbol"))
    ipmc = tab("rocpd_info_pmc")
    names = dict(cur.execute(f"SELECT id, name FROM {ipmc}").fetchall())
    rows = cur.execute(f"""
      SELECT k.display_name, p.pmc_id, SUM(p.value),
             SUM(d.end - d.start), COUNT(DISTINCT d.id)
      FROM {pmc} p
      JOIN {kd} d ON d.event_id = p.event_id
      JOIN {ks} k ON d.kernel_id = k.id
      GROUP BY k.display_name, p.pmc_id
    """).fetchall()
    agg = {}
    for disp, pid, val, ns, calls in rows:
        short = disp.split("(")[0].split("grapehip::")[-1][:48]
        e = agg.setdefault(short, {"ns": ns, "calls": calls})
        e[names[pid]] = val
    out = ["| kernel | calls | ms | HBM GB | GB/s | stall% (wait/wave) |",
           "|---|---|---|---|---|---|"]
    for k, e in sorted(agg.items(), key=lambda kv: -kv[1]["ns"]):
        ms = e["ns"] / 1e6
        # FETCH_SIZE counts KB (TCC_EA_RDREQ x 64B, reported in KB)
        fetch = e.get("FETCH_SIZE", 0) * 1024 / 1e9
        wave = e.get("SQ_WAVE_CYCLES", 0)
        wait = e.get("SQ_WAIT_ANY", 0)
        stall = 100.0 * wait / wave if wave else 0
        bw = fetch / (ms / 1e3) if ms else 0
        out.append("| `%s` | %d | %.2f | %.2f | %.0f | %.0f |"
                   % (k, e["calls"], ms, fetch, bw, stall))
    text = "\n".join(out)
    if len(sys.argv) > 2:
        open(sys.argv[2], "w").write(text + "\n")
    print(text)


if __name__ == "__main__":
    main()
