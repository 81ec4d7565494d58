#!/usr/bin/env python3
"""Summarize a rocprofv3 results DB (kernel totals + steady-decode window)
into a markdown table for profiles/. Usage: profile_summary.py <db> <out.md>"""
import sqlite3
import sys


def main(db_path, out_path):
    con = sqlite3.connect(db_path)
    tables = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")]
    sfx = tables[0].replace("rocpd_kernel_dispatch_", "")
    lines = ["# rocprofv3 kernel summary", "",
             f"source: {db_path}", "",
             "## All-run kernel totals", "",
             "| total ms | calls | us/call | kernel |", "|---|---|---|---|"]
    q = f"""
SELECT ks.display_name, COUNT(*) n, SUM(k.end-k.start)/1e6 total_ms,
       AVG(k.end-k.start)/1e3 avg_us
FROM rocpd_kernel_dispatch_{sfx} k
JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
GROUP BY ks.display_name ORDER BY total_ms DESC LIMIT 20"""
    for name, n, tot, avg in con.execute(q):
        lines.append(f"| {tot:.1f} | {n} | {avg:.1f} | {name[:90]} |")
    lo, hi = con.execute(
        f"SELECT MIN(start), MAX(end) FROM rocpd_kernel_dispatch_{sfx}"
    ).fetchone()
    w0 = hi - (hi - lo) * 0.2
    lines += ["", "## Last-20% window (steady decode)", "",
              "| total ms | calls | kernel |", "|---|---|---|"]
    busy = 0.0
    q2 = f"""
SELECT ks.display_name, COUNT(*) n, SUM(k.end-k.start)/1e6 total_ms
FROM rocpd_kernel_dispatch_{sfx} k
JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
WHERE k.start > {w0}
GROUP BY ks.display_name ORDER BY total_ms DESC LIMIT 15"""
    for name, n, tot in con.execute(q2):
        busy += tot
        lines.append(f"| {tot:.1f} | {n} | {name[:90]} |")
    lines += ["", f"window wall: {(hi-w0)/1e6:.1f} ms, kernel busy: "
              f"{busy:.1f} ms ({busy/((hi-w0)/1e6)*100:.0f}%)", ""]
    with open(out_path, "w") as f:
        f.write("\n".join(lines))
    print(f"wrote {out_path}")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2])
