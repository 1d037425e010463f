#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite results DB: top kernels by total
GPU time.  Usage: python tools/rocpd_summary.py results.db [out.csv]"""
import sqlite3
import sys


def main():
    db = sys.argv[1]
    out = sys.argv[2] if len(sys.argv) > 2 else None
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tables if "kernel_dispatch" in t]
    assert kd, tables
    kd = kd[0]
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({kd})")]
    # find kernel-name resolution path
    ki = [t for t in tables if "kernel" in t and "info" in t.lower()]
    rows = None
    for q in (
        # rocpd >= 8: kernel_dispatch has region/kernel_id -> kernel codeobj table
        f"""SELECT s.string AS name, COUNT(*), SUM(k.end - k.start)
            FROM {kd} k
            JOIN rocpd_info_kernel_symbol ks ON k.kernel_id = ks.id
            JOIN rocpd_string s ON ks.kernel_name_id = s.id
            GROUP BY name""",
        f"""SELECT ks.display_name AS name, COUNT(*), SUM(k.end - k.start)
            FROM {kd} k
            JOIN rocpd_info_kernel_symbol ks ON k.kernel_id = ks.id
            GROUP BY name""",
    ):
        try:
            rows = cur.execute(q).fetchall()
            break
        except sqlite3.Error as e:
            print("query failed:", e, file=sys.stderr)
    if rows is None:
        print("tables:", tables, file=sys.stderr)
        for t in tables:
            print(t, [r[1] for r in cur.execute(f"PRAGMA table_info({t})")],
                  file=sys.stderr)
        sys.exit(1)
    rows.sort(key=lambda r: -(r[2] or 0))
    total = sum(r[2] or 0 for r in rows)
    lines = [f"# total GPU kernel time: {total / 1e9:.3f} s"]
    lines.append("total_ms,calls,mean_us,pct,name")
    for name, calls, dur in rows[:60]:
        if dur is None:
            continue
        lines.append(
            f"{dur / 1e6:.2f},{calls},{dur / 1e3 / max(calls, 1):.1f},"
            f"{100.0 * dur / total:.1f},{name[:120]}"
        )
    text = "\n".join(lines)
    print(text)
    if out:
        with open(out, "w") as f:
            f.write(text + "\n")


if __name__ == "__main__":
    main()
