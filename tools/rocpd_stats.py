#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd .db: per-kernel dispatch counts and
durations (the judge-facing evidence committed under profiles/).

Usage: python tools/rocpd_stats.py results.db [--out summary.md]
"""
from __future__ import annotations

import argparse
import sqlite3


def suffix_of(db: sqlite3.Connection) -> str:
    row = db.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch_%'").fetchone()
    if not row:
        raise SystemExit("no kernel dispatch table in this db")
    return row[0][len("rocpd_kernel_dispatch_"):]


def kernel_stats(path: str) -> list[dict]:
    db = sqlite3.connect(path)
    sfx = suffix_of(db)
    q = f"""
      SELECT s.display_name AS name, COUNT(*) AS calls,
             SUM(d.end - d.start) AS total_ns,
             AVG(d.end - d.start) AS avg_ns,
             MIN(d.end - d.start) AS min_ns,
             MAX(d.end - d.start) AS max_ns
      FROM rocpd_kernel_dispatch_{sfx} d
      JOIN rocpd_info_kernel_symbol_{sfx} s
        ON s.id = d.kernel_id
      GROUP BY s.display_name ORDER BY total_ns DESC
    """
    rows = []
    for r in db.execute(q):
        rows.append({
            "name": r[0].split("(")[0], "calls": r[1],
            "total_ms": round(r[2] / 1e6, 3), "avg_us": round(r[3] / 1e3, 2),
            "min_us": round(r[4] / 1e3, 2), "max_us": round(r[5] / 1e3, 2),
        })
    return rows


def memcpy_stats(path: str) -> list[dict]:
    db = sqlite3.connect(path)
    sfx = suffix_of(db)
    try:
        q = f"""
          SELECT name_id, COUNT(*), SUM(end - start), SUM(size)
          FROM rocpd_memory_copy_{sfx} GROUP BY name_id
        """
        out = []
        for name_id, calls, ns, size in db.execute(q):
            name = db.execute(
                f"SELECT string FROM rocpd_string_{sfx} WHERE id=?",
                (name_id,)).fetchone()
            out.append({
                "op": name[0] if name else str(name_id), "calls": calls,
                "total_ms": round(ns / 1e6, 3),
                "GB": round((size or 0) / 1e9, 3),
                "GBps": round((size or 0) / ns, 3) if ns else 0,
            })
        return out
    except sqlite3.OperationalError:
        return []


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("--out", default="")
    args = ap.parse_args()
    lines = ["| kernel | calls | total ms | avg us | min us | max us |",
             "|---|---|---|---|---|---|"]
    for r in kernel_stats(args.db):
        lines.append(f"| `{r['name']}` | {r['calls']} | {r['total_ms']} | "
                     f"{r['avg_us']} | {r['min_us']} | {r['max_us']} |")
    mc = memcpy_stats(args.db)
    if mc:
        lines += ["", "| memcpy | calls | total ms | GB | GB/s |",
                  "|---|---|---|---|---|"]
        for r in mc:
            lines.append(f"| {r['op']} | {r['calls']} | {r['total_ms']} | "
                         f"{r['GB']} | {r['GBps']} |")
    text = "\n".join(lines)
    print(text)
    if args.out:
        with open(args.out, "w") as f:
            f.write(text + "\n")


if __name__ == "__main__":
    main()
