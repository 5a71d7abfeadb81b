#!/usr/bin/env python3
"""Summarize rocprofv3 rocpd SQLite results into a small markdown table
(kernel-time stats, and PMC counters when present) — run on the GPU box so
only the summary travels back."""
import glob
import sqlite3
import sys


def main(pattern: str, out_path: str | None = None):
    paths = sorted(glob.glob(pattern))
    if not paths:
        print(f"no dbs match {pattern}", file=sys.stderr)
        sys.exit(1)
    lines = []
    for path in paths:
        db = sqlite3.connect(path)
        cur = db.cursor()
        tables = [
            r[0]
            for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")
        ]
        kd = [t for t in tables if t.startswith("rocpd_kernel_dispatch")]
        if not kd:
            continue
        sfx = kd[0][len("rocpd_kernel_dispatch_"):]
        lines.append(f"## {path}")
        lines.append("")
        lines.append("| total ms | calls | us/call | kernel |")
        lines.append("|---|---|---|---|")
        total = 0.0
        for name, n, ms, us in cur.execute(
            f"""SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
                       AVG(kd.end-kd.start)/1e3
                FROM rocpd_kernel_dispatch_{sfx} kd
                JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
                GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 20"""
        ):
            total += ms
            lines.append(f"| {ms:.2f} | {n} | {us:.1f} | `{name[:80]}` |")
        lines.append("")
        lines.append(f"Total kernel time: {total:.1f} ms")
        span = cur.execute(
            f"SELECT MIN(start), MAX(end) FROM rocpd_kernel_dispatch_{sfx}"
        ).fetchone()
        if span and span[0]:
            lines.append(f"Dispatch span: {(span[1] - span[0]) / 1e9:.2f} s")
        lines.append("")
        # PMC events, aggregated per kernel per counter
        pe = [t for t in tables if t.startswith("rocpd_pmc_event_")]
        if pe:
            try:
                rows = list(cur.execute(
                    f"""SELECT ks.display_name, pi.name, SUM(pev.value), COUNT(*)
                        FROM rocpd_pmc_event_{sfx} pev
                        JOIN rocpd_info_pmc_{sfx} pi ON pev.pmc_id = pi.id
                        JOIN rocpd_kernel_dispatch_{sfx} kd ON pev.event_id = kd.event_id
                        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
                        GROUP BY ks.display_name, pi.name
                        ORDER BY 3 DESC LIMIT 60"""
                ))
                if rows:
                    lines.append("| counter sum | n | counter | kernel |")
                    lines.append("|---|---|---|---|")
                    for kname, cname, val, n in rows:
                        lines.append(f"| {val:.3e} | {n} | {cname} | `{kname[:60]}` |")
                    lines.append("")
            except sqlite3.Error as e:
                lines.append(f"(pmc join failed: {e})")
        db.close()
    text = "\n".join(lines)
    if out_path:
        with open(out_path, "w") as f:
            f.write(text)
    print(text)


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
