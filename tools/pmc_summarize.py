#!/usr/bin/env python3
"""Summarize rocprofv3 --pmc results (rocpd SQLite) into a per-kernel
counter table.  Schema-introspecting: rocpd table suffixes and counter
table names vary across rocprof versions."""
import glob
import sqlite3
import sys
from collections import defaultdict


def main(pattern: str, out_path: str | None = None):
    agg = defaultdict(lambda: defaultdict(float))
    counters_seen = set()
    for path in sorted(glob.glob(pattern)):
        db = sqlite3.connect(path)
        tables = [r[0] for r in db.execute(
            "SELECT name FROM sqlite_master WHERE type='table'")]
        ctab = [t for t in tables if "counter" in t.lower() or "pmc" in t.lower()]
        if not ctab:
            print(f"{path}: no counter tables; tables={tables[:25]}",
                  file=sys.stderr)
            continue
        for t in ctab:
            cols = [r[1] for r in db.execute(f"PRAGMA table_info({t})")]
            # expected shapes: (…, kernel name or id, counter name, value)
            name_col = next((c for c in cols if "kernel_name" in c.lower()), None)
            cname_col = next((c for c in cols if "counter_name" in c.lower()), None)
            val_col = next((c for c in cols if c.lower() in ("value", "counter_value")), None)
            if not (cname_col and val_col):
                print(f"{path}:{t}: cols={cols}", file=sys.stderr)
                continue
            if name_col:
                q = (f"SELECT {name_col}, {cname_col}, SUM({val_col}) "
                     f"FROM {t} GROUP BY 1, 2")
                rows = db.execute(q)
            else:
                # join against a kernel-dispatch/info table via kernel_id
                kid = next((c for c in cols if "kernel_id" in c.lower()), None)
                ktab = next((x for x in tables if "kernel" in x.lower()
                             and "info" in x.lower()), None)
                if not (kid and ktab):
                    print(f"{path}:{t}: no kernel name path; cols={cols}",
                          file=sys.stderr)
                    continue
                kcols = [r[1] for r in db.execute(f"PRAGMA table_info({ktab})")]
                kname = next((c for c in kcols if "name" in c.lower()), None)
                q = (f"SELECT k.{kname}, p.{cname_col}, SUM(p.{val_col}) "
                     f"FROM {t} p JOIN {ktab} k ON k.id = p.{kid} GROUP BY 1, 2")
                rows = db.execute(q)
            for kname_v, cname_v, val in rows:
                key = str(kname_v).split("(")[0][:52]
                agg[key][cname_v] += float(val or 0)
                counters_seen.add(cname_v)
    if not agg:
        print("no counter data found", file=sys.stderr)
        sys.exit(1)
    counters = sorted(counters_seen)
    lines = ["| kernel | " + " | ".join(counters) + " |",
             "|" + "---|" * (len(counters) + 1)]
    first = counters[0]
    for k, c in sorted(agg.items(), key=lambda kv: -max(kv[1].values()))[:12]:
        lines.append(
            "| `" + k + "` | " + " | ".join(f"{c.get(cn, 0):.3e}" for cn in counters) + " |"
        )
    text = "\n".join(lines) + "\n"
    if out_path:
        with open(out_path, "w") as f:
            f.write(text)
    print(text)


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
