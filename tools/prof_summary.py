"""Aggregate a rocprofv3 rocpd SQLite database into a per-kernel table
(count, total us, mean us, share).  Usage:
    python tools/prof_summary.py <results.db> [out.md]
"""
import sqlite3
import sys


def main():
    db = sqlite3.connect(sys.argv[1])
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next((t for t in tables if "kernel_dispatch" in t), None)
    if kd is None:
        print("tables:", tables)
        raise SystemExit("no kernel_dispatch table")
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({kd})")]
    # find the kernel-name join: kernel_id -> kernel info -> string table
    ki = next((t for t in tables if "kernel" in t and "info" in t.replace(
        "_", " ")), None)
    rows = None
    if ki:
        kcols = [r[1] for r in cur.execute(f"PRAGMA table_info({ki})")]
        name_col = next((c for c in kcols if "name" in c), None)
        if name_col and "kernel_id" in cols:
            q = (f"SELECT k.{name_col}, COUNT(*), "
                 f"SUM(d.end - d.start) FROM {kd} d "
                 f"JOIN {ki} k ON d.kernel_id = k.id "
                 f"GROUP BY k.{name_col}")
            try:
                rows = cur.execute(q).fetchall()
            except sqlite3.OperationalError as e:
                print("join failed:", e, "kcols:", kcols)
    if rows is None:
        print(f"{kd} cols:", cols)
        if ki:
            print(f"{ki} cols:",
                  [r[1] for r in cur.execute(f"PRAGMA table_info({ki})")])
        raise SystemExit("adapt query")
    # string table indirection?
    if rows and isinstance(rows[0][0], int):
        st = next((t for t in tables if t.endswith("string")), None)
        smap = dict(cur.execute(f"SELECT id, string FROM {st}"))
        rows = [(smap.get(r[0], str(r[0])), r[1], r[2]) for r in rows]
    total = sum(r[2] for r in rows) or 1
    rows.sort(key=lambda r: -r[2])
    out = []
    out.append(f"| kernel | calls | total ms | mean us | share |")
    out.append("|---|---|---|---|---|")
    for name, cnt, ns in rows[:40]:
        short = name.split("(")[0][:70]
        out.append(f"| {short} | {cnt} | {ns/1e6:.2f} | "
                   f"{ns/1e3/cnt:.1f} | {100*ns/total:.1f}% |")
    text = "\n".join(out)
    print(text)
    if len(sys.argv) > 2:
        with open(sys.argv[2], "w") as f:
            f.write(text + "\n")


if __name__ == "__main__":
    main()
