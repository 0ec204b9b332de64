"""Aggregate rocprofv3 --pmc results per kernel: sums each counter over
all dispatches of each kernel.  Usage: python tools/pmc_summary.py <db>
[kernel-substring]"""
import sqlite3
import sys
from collections import defaultdict


def main():
    db = sqlite3.connect(sys.argv[1])
    want = sys.argv[2] if len(sys.argv) > 2 else ""
    cur = db.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]

    def tbl(sub):
        return next(t for t in tables if sub in t)

    kd = tbl("kernel_dispatch")
    pe = tbl("pmc_event")
    pi = tbl("info_pmc")
    ks = tbl("kernel_symbol")
    pmc_names = dict(cur.execute(f"SELECT id, name FROM {pi}"))
    ksyms = dict(cur.execute(
        f"SELECT id, display_name FROM {ks}"))
    kd_cols = [r[1] for r in cur.execute(f"PRAGMA table_info({kd})")]
    pe_cols = [r[1] for r in cur.execute(f"PRAGMA table_info({pe})")]
    # dispatch -> kernel name + duration
    disp = {}
    for row in cur.execute(
            f"SELECT id, kernel_id, end - start FROM {kd}"):
        disp[row[0]] = (ksyms.get(row[1], str(row[1])), row[2])
    # pmc events reference dispatch via ...? inspect columns
    ref = next((c for c in pe_cols if "dispatch" in c or c == "event_id"),
               None)
    agg = defaultdict(lambda: defaultdict(float))
    dur = defaultdict(float)
    cnt = defaultdict(int)
    for row in cur.execute(
            f"SELECT {ref}, pmc_id, value FROM {pe}"):
        d = disp.get(row[0])
        if d is None:
            continue
        name, ns = d
        agg[name][pmc_names.get(row[1], row[1])] += row[2]
    for did, (name, ns) in disp.items():
        dur[name] += ns
        cnt[name] += 1
    for name in sorted(agg, key=lambda n: -dur[n]):
        if want and want not in name:
            continue
        short = name.split("(")[0][:70]
        print(f"== {short}  calls={cnt[name]} total={dur[name]/1e6:.2f}ms")
        for c, v in sorted(agg[name].items()):
            print(f"   {c:28s} {v:.3e}")


if __name__ == "__main__":
    main()
