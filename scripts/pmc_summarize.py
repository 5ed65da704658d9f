"""Summarize rocprofv3 rocpd .db outputs from scripts/pmc_profile.sh.

    python scripts/pmc_summarize.py gpurun_out/pmc

Joins kernel dispatches with their PMC values (summed over the counter's
hardware instances), aggregates per kernel symbol, and prints one table
per database: mean per-dispatch counter values + mean dispatch time.
"""
import glob
import os
import sqlite3
import sys
from collections import defaultdict


def table(cur, stem):
    row = cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE ?",
        (stem + "%",)).fetchone()
    return row[0] if row else None


def summarize(path):
    db = sqlite3.connect(path)
    cur = db.cursor()
    t_disp = table(cur, "rocpd_kernel_dispatch")
    t_pmc = table(cur, "rocpd_pmc_event")
    t_info = table(cur, "rocpd_info_pmc")
    t_sym = table(cur, "rocpd_info_kernel_symbol")
    if not t_disp:
        return
    names = dict(cur.execute(
        "SELECT id, COALESCE(display_name, kernel_name) FROM " + t_sym))
    counters = dict(cur.execute("SELECT id, name FROM " + t_info)) \
        if t_info else {}
    disp = list(cur.execute(
        "SELECT id, kernel_id, start, end FROM " + t_disp))
    # event_id in pmc_event == dispatch id (1-based sequence)
    pmc = defaultdict(lambda: defaultdict(float))
    if t_pmc and counters:
        for eid, cid, val in cur.execute(
                "SELECT event_id, pmc_id, value FROM " + t_pmc):
            pmc[eid][counters.get(cid, str(cid))] += val

    agg = defaultdict(lambda: defaultdict(float))
    cnt = defaultdict(int)
    tsum = defaultdict(float)
    for did, kid, start, end in disp:
        key = names.get(kid, str(kid))
        key = key.split("(")[0][:60]
        cnt[key] += 1
        tsum[key] += (end - start) * 1e-6  # ns -> ms
        for c, v in pmc.get(did, {}).items():
            agg[key][c] += v

    cols = sorted({c for k in agg.values() for c in k})
    print("\n##", os.path.basename(path))
    hdr = "%-48s %5s %9s" % ("kernel", "n", "ms/disp")
    for c in cols:
        hdr += " %18s" % c[-18:]
    print(hdr)
    for key in sorted(cnt, key=lambda k: -tsum[k]):
        if cnt[key] == 0:
            continue
        line = "%-48s %5d %9.4f" % (key, cnt[key], tsum[key] / cnt[key])
        for c in cols:
            line += " %18.3e" % (agg[key][c] / cnt[key])
        print(line)


def main():
    root = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/pmc"
    for path in sorted(glob.glob(os.path.join(root, "*", "*_results.db"))):
        summarize(path)


if __name__ == "__main__":
    main()
