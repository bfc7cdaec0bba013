#!/usr/bin/env python3
"""Summarize rocprofv3 rocpd sqlite outputs: per-kernel dispatch stats
and (when present) per-kernel PMC counter sums.

Usage: python tools/rocpd_summary.py <results.db> [counter_name]
Prints a text table; used to produce the committed profiles/ summaries.
"""
import sqlite3
import sys
from collections import defaultdict


def main(path, counter=None):
    db = sqlite3.connect(path)
    tabs = {r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")}

    def t(prefix):
        for name in tabs:
            if name.startswith(prefix):
                return name
        raise KeyError(prefix)

    strings = dict(db.execute(f"SELECT id, string FROM {t('rocpd_string')}"))
    sym = {}
    for r in db.execute(f"SELECT id, display_name FROM {t('rocpd_info_kernel_symbol')}"):
        sym[r[0]] = strings.get(r[1], str(r[1]))

    stats = defaultdict(lambda: [0, 0.0, 0.0])  # name -> [n, total_ns, max]
    disp = {}
    for r in db.execute(
            f"SELECT id, kernel_id, start, end FROM {t('rocpd_kernel_dispatch')}"):
        name = sym.get(r[1], str(r[1]))
        dur = r[3] - r[2]
        s = stats[name]
        s[0] += 1
        s[1] += dur
        s[2] = max(s[2], dur)
        disp[r[0]] = name

    print(f"== kernel dispatches: {path}")
    print(f"{'kernel':60s} {'n':>5s} {'total_ms':>10s} {'avg_us':>10s}")
    for name, (n, tot, mx) in sorted(stats.items(), key=lambda kv: -kv[1][1]):
        short = name.split("(")[0][:60]
        print(f"{short:60s} {n:5d} {tot/1e6:10.3f} {tot/1e3/n:10.1f}")

    if "rocpd_pmc_event" in {x.rsplit('_00', 1)[0] for x in tabs}:
        pmc_info = dict(db.execute(
            f"SELECT id, name FROM {t('rocpd_info_pmc')}"))
        sums = defaultdict(lambda: defaultdict(float))
        cnts = defaultdict(lambda: defaultdict(int))
        for r in db.execute(
                f"SELECT event_id, pmc_id, value FROM {t('rocpd_pmc_event')}"):
            # event_id links to dispatch via event table? try direct
            pass
        # rocpd links pmc_event.event_id -> kernel_dispatch.event_id
        try:
            q = (f"SELECT kd.kernel_id, pi.name, pe.value "
                 f"FROM {t('rocpd_pmc_event')} pe "
                 f"JOIN {t('rocpd_kernel_dispatch')} kd ON pe.event_id = kd.event_id "
                 f"JOIN {t('rocpd_info_pmc')} pi ON pe.pmc_id = pi.id")
            for kid, cname, val in db.execute(q):
                name = sym.get(kid, str(kid)).split("(")[0][:60]
                sums[name][cname] += val
                cnts[name][cname] += 1
        except sqlite3.OperationalError as e:
            print("PMC join failed:", e)
            # dump schemas to adapt
            for tb in (t('rocpd_pmc_event'), t('rocpd_kernel_dispatch')):
                cols = [c[1] for c in db.execute(f"PRAGMA table_info({tb})")]
                print(tb, cols)
            return
        if sums:
            print("\n== PMC sums (value, n_dispatches, value/dispatch)")
            for name in sums:
                for cname, v in sums[name].items():
                    n = cnts[name][cname]
                    print(f"{name:60s} {cname:12s} {v:16.0f} {n:5d} "
                          f"{v/n:14.1f}")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
