#!/usr/bin/env python3
"""Summarize PMC counters from a rocprofv3 rocpd .db (per-kernel means).

Usage: python pmc_db_summarize.py <results.db> [--schema]
"""
import sqlite3
import sys
from collections import defaultdict


def main():
    db = sys.argv[1]
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    if "--schema" in sys.argv:
        for t in tables:
            cols = [c[1] for c in cur.execute(f"PRAGMA table_info({t})")]
            print(t, cols)
        return

    def tab(prefix):
        m = [t for t in tables if t.startswith(prefix)]
        return m[0] if m else None

    kd = tab("rocpd_kernel_dispatch")
    ks = tab("rocpd_info_kernel_symbol")
    pe = tab("rocpd_pmc_event")
    pi = tab("rocpd_info_pmc")
    if not all([kd, ks, pe, pi]):
        print("missing tables; run with --schema", kd, ks, pe, pi)
        return
    kd_cols = [c[1] for c in cur.execute(f"PRAGMA table_info({kd})")]
    pe_cols = [c[1] for c in cur.execute(f"PRAGMA table_info({pe})")]
    pi_cols = [c[1] for c in cur.execute(f"PRAGMA table_info({pi})")]
    # column-name guesses with fallbacks
    def pick(cols, *cands):
        for c in cands:
            if c in cols:
                return c
        raise KeyError((cands, cols))

    pe_val = pick(pe_cols, "value", "counter_value", "Counter_Value")
    pe_pmc = pick(pe_cols, "pmc_id", "pmc_info_id", "counter_id")
    pe_disp = pick(pe_cols, "event_id", "dispatch_id", "correlation_id")
    pi_name = pick(pi_cols, "name", "symbol", "counter_name")
    kd_start = pick(kd_cols, "start", "start_timestamp")
    kd_end = pick(kd_cols, "end", "end_timestamp")
    kd_kid = pick(kd_cols, "kernel_id")
    kd_disp = pick(kd_cols, "event_id", "dispatch_id", "id")

    q = f"""
      SELECT s.display_name, p.{pi_name}, e.{pe_val},
             d.{kd_end} - d.{kd_start}
      FROM {pe} e
      JOIN {pi} p ON e.{pe_pmc} = p.id
      JOIN {kd} d ON e.{pe_disp} = d.{kd_disp}
      JOIN {ks} s ON d.{kd_kid} = s.id
    """
    agg = defaultdict(lambda: [0.0, 0])
    times = defaultdict(lambda: [0.0, 0])
    def clean(n):
        n = n.replace("(anonymous namespace)::", "")
        return n.split("(")[0].split("<")[0]

    nrows = 0
    for name, cname, val, dt in cur.execute(q):
        nrows += 1
        name = clean(name)
        if "rocclr" in name:
            continue
        a = agg[(name, cname)]
        a[0] += float(val)
        a[1] += 1
        key = (name, cname)
        # dispatch time: count once per (dispatch) -> approximate by first
        # counter of each dispatch; use per-counter row count as n
    # per-kernel dispatch times from kd directly
    q2 = f"""SELECT s.display_name, COUNT(*), AVG(d.{kd_end}-d.{kd_start})
             FROM {kd} d JOIN {ks} s ON d.{kd_kid}=s.id
             GROUP BY s.display_name"""
    for name, n, avg in cur.execute(q2):
        name = clean(name)
        t = times[name]
        # merge groups that clean() collapses together
        tot = t[0] * t[1] + (avg / 1e3) * n
        t[1] += n
        t[0] = tot / t[1]
    if nrows == 0:
        print("NO pmc rows; counters recorded in rocpd_info_pmc:")
        for r in cur.execute(f"SELECT * FROM {pi} LIMIT 40"):
            print(" ", r)
        return
    kernels = sorted({k for (k, _) in agg})
    for k in kernels:
        us, n = times.get(k, [0, 0])
        print(f"\n{k}  ({n} dispatches, avg {us:.1f} us)")
        w = m = wa = ia = None
        for (kk, c) in sorted(agg):
            if kk != k:
                continue
            s, cnt = agg[(kk, c)]
            per = s / max(1, n)
            if "SIZE" in c:
                print(f"  {c:<28} {per * 32 / 1e6:12.1f} MB/dispatch")
            else:
                print(f"  {c:<28} {per:12.3e} /dispatch")
            if c == "SQ_WAVE_CYCLES":
                w = s
            if c == "SQ_VALU_MFMA_BUSY_CYCLES":
                m = s
            if c == "SQ_WAIT_ANY":
                wa = s
            if c == "SQ_WAIT_INST_ANY":
                ia = s
        if w:
            if m is not None:
                print(f"  {'-> MFMA busy / wave cyc':<28} "
                      f"{m / (4 * w) * 100:11.1f} %")
            if wa is not None:
                print(f"  {'-> WAIT_ANY / wave cyc':<28} "
                      f"{wa / w * 100:11.1f} %")
            if ia is not None:
                print(f"  {'-> WAIT_INST / wave cyc':<28} "
                      f"{ia / w * 100:11.1f} %")


if __name__ == "__main__":
    main()
