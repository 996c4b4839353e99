#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd .db: kernel stats and (if present) PMC counters.

Usage: python tools/rocpd_stats.py results.db [--pmc]
"""
import sqlite3, sys

def main(path):
    db = sqlite3.connect(path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    u = next(t for t in tabs if t.startswith('rocpd_kernel_dispatch_')).split('rocpd_kernel_dispatch_')[1]
    print("== kernels ==")
    q = f"""SELECT ks.display_name, COUNT(*), AVG(k.end-k.start)/1000.0, SUM(k.end-k.start)/1000.0
        FROM rocpd_kernel_dispatch_{u} k JOIN rocpd_info_kernel_symbol_{u} ks ON k.kernel_id=ks.id
        GROUP BY 1 ORDER BY 4 DESC LIMIT 15"""
    for name, n, avg, tot in cur.execute(q):
        print(f"{n:5d} x {avg:9.1f} us  tot {tot:10.1f} us  {name[:88]}")
    if f'rocpd_pmc_event_{u}' in tabs:
        try:
            q = f"""SELECT ks.display_name, pi.name, AVG(pe.value)
                FROM rocpd_pmc_event_{u} pe
                JOIN rocpd_kernel_dispatch_{u} k ON pe.event_id=k.event_id
                JOIN rocpd_info_kernel_symbol_{u} ks ON k.kernel_id=ks.id
                JOIN rocpd_info_pmc_{u} pi ON pe.pmc_id=pi.id
                GROUP BY 1,2 ORDER BY 1,2"""
            rows = list(cur.execute(q))
        except sqlite3.OperationalError as e:
            print("pmc join failed:", e)
            for t in tabs:
                if 'pmc' in t:
                    print(t, [r[1] for r in cur.execute(f'PRAGMA table_info({t})')])
            return
        cur_k = None
        vals = {}
        def flush():
            if not vals: return
            wc = vals.get('SQ_WAVE_CYCLES')
            print(f"-- {cur_k[:80]}")
            for nm, v in vals.items():
                pct = f" ({100*v/wc:.1f}% of wave cyc)" if wc and nm != 'SQ_WAVE_CYCLES' else ""
                print(f"   {nm:30s} {v:18.0f}{pct}")
        for name, pmc, v in rows:
            if name != cur_k:
                flush(); cur_k = name; vals = {}
            vals[pmc] = v
        flush()

if __name__ == '__main__':
    main(sys.argv[1])
