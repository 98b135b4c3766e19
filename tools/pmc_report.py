"""Aggregate PMC counters for our kernels from a rocpd db (run on the box)."""
import glob, sqlite3, sys
db = glob.glob(sys.argv[1])[0]
con = sqlite3.connect(db)
tabs = [r[0] for r in con.execute("SELECT name FROM sqlite_master WHERE type='table'")]
kd = next(t for t in tabs if 'kernel_dispatch' in t)
ks = next(t for t in tabs if 'kernel_symbol' in t)
pmc = next(t for t in tabs if t.startswith('rocpd_pmc_event'))
pi = next(t for t in tabs if 'info_pmc' in t)
rows = con.execute(f"""
SELECT s.display_name, p2.name, SUM(p.value)
FROM {pmc} p JOIN {kd} kd ON p.event_id = kd.event_id
JOIN {ks} s ON kd.kernel_id = s.id JOIN {pi} p2 ON p.pmc_id = p2.id
WHERE s.display_name LIKE '%conv3d%' OR s.display_name LIKE '%wgrad%'
GROUP BY s.display_name, p2.name""").fetchall()
agg = {}
for name, ctr, val in rows:
    agg.setdefault(name.split('(')[0][:48], {})[ctr] = val
for name, c in agg.items():
    wc = c.get('SQ_WAVE_CYCLES', 1)
    print(f"{name}")
    for k in ['SQ_WAIT_ANY', 'SQ_WAIT_INST_ANY', 'SQ_ACTIVE_INST_ANY',
              'SQ_ACTIVE_INST_VALU']:
        v = c.get(k, 0)
        print(f"   {k:22s} {v/wc*100:5.1f}% of wave cycles")
