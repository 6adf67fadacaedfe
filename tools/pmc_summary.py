"""Summarize rocprofv3 rocpd .db: per-kernel time and SQ counter ratios."""
import collections
import sqlite3
import sys

db = sqlite3.connect(sys.argv[1])
cur = db.cursor()
names = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
sfx = [n for n in names if 'kernel_dispatch' in n][0].replace('rocpd_kernel_dispatch_', '')
q = f"""SELECT s.display_name, COUNT(DISTINCT k.id), SUM(k.end-k.start)/1e6
FROM rocpd_kernel_dispatch_{sfx} k
JOIN rocpd_info_kernel_symbol_{sfx} s ON k.kernel_id=s.id GROUP BY 1 ORDER BY 3 DESC LIMIT 12"""
for r in cur.execute(q):
    print(f"{r[2]:9.2f} ms {r[1]:6d} calls  {r[0][:58]}")
if any('pmc_event' in n for n in names):
    q = f"""SELECT s.display_name, pi.name, AVG(p.value) FROM rocpd_pmc_event_{sfx} p
    JOIN rocpd_kernel_dispatch_{sfx} k ON p.event_id=k.event_id
    JOIN rocpd_info_kernel_symbol_{sfx} s ON k.kernel_id=s.id
    JOIN rocpd_info_pmc_{sfx} pi ON p.pmc_id=pi.id GROUP BY 1,2"""
    agg = collections.defaultdict(dict)
    for n, c, v in cur.execute(q):
        agg[n.split('(')[0][:34]][c] = v
    for k, v in sorted(agg.items()):
        wc = v.get('SQ_WAVE_CYCLES', 1)
        wa = v.get('SQ_WAIT_ANY', 0) / wc
        wi = v.get('SQ_WAIT_INST_ANY', 0) / wc
        mf = v.get('SQ_VALU_MFMA_BUSY_CYCLES', 0) / (wc * 4)
        print(f"{k:36s} waitANY={wa:5.1%} waitINST={wi:5.1%} mfmaBusy={mf:5.1%}")
