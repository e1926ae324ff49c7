"""Aggregate a rocprofv3 PMC db into per-kernel counter means (runs on the GPU box)."""
import glob, sqlite3, sys, re, json

db = sqlite3.connect(sys.argv[1])
cur = db.cursor()
names = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
sfx = [n for n in names if n.startswith('rocpd_kernel_dispatch')][0].split('dispatch_')[1]
has_pmc = any(n.startswith('rocpd_pmc_event') for n in names)
q = f"""
SELECT k.display_name, p.name, COUNT(*), AVG(e.value), SUM(e.value)
FROM rocpd_pmc_event_{sfx} e
JOIN rocpd_info_pmc_{sfx} p ON e.pmc_id = p.id
JOIN rocpd_kernel_dispatch_{sfx} d ON e.event_id = d.event_id
JOIN rocpd_info_kernel_symbol_{sfx} k ON d.kernel_id = k.id
GROUP BY 1, 2 ORDER BY 1, 2
"""
out = {}
try:
    for kname, cname, cnt, avg, total in cur.execute(q):
        key = re.sub(r'[(<].*', '', kname)[:60]
        out.setdefault(key, {})[cname] = {"n": cnt, "avg": round(avg, 2)}
except Exception as ex:
    # schema discovery fallback
    for n in names:
        if 'pmc' in n:
            cols = [r[1] for r in cur.execute(f"PRAGMA table_info({n})")]
            out[n] = cols
print(json.dumps(out, indent=1))
