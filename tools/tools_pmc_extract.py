"""Extract per-kernel PMC sums from a rocpd .db into a small JSON."""
import json, sqlite3, sys

out = {}
for path in sys.argv[1:-1]:
    db = sqlite3.connect(path)
    cur = db.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' AND name LIKE"
        " 'rocpd_pmc_event%'")]
    if not t:
        continue
    sfx = t[0].replace('rocpd_pmc_event_', '')
    q = f"""
    SELECT ks.display_name, pi.name, SUM(pe.value), COUNT(*)
    FROM rocpd_pmc_event_{sfx} pe
    JOIN rocpd_kernel_dispatch_{sfx} kd ON pe.event_id = kd.event_id
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
    JOIN rocpd_info_pmc_{sfx} pi ON pe.pmc_id = pi.id
    GROUP BY ks.display_name, pi.name"""
    rows = {}
    for dn, cn, v, c in cur.execute(q):
        key = dn.split('(')[0]
        rows.setdefault(key, {})[cn] = [v, c]
    out[path] = rows
json.dump(out, open(sys.argv[-1], 'w'), indent=1)
print("wrote", sys.argv[-1])
