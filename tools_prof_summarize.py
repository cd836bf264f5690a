"""Summarizes a rocprofv3 db dir into a small text file and deletes it."""
import glob, shutil, sqlite3, sys

d = sys.argv[1]
out = sys.argv[2]
dbs = glob.glob(f'{d}/**/*.db', recursive=True)
lines = []
for db in dbs:
    con = sqlite3.connect(db)
    tables = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = [t for t in tables if t.startswith('rocpd_kernel_dispatch_')]
    if not kd:
        continue
    sfx = kd[0][len('rocpd_kernel_dispatch_'):]
    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
           AVG(k.end-k.start)/1e3
    FROM rocpd_kernel_dispatch_{sfx} k
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 20"""
    for name, n, tot, avg in con.execute(q):
        lines.append(f'{name[:70]:70s} n={n:6d} total={tot:9.2f}ms '
                     f'avg={avg:8.2f}us')
    con.close()
open(out, 'w').write('\n'.join(lines) + '\n')
print('\n'.join(lines[:20]))
shutil.rmtree(d, ignore_errors=True)
