"""Top-kernel summary from a rocprofv3 rocpd SQLite db (UUID-suffixed
tables). Usage: python scripts/summarize_rocpd.py <results.db> [top_n]"""
import sqlite3
import sys

db = sys.argv[1]
top = int(sys.argv[2]) if len(sys.argv) > 2 else 30
c = sqlite3.connect(db)
tables = [r[0] for r in c.execute(
    "SELECT name FROM sqlite_master WHERE type='table'").fetchall()]
kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
ks = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
rows = c.execute(f"""
    SELECT ks.display_name, COUNT(*) AS calls,
           SUM(kd.end - kd.start) / 1e6 AS total_ms,
           AVG(kd.end - kd.start) / 1e3 AS avg_us
    FROM {kd} kd JOIN {ks} ks ON kd.kernel_id = ks.id
    GROUP BY 1 ORDER BY 3 DESC LIMIT {top}
""").fetchall()
total_ms = c.execute(
    f"SELECT SUM(end - start) / 1e6 FROM {kd}").fetchone()[0]
print(f"{'total_ms':>9} {'calls':>7} {'avg_us':>8}  kernel  "
      f"(all kernels: {total_ms:.1f} ms)")
for name, calls, ms, avg in rows:
    print(f"{ms:9.2f} {calls:7d} {avg:8.1f}  {name[:100]}")
