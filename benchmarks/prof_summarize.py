"""Summarize a rocprofv3 results.db into per-kernel totals (text)."""
import collections
import glob
import re
import sqlite3
import sys

db_path = sorted(glob.glob(sys.argv[1]))[-1]
out_path = sys.argv[2]
db = sqlite3.connect(db_path)
cur = db.cursor()
t = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type='table'")
     if 'kernel_dispatch' in r[0]][0]
sfx = t.replace('rocpd_kernel_dispatch_', '')
q = f"""SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)
FROM rocpd_kernel_dispatch_{sfx} kd
JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id=ks.id
GROUP BY ks.display_name ORDER BY SUM(kd.end-kd.start) DESC"""
rows = list(cur.execute(q))
tot = sum(r[2] for r in rows)
with open(out_path, "w") as f:
    f.write(f"total GPU time: {tot/1e9:.2f} s over {sum(r[1] for r in rows)} dispatches\n")
    for name, n, dur in rows[:40]:
        short = re.sub(r'\(anonymous namespace\)::', '', name); short = re.sub(r'\(.*', '', short)[:84]
        f.write(f"{dur/tot*100:5.1f}%  {dur/1e6:9.1f} ms  n={n:7d}  {short}\n")
    miopen = [(name, n, dur) for name, n, dur in rows
              if 'miopen' in name.lower() or 'Miopen' in name]
    f.write(f"\nMIOpen kernels: {len(miopen)} "
            f"({sum(d for _,_,d in miopen)/tot*100:.2f}% of GPU time)\n")
    for name, n, dur in miopen[:10]:
        f.write(f"  {dur/1e6:.1f} ms n={n} {name[:80]}\n")
print("wrote", out_path)
