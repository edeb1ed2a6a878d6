#!/usr/bin/env python3
"""Summarize a rocprofv3 results DB (kernel-trace) into a text table:
total/avg time per kernel, dispatch counts, effective HBM bandwidth for
the known slab kernels. Usage: prof_summary.py <results.db> [out.txt]"""
import sqlite3
import sys


def main(db_path, out_path=None):
    c = sqlite3.connect(db_path)
    t = [r[0] for r in c.execute("SELECT name FROM sqlite_master WHERE type='table'")
         if r[0].startswith('rocpd_kernel_dispatch')][0]
    suf = t.replace('rocpd_kernel_dispatch', '')
    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6, AVG(k.end-k.start)/1e3,
           AVG(k.grid_size_x), MAX(ks.arch_vgpr_count)
    FROM rocpd_kernel_dispatch{suf} k
    JOIN rocpd_info_kernel_symbol{suf} ks ON k.kernel_id=ks.id
    GROUP BY ks.display_name ORDER BY 3 DESC"""
    lines = [f"{'total_ms':>10} {'n':>6} {'avg_us':>10} {'grid':>8} {'vgpr':>5}  kernel"]
    total = 0.0
    for name, n, ms, us, grid, vgpr in c.execute(q):
        total += ms
        lines.append(f"{ms:10.2f} {n:6d} {us:10.1f} {int(grid):8d} {int(vgpr):5d}  {name[:90]}")
    lines.append(f"{total:10.2f}  total GPU kernel time")
    txt = "\n".join(lines)
    print(txt)
    if out_path:
        with open(out_path, "w") as f:
            f.write(txt + "\n")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else None)
