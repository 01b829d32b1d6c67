#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd results DB (kernel-trace) into the per-kernel
table committed under profiles/: n / total / avg / min / max ms + dispatch
geometry + register counts. Usage: rocpd_summary.py <results.db> [out.txt]"""
import sqlite3
import sys


def summarize(db_path):
    con = sqlite3.connect(db_path)
    names = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(n for n in names if 'rocpd_kernel_dispatch' in n)
    ks = next(n for n in names if 'rocpd_info_kernel_symbol' in n)
    q = f"""
      SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6,
             AVG(d.end-d.start)/1e6, MIN(d.end-d.start)/1e6,
             MAX(d.end-d.start)/1e6,
             MAX(d.grid_size_x*d.grid_size_y*d.grid_size_z),
             MAX(d.workgroup_size_x*d.workgroup_size_y*d.workgroup_size_z),
             MAX(s.arch_vgpr_count), MAX(s.sgpr_count),
             MAX(d.group_segment_size), MAX(d.private_segment_size)
      FROM {kd} d JOIN {ks} s ON d.kernel_id = s.id
      GROUP BY s.display_name ORDER BY 3 DESC"""
    rows = con.execute(q).fetchall()
    out = [f"{'kernel':60s} {'n':>4} {'total_ms':>9} {'avg_ms':>8} "
           f"{'min_ms':>8} {'max_ms':>8} {'grid_thr':>11} {'blk':>4} "
           f"{'vgpr':>5} {'sgpr':>5} {'lds_B':>7} {'scr_B':>6}"]
    for r in rows:
        name = r[0][:60]
        out.append(f"{name:60s} {r[1]:>4} {r[2]:>9.3f} {r[3]:>8.3f} "
                   f"{r[4]:>8.3f} {r[5]:>8.3f} {r[6]:>11} {r[7]:>4} "
                   f"{r[8]:>5} {r[9]:>5} {r[10]:>7} {r[11]:>6}")
    return "\n".join(out)


if __name__ == "__main__":
    text = summarize(sys.argv[1])
    if len(sys.argv) > 2:
        with open(sys.argv[2], "w") as f:
            f.write(text + "\n")
    print(text)
