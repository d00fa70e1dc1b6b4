"""Aggregate a rocprofv3 rocpd .db into a small per-kernel summary CSV.

Usage: python tools/prof_summarize.py <results.db> <out.csv>
Run ON the GPU box right after rocprofv3 so only the summary (KBs) travels
back through gpurun_out (raw dispatch DBs can exceed the 64 MiB merge cap)."""
import re
import sqlite3
import sys


def main(db_path: str, out_path: str):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    q = f"""
    SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6, AVG(d.end-d.start)/1e3,
           MAX(s.arch_vgpr_count), MAX(s.accum_vgpr_count), MAX(s.group_segment_size)
    FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
    GROUP BY s.display_name ORDER BY SUM(d.end-d.start) DESC
    """
    rows = list(cur.execute(q))
    total = sum(r[2] for r in rows)
    with open(out_path, "w") as f:
        f.write("kernel,count,total_ms,avg_us,pct,vgpr,agpr,lds_bytes\n")
        for name, cnt, ms, avg, vgpr, agpr, lds in rows:
            nm = re.sub(r"[,\n]", ";", re.sub(r"\(.*", "", name))[:110]
            f.write(f"{nm},{cnt},{ms:.2f},{avg:.1f},{100*ms/total:.1f},{vgpr},{agpr},{lds}\n")
        f.write(f"TOTAL,,{total:.2f},,,,,\n")
    print(f"wrote {out_path}: total {total:.1f} ms over {len(rows)} kernels")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2])
