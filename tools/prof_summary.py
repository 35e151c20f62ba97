"""Summarize a rocprofv3 results DB (rocpd sqlite) into a small CSV:
per-kernel call count, total/avg time. Usage:
  python tools/prof_summary.py <results.db> <out.csv> [top_n]
"""
import sqlite3
import sys


def summarize(db_path: str, out_path: str, top_n: int = 60):
    c = sqlite3.connect(db_path)
    tabs = [r[0] for r in c.execute(
        "select name from sqlite_master where type='table'")]
    kd = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    ks = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    rows = list(c.execute(f"""
        select s.display_name, count(*), sum(k.end-k.start)/1e6,
               avg(k.end-k.start)/1e3
        from {kd} k join {ks} s on k.kernel_id = s.id
        group by s.display_name order by 3 desc limit {top_n}"""))
    total_ms = sum(r[2] for r in rows)
    with open(out_path, "w") as f:
        f.write("total_ms,calls,avg_us,pct,kernel\n")
        for name, n, ms, us in rows:
            f.write(f"{ms:.3f},{n},{us:.1f},{100*ms/max(total_ms,1e-9):.1f},"
                    f"\"{name[:140]}\"\n")
    print(f"wrote {out_path}: {len(rows)} kernels, total {total_ms:.2f} ms")


if __name__ == "__main__":
    db, out = sys.argv[1], sys.argv[2]
    summarize(db, out, int(sys.argv[3]) if len(sys.argv) > 3 else 60)
