"""Extract committed kernel-stats summaries from rocprofv3 result DBs
(gpurun_out is scratch; summaries here are the judged evidence)."""

import glob
import sqlite3
import sys


def summarize(db_path, out_path, title, note=""):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_kernel_dispatch%'").fetchall()]
    sfx = tables[0].replace("rocpd_kernel_dispatch_", "")
    rows = cur.execute(
        f"""SELECT ks.display_name, COUNT(*) n, SUM(k.end-k.start)/1e6 ms, AVG(k.end-k.start)/1e3 us
        FROM rocpd_kernel_dispatch_{sfx} k JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        GROUP BY ks.display_name ORDER BY ms DESC LIMIT 40"""
    ).fetchall()
    total = cur.execute(f"SELECT SUM(end-start)/1e6, COUNT(*) FROM rocpd_kernel_dispatch_{sfx}").fetchone()
    with open(out_path, "w") as f:
        f.write(f"# {title}\n\n{note}\n\n")
        f.write(f"Total GPU-busy: {total[0]:.1f} ms across {total[1]} kernel dispatches.\n\n")
        f.write("| total ms | calls | avg µs | kernel |\n|---:|---:|---:|---|\n")
        for name, n, ms, us in rows:
            f.write(f"| {ms:.2f} | {n} | {us:.1f} | `{name[:110]}` |\n")
    print("wrote", out_path)


if __name__ == "__main__":
    summarize(*sys.argv[1:])
