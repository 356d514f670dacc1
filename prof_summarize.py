"""Summarize a rocprofv3 rocpd SQLite db into a small text report.

Usage: python prof_summarize.py <dir-with-*_results.db> <out.txt>
Run ON the GPU box so only the summary travels back through gpurun_out.
"""

import glob
import sqlite3
import sys


def main() -> None:
    dbs = glob.glob(sys.argv[1] + "/**/*_results.db", recursive=True)
    out = sys.argv[2]
    with open(out, "w") as f:
        for path in dbs:
            db = sqlite3.connect(path)
            cur = db.cursor()
            tables = [
                r[0] for r in cur.execute(
                    "SELECT name FROM sqlite_master WHERE type='table'"
                )
            ]
            disp = [t for t in tables if t.startswith("rocpd_kernel_dispatch")]
            if not disp:
                continue
            u = disp[0].replace("rocpd_kernel_dispatch_", "")
            f.write(f"== {path} ==\n")
            q = (
                f"SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6,"
                f" AVG(d.end-d.start)/1e3, MAX(d.grid_size_x)"
                f" FROM rocpd_kernel_dispatch_{u} d"
                f" JOIN rocpd_info_kernel_symbol_{u} s ON d.kernel_id=s.id"
                f" GROUP BY s.display_name ORDER BY 3 DESC LIMIT 30"
            )
            for name, n, ms, us, gx in cur.execute(q):
                f.write(f"{ms:9.2f} ms n={n:6d} avg={us:8.1f}us "
                        f"maxgrid={gx:9d} {name[:90]}\n")
            tot = cur.execute(
                f"SELECT SUM(end-start)/1e6 FROM rocpd_kernel_dispatch_{u}"
            ).fetchone()[0]
            lo, hi = cur.execute(
                f"SELECT MIN(start), MAX(end) FROM rocpd_kernel_dispatch_{u}"
            ).fetchone()
            f.write(f"TOTAL {tot:.1f} ms kernels; span {(hi-lo)/1e6:.1f} ms\n\n")
            # PMC events, aggregated per kernel per counter
            npmc = cur.execute(
                f"SELECT COUNT(*) FROM rocpd_pmc_event_{u}"
            ).fetchone()[0]
            if npmc:
                cols = [r[1] for r in cur.execute(
                    f"PRAGMA table_info(rocpd_pmc_event_{u})")]
                f.write(f"PMC events: {npmc} (cols: {cols})\n")
                try:
                    q2 = (
                        f"SELECT s.display_name, i.name,"
                        f" SUM(p.value), COUNT(*)"
                        f" FROM rocpd_pmc_event_{u} p"
                        f" JOIN rocpd_info_pmc_{u} i ON p.pmc_id = i.id"
                        f" JOIN rocpd_kernel_dispatch_{u} d"
                        f"   ON p.event_id = d.event_id"
                        f" JOIN rocpd_info_kernel_symbol_{u} s"
                        f"   ON d.kernel_id = s.id"
                        f" GROUP BY s.display_name, i.name"
                        f" ORDER BY s.display_name, i.name"
                    )
                    for name, cname, val, n in cur.execute(q2):
                        f.write(f"  {cname:34s} {val:16.0f} n={n:6d} "
                                f"{name[:60]}\n")
                except Exception as e:
                    f.write(f"  pmc join failed: {e}\n")
            # optional: dump dispatch order (argv[3] = count, argv[4] = skip)
            if len(sys.argv) > 3:
                cnt = int(sys.argv[3])
                skip = int(sys.argv[4]) if len(sys.argv) > 4 else 0
                q3 = (
                    f"SELECT s.display_name, d.grid_size_x, d.grid_size_y,"
                    f" d.grid_size_z, (d.end-d.start)/1e3"
                    f" FROM rocpd_kernel_dispatch_{u} d"
                    f" JOIN rocpd_info_kernel_symbol_{u} s"
                    f"   ON d.kernel_id=s.id"
                    f" ORDER BY d.start LIMIT {cnt} OFFSET {skip}"
                )
                f.write("== dispatch order ==\n")
                for name, gx, gy, gz, us in cur.execute(q3):
                    f.write(f"{us:8.1f}us ({gx:>9},{gy:>3},{gz:>3}) "
                            f"{name[:80]}\n")


if __name__ == "__main__":
    main()
