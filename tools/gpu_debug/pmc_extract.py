"""Extract per-kernel PMC counter sums from a rocprofv3 rocpd db.

Usage: python pmc_extract.py <dir-with-*_results.db> <out.txt>
"""
import glob
import sqlite3
import sys


def main():
    dbs = glob.glob(sys.argv[1] + "/**/*_results.db", recursive=True)
    with open(sys.argv[2], "w") as out:
        for path in dbs:
            db = sqlite3.connect(path)
            cur = db.cursor()
            tables = [r[0] for r in cur.execute(
                "SELECT name FROM sqlite_master WHERE type='table'")]
            disp = [t for t in tables
                    if t.startswith("rocpd_kernel_dispatch")]
            if not disp:
                continue
            u = disp[0].replace("rocpd_kernel_dispatch_", "")
            out.write(f"== {path} ==\n")
            cols = {t: [r[1] for r in cur.execute(
                f"PRAGMA table_info({t})")] for t in tables}
            pmc_ev = f"rocpd_pmc_event_{u}"
            out.write(f"# pmc_event cols: {cols.get(pmc_ev)}\n")
            out.write(f"# info_pmc cols: "
                      f"{cols.get(f'rocpd_info_pmc_{u}')}\n")
            # rocpd_pmc_event: (event_id, pmc_id, value); events link to
            # dispatches through rocpd_event / kernel dispatch event_id
            try:
                q = f"""
                SELECT ks.display_name, ip.name, SUM(pe.value),
                       COUNT(*)
                FROM rocpd_pmc_event_{u} pe
                JOIN rocpd_info_pmc_{u} ip ON pe.pmc_id = ip.id
                JOIN rocpd_kernel_dispatch_{u} d
                     ON pe.event_id = d.event_id
                JOIN rocpd_info_kernel_symbol_{u} ks
                     ON d.kernel_id = ks.id
                GROUP BY 1, 2 ORDER BY 1, 2"""
                for name, cn, v, n in cur.execute(q):
                    out.write(f"  {cn:28s} {v:18.0f} n={n:6d} "
                              f"{name[:80]}\n")
            except Exception as e:
                out.write(f"JOIN-ERR {e}\n")
                # dump schemas of candidate tables to adapt offline
                for t in tables:
                    out.write(f"{t}: {cols[t]}\n")


if __name__ == "__main__":
    main()
