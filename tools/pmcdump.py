"""Aggregate rocprofv3 rocpd PMC counters per kernel-name substring."""

import glob
import sqlite3
import sys


def main(path_glob, needle):
    for db in glob.glob(path_glob, recursive=True):
        con = sqlite3.connect(db)
        tabs = {t[0] for t in con.execute(
            "SELECT name FROM sqlite_master WHERE type='table'")}

        def tab(prefix):
            for t in tabs:
                if t.startswith(prefix):
                    return t
            return None

        pmc_ev = tab("rocpd_pmc_event")
        info_pmc = tab("rocpd_info_pmc")
        disp = tab("rocpd_kernel_dispatch")
        ksym = tab("rocpd_info_kernel_symbol")
        for t in (pmc_ev, info_pmc, disp, ksym):
            if t:
                cols = [c[1] for c in con.execute(f"PRAGMA table_info({t})")]
                print(t.split("_0000")[0], cols)
        if not (pmc_ev and info_pmc and disp and ksym):
            continue
        q = f"""
        SELECT p.name, SUM(e.value)
        FROM {pmc_ev} e
        JOIN {info_pmc} p ON p.id = e.pmc_id
        JOIN {disp} d ON d.event_id = e.event_id
        JOIN {ksym} s ON s.id = d.kernel_id
        WHERE s.display_name LIKE '%' || ? || '%'
        GROUP BY p.name
        """
        try:
            for name, val in con.execute(q, (needle,)):
                print(f"{name:28s} {val:.4e}")
        except Exception as exc:
            print("query failed:", exc)


if __name__ == "__main__":
    main(sys.argv[1] if len(sys.argv) > 1 else "/tmp/pm/**/*.db",
         sys.argv[2] if len(sys.argv) > 2 else "fp4")
