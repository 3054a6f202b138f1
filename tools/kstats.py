"""Per-kernel time aggregation from a rocprofv3 rocpd database."""

import glob
import sys


def main(pattern):
    import sqlite3

    for db in glob.glob(pattern, recursive=True):
        con = sqlite3.connect(db)
        tabs = {t[0] for t in con.execute(
            "SELECT name FROM sqlite_master WHERE type='table'")}

        def tab(p):
            return next((t for t in tabs if t.startswith(p)), None)

        d, s = tab("rocpd_kernel_dispatch"), tab("rocpd_info_kernel_symbol")
        if not (d and s):
            continue
        q = (f"SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6, "
             f"AVG(d.end-d.start)/1e3 FROM {d} d JOIN {s} s ON s.id=d.kernel_id "
             f"GROUP BY s.display_name ORDER BY SUM(d.end-d.start) DESC LIMIT 14")
        for name, n, tot_ms, avg_us in con.execute(q):
            print(f"{tot_ms:9.2f} ms  n={n:4d}  avg={avg_us:9.1f} us  {name[:76]}")


if __name__ == "__main__":
    main(sys.argv[1] if len(sys.argv) > 1 else "/tmp/ks3/**/*.db")
