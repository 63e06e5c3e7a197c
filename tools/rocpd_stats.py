"""Aggregate per-kernel time from a rocprofv3 rocpd sqlite DB
(rocprofv3 --kernel-trace --stats output format on ROCm 7.2).

    python tools/rocpd_stats.py <results.db> [steps]
"""
import sqlite3
import sys


def main():
    db = sys.argv[1]
    steps = float(sys.argv[2]) if len(sys.argv) > 2 else 1.0
    c = sqlite3.connect(db)
    uuids = [r[0].split("rocpd_kernel_dispatch_")[1] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")]
    agg = {}
    for uuid in uuids:
        cols = [r[1] for r in c.execute(
            f"PRAGMA table_info(rocpd_info_kernel_symbol_{uuid})")]
        if "kernel_name_id" in cols:
            q = f"""
            SELECT s.string, COUNT(*), SUM(d.end - d.start)
            FROM rocpd_kernel_dispatch_{uuid} d
            JOIN rocpd_info_kernel_symbol_{uuid} k ON d.kernel_id = k.id
            JOIN rocpd_string_{uuid} s ON k.kernel_name_id = s.id
            GROUP BY s.string
            """
        else:
            q = f"""
            SELECT k.kernel_name, COUNT(*), SUM(d.end - d.start)
            FROM rocpd_kernel_dispatch_{uuid} d
            JOIN rocpd_info_kernel_symbol_{uuid} k ON d.kernel_id = k.id
            GROUP BY k.kernel_name
            """
        for name, calls, ns in c.execute(q):
            e = agg.setdefault(name, [0, 0])
            e[0] += calls
            e[1] += ns or 0
    rows = sorted(((n, v[0], v[1]) for n, v in agg.items()),
                  key=lambda r: -r[2])
    total = sum(r[2] for r in rows)
    print(f"total kernel ns: {total:,} -> {total/1e6/steps:.2f} ms/step")
    print(f"{'kernel':64s} {'calls':>6s} {'ms/step':>8s} {'avg us':>7s} {'%':>5s}")
    for name, calls, ns in rows[:40]:
        nm = name.split("(")[0][:64]
        print(f"{nm:64s} {calls:6d} {ns/1e6/steps:8.3f} "
              f"{ns/1e3/calls:7.1f} {100.0*ns/total:5.1f}")


if __name__ == "__main__":
    main()
