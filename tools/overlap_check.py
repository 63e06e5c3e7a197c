"""Comm/compute overlap evidence from a rocprofv3 rocpd DB:
how much of the RCCL all-reduce time runs CONCURRENTLY with backward
compute kernels.

    python tools/overlap_check.py <results.db>
"""
import json
import sqlite3
import sys


def main():
    db = sys.argv[1]
    c = sqlite3.connect(db)
    uuids = [r[0].split("rocpd_kernel_dispatch_")[1] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")]
    rows = []
    for uuid in uuids:
        cols = [r[1] for r in c.execute(
            f"PRAGMA table_info(rocpd_info_kernel_symbol_{uuid})")]
        name_expr = ("s.string" if "kernel_name_id" in cols
                     else "k.kernel_name")
        join = (f"JOIN rocpd_string_{uuid} s ON k.kernel_name_id = s.id"
                if "kernel_name_id" in cols else "")
        rows += list(c.execute(f"""
            SELECT {name_expr}, d.start, d.end
            FROM rocpd_kernel_dispatch_{uuid} d
            JOIN rocpd_info_kernel_symbol_{uuid} k ON d.kernel_id = k.id
            {join}"""))
    def is_comm(n):
        nl = n.lower()
        return ("ncclDevKernel" in n) or ("rccl" in nl) or \
               ("nccl" in nl and "rocclr" not in nl)
    comm = [(s, e) for n, s, e in rows if is_comm(n)]
    comp = [(s, e) for n, s, e in rows if not is_comm(n)]
    if not comm:
        print(json.dumps({"error": "no RCCL kernels in trace"}))
        return
    # merge compute intervals
    comp.sort()
    merged = []
    for s, e in comp:
        if merged and s <= merged[-1][1]:
            merged[-1][1] = max(merged[-1][1], e)
        else:
            merged.append([s, e])
    total_comm = sum(e - s for s, e in comm)
    overl = 0
    import bisect
    starts = [m[0] for m in merged]
    for s, e in comm:
        i = bisect.bisect_right(starts, e) - 1
        while i >= 0 and merged[i][1] > s:
            overl += max(0, min(e, merged[i][1]) - max(s, merged[i][0]))
            i -= 1
    print(json.dumps({
        "rccl_kernels": len(comm),
        "rccl_total_ms": total_comm / 1e6,
        "rccl_overlapped_with_compute_ms": overl / 1e6,
        "overlap_fraction": overl / total_comm,
        "compute_kernels": len(comp)}))


if __name__ == "__main__":
    main()
