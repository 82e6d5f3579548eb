"""Summarize a rocprofv3 SQLite results db into a top-kernels CSV.

rocprofv3 on this image (ROCm 7.2) writes `*_results.db` (rocpd schema,
GUID-suffixed tables); the raw trace exceeds gpurun's 64 MiB copy-back,
so this runs ON the GPU box and emits only the per-kernel aggregate:

  python scripts/rocpd_top.py /tmp/prof/prof_results.db out.csv [N]
"""

import csv
import sqlite3
import sys


def main(db_path: str, out_csv: str, top_n: int = 40) -> None:
    con = sqlite3.connect(db_path)
    tabs = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    rows = list(con.execute(
        f"""SELECT s.display_name, COUNT(*) AS calls,
                   SUM(d.end - d.start) / 1e6 AS total_ms,
                   AVG(d.end - d.start) / 1e3 AS avg_us
            FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id
            GROUP BY s.display_name ORDER BY total_ms DESC LIMIT ?""",
        (top_n,),
    ))
    total = sum(r[2] for r in con.execute(
        f"SELECT s.display_name, 0, SUM(d.end - d.start) / 1e6 FROM {disp} d "
        f"JOIN {sym} s ON d.kernel_id = s.id GROUP BY s.display_name"))
    with open(out_csv, "w", newline="") as f:
        w = csv.writer(f)
        w.writerow(["kernel", "calls", "total_ms", "avg_us", "pct_gpu_busy"])
        for name, calls, ms, avg in rows:
            w.writerow([name[:120], calls, round(ms, 1), round(avg, 1),
                        round(100 * ms / total, 1)])
    print(f"wrote {out_csv}: {len(rows)} kernels, {total:.0f} ms GPU busy total")
    for name, calls, ms, avg in rows[:15]:
        print(f"{100*ms/total:5.1f}%  {ms:9.1f} ms  {calls:7d}x  {avg:8.1f} us  {name[:70]}")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2], int(sys.argv[3]) if len(sys.argv) > 3 else 40)
