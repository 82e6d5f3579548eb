"""Summarize rocprofv3 --pmc results: per-kernel counter totals.

  python scripts/rocpd_pmc.py results.db [kernel-substring]

Schema-adaptive: discovers the GUID-suffixed rocpd tables, joins
pmc_event -> (event/dispatch) -> kernel symbol, prints counter sums per
kernel name.
"""

import sqlite3
import sys
from collections import defaultdict


def main(db_path: str, match: str = "") -> None:
    con = sqlite3.connect(db_path)
    tabs = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]

    def tab(prefix):
        return next((t for t in tabs if t.startswith(prefix)), None)

    t_pmc = tab("rocpd_pmc_event")
    t_info = tab("rocpd_info_pmc")
    t_disp = tab("rocpd_kernel_dispatch")
    t_sym = tab("rocpd_info_kernel_symbol")
    t_event = tab("rocpd_event")
    for name, t in [("pmc", t_pmc), ("info", t_info), ("disp", t_disp),
                    ("sym", t_sym), ("event", t_event)]:
        print(f"-- {name}: {t}")
        if t:
            cols = [d[1] for d in con.execute(f"PRAGMA table_info({t})")]
            print("   cols:", cols)

    names = {}
    if t_info:
        cols = [d[1] for d in con.execute(f"PRAGMA table_info({t_info})")]
        name_col = "name" if "name" in cols else cols[-1]
        for r in con.execute(f"SELECT id, {name_col} FROM {t_info}"):
            names[r[0]] = r[1]
        print("counters:", list(names.values()))

    # try the direct join: dispatch rows reference an event id column
    dcols = [d[1] for d in con.execute(f"PRAGMA table_info({t_disp})")]
    # dispatch ids and pmc event_ids may share the event table; attempt
    # pmc_event.event_id == kernel_dispatch.id first, then via rocpd_event
    joined = list(con.execute(
        f"""SELECT s.display_name, p.pmc_id, SUM(p.value)
            FROM {t_pmc} p JOIN {t_disp} d ON p.event_id = d.id
            JOIN {t_sym} s ON d.kernel_id = s.id
            GROUP BY s.display_name, p.pmc_id"""))
    if not joined and t_event:
        ecols = [d[1] for d in con.execute(f"PRAGMA table_info({t_event})")]
        print("event cols:", ecols)
        joined = list(con.execute(
            f"""SELECT s.display_name, p.pmc_id, SUM(p.value)
                FROM {t_pmc} p JOIN {t_event} e ON p.event_id = e.id
                JOIN {t_disp} d ON d.event_id = e.id
                JOIN {t_sym} s ON d.kernel_id = s.id
                GROUP BY s.display_name, p.pmc_id"""))
    per_kernel: dict = defaultdict(dict)
    for disp_name, pmc_id, total in joined:
        per_kernel[disp_name][names.get(pmc_id, pmc_id)] = total
    for k, counters in sorted(per_kernel.items()):
        if match and match not in k:
            continue
        print(f"\n== {k[:90]}")
        for cname, v in sorted(counters.items(), key=lambda x: str(x[0])):
            print(f"   {cname}: {v:.3e}")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2] if len(sys.argv) > 2 else "")
