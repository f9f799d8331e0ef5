"""Summarize a rocprofv3 --pmc rocpd SQLite DB: per-kernel counter sums.

FETCH_SIZE is reported in KB and needs the gfx950 x2 correction for wide
coalesced reads (MI355X_MICROARCH.md, HBM/rocprofv3 section); WRITE_SIZE is
KB with no correction. Prints one JSON object {kernel: {pmc: GB}}.
"""
import json
import sqlite3
import sys


def summarize(db_path):
    c = sqlite3.connect(db_path)
    t = c.execute("SELECT name FROM sqlite_master WHERE type='table' "
                  "AND name LIKE 'rocpd_pmc_event%'").fetchone()[0]
    s = t[len('rocpd_pmc_event'):]
    rows = c.execute(f"""
      SELECT k.display_name, p.name, SUM(e.value), COUNT(*)
      FROM rocpd_pmc_event{s} e
      JOIN rocpd_info_pmc{s} p ON e.pmc_id=p.id
      JOIN rocpd_kernel_dispatch{s} d ON e.event_id=d.event_id
      JOIN rocpd_info_kernel_symbol{s} k ON d.kernel_id=k.id
      GROUP BY 1,2 ORDER BY 3 DESC""").fetchall()
    out = {}
    for name, pmc, val, n in rows:
        name = name.split('(')[0]
        corr = 2.0 if pmc == "FETCH_SIZE" else 1.0
        gb = val * 1024.0 * corr / 1e9
        out.setdefault(name, {})[pmc] = round(gb, 4)
        out[name][pmc + "_dispatches"] = n
    return out


if __name__ == "__main__":
    print(json.dumps({p: summarize(p) for p in sys.argv[1:]}, indent=1))
