"""Summarize a rocprofv3 rocpd results database into a per-kernel counter
table (CSV to stdout). Usage:

    python tools/pmc_summarize.py results.db [name-filter ...]

Counters are averaged per dispatch (and across counter instances, i.e.
shader engines / TCC channels — multiply by the instance count for chip
totals where that matters)."""

import sqlite3
import sys


def main():
    db = sys.argv[1]
    filters = [f.lower() for f in sys.argv[2:]]
    con = sqlite3.connect(db)
    cur = con.cursor()
    sfx_rows = [
        r[0] for r in cur.execute(
            "SELECT name FROM sqlite_master WHERE type='table' "
            "AND name LIKE 'rocpd_kernel_dispatch%'"
        )
    ]
    if not sfx_rows:
        print("no kernel dispatch table found", file=sys.stderr)
        sys.exit(1)
    sfx = sfx_rows[0].replace("rocpd_kernel_dispatch", "")
    q = f"""
    SELECT ks.display_name, p.name, COUNT(*), AVG(pe.value),
           AVG(kd.end - kd.start)
    FROM rocpd_pmc_event{sfx} pe
    JOIN rocpd_kernel_dispatch{sfx} kd ON pe.event_id = kd.event_id
    JOIN rocpd_info_kernel_symbol{sfx} ks ON kd.kernel_id = ks.id
    JOIN rocpd_info_pmc{sfx} p ON pe.pmc_id = p.id
    GROUP BY ks.display_name, p.name
    """
    # per-kernel: dispatch count, mean duration, counters
    table = {}
    for name, counter, n, val, dur in cur.execute(q):
        short = name.split("<")[0].split("(")[0].strip()
        if filters and not any(f in short.lower() for f in filters):
            continue
        ent = table.setdefault(short, {"n": n, "dur_us": dur / 1e3})
        ent[counter] = val
    if not table:
        print("no kernels matched", file=sys.stderr)
        sys.exit(1)
    counters = sorted({c for e in table.values() for c in e if c not in ("n", "dur_us")})
    print("kernel,n_samples,avg_dur_us," + ",".join(counters))
    for name, e in sorted(table.items(), key=lambda kv: -kv[1]["dur_us"]):
        row = [name, str(e["n"]), f"{e['dur_us']:.2f}"]
        row += [f"{e.get(c, 0):.1f}" for c in counters]
        print(",".join(row))


if __name__ == "__main__":
    main()
