"""Summarize a rocprofv3 capture into a committed profile report.

Handles both output formats:
  - kernel_stats.csv (rocprofv3 --stats csv output)
  - *_results.db (the sqlite rocpd database some rocprofv3 builds emit
    instead; stats are aggregated here, optionally restricted to the
    last N milliseconds to exclude init)

Usage:
  python tools/summarize_profile.py <prof_dir> <out.md> "context" [last_ms]
"""

import csv
import glob
import sys


def _rows_from_csv(path):
    rows = list(csv.DictReader(open(path)))
    return [(r["Name"], int(r["Calls"]), float(r["TotalDurationNs"]),
             float(r["AverageNs"])) for r in rows]


def _rows_from_db(path, last_ms=None):
    import sqlite3
    db = sqlite3.connect(path)
    tabs = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = [t for t in tabs if t.startswith("rocpd_metadata_")][0] \
        .split("rocpd_metadata_")[-1]
    cond = ""
    if last_ms:
        t1 = db.execute(
            f"SELECT MAX(end) FROM rocpd_kernel_dispatch_{sfx}").fetchone()[0]
        cond = f"WHERE d.start >= {t1 - int(last_ms * 1e6)}"
    q = f"""SELECT s.display_name, COUNT(*), SUM(d.end-d.start),
                   AVG(d.end-d.start)
            FROM rocpd_kernel_dispatch_{sfx} d
            JOIN rocpd_info_kernel_symbol_{sfx} s ON d.kernel_id = s.id
            {cond} GROUP BY 1"""
    return [tuple(r) for r in db.execute(q)]


def main(prof_dir, out_md, context="", last_ms=None):
    csvs = glob.glob(f"{prof_dir}/**/*kernel_stats.csv", recursive=True)
    dbs = glob.glob(f"{prof_dir}/**/*results.db", recursive=True)
    if csvs:
        rows = _rows_from_csv(csvs[0])
        src = csvs[0]
    else:
        assert dbs, f"no kernel_stats.csv or results.db under {prof_dir}"
        rows = _rows_from_db(dbs[0], last_ms)
        src = dbs[0]
    rows.sort(key=lambda r: -r[2])
    total = sum(r[2] for r in rows)
    with open(out_md, "w") as f:
        f.write(f"# Kernel profile: {context}\n\n")
        f.write(f"Source: rocprofv3 on MI355X ({src.split('/')[-1]}"
                f"{f', last {last_ms} ms' if last_ms else ''})\n\n")
        f.write(f"Total kernel time: {total / 1e6:.1f} ms, "
                f"{len(rows)} distinct kernels\n\n")
        f.write("| kernel | % | calls | avg µs | total ms |\n"
                "|---|---|---|---|---|\n")
        for name, calls, tot, avg in rows[:25]:
            nm = name.split("(")[0].replace("void ", "")[:70]
            f.write(f"| `{nm}` | {tot / total * 100:.2f} | {calls} "
                    f"| {avg / 1e3:.1f} | {tot / 1e6:.2f} |\n")
    print(f"wrote {out_md}")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2],
         sys.argv[3] if len(sys.argv) > 3 else "",
         float(sys.argv[4]) if len(sys.argv) > 4 else None)
