"""Summarize a rocprofv3 kernel_stats.csv into a committed profile report.

Usage: python tools/summarize_profile.py gpurun_out/profN profiles/NAME.md "context"
"""

import csv
import glob
import sys


def main(prof_dir, out_md, context=""):
    files = glob.glob(f"{prof_dir}/**/*kernel_stats.csv", recursive=True)
    assert files, f"no kernel_stats.csv under {prof_dir}"
    rows = list(csv.DictReader(open(files[0])))
    rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
    total = sum(float(r["TotalDurationNs"]) for r in rows)
    with open(out_md, "w") as f:
        f.write(f"# Kernel profile: {context}\n\n")
        f.write(f"Source: rocprofv3 --kernel-trace --stats on MI355X "
                f"({files[0].split('/')[-1]})\n\n")
        f.write(f"Total kernel time: {total / 1e6:.1f} ms, {len(rows)} distinct kernels\n\n")
        f.write("| kernel | % | calls | avg µs | total ms |\n|---|---|---|---|---|\n")
        for r in rows[:25]:
            name = r["Name"].split("(")[0].replace("void ", "")[:70]
            f.write(f"| `{name}` | {float(r['TotalDurationNs']) / total * 100:.2f} "
                    f"| {r['Calls']} | {float(r['AverageNs']) / 1e3:.1f} "
                    f"| {float(r['TotalDurationNs']) / 1e6:.2f} |\n")
    print(f"wrote {out_md}")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2], sys.argv[3] if len(sys.argv) > 3 else "")
