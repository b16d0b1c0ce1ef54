"""Summarize a rocprofv3 kernel_stats.csv into a committed markdown table.

  python scripts/summarize_prof.py <kernel_stats.csv-or-glob> <steps> \
      <out.md> "<title>" ["<note>"]
"""
from __future__ import annotations

import csv
import glob
import sys


def main():
    pattern, steps, out_md, title = sys.argv[1:5]
    note = sys.argv[5] if len(sys.argv) > 5 else ""
    steps = int(steps)
    files = sorted(glob.glob(pattern))
    if not files:
        raise SystemExit(f"no files match {pattern}")
    rows = list(csv.DictReader(open(files[-1])))
    tot = sum(float(r["TotalDurationNs"]) for r in rows)
    lines = [f"# {title}", "",
             f"Source: `rocprofv3 --kernel-trace --stats` over {steps} "
             f"learner steps on a gpurun MI355X box (raw: `{files[-1]}`).",
             ""]
    if note:
        lines += [note, ""]
    lines += [f"Total GPU kernel time: **{tot/steps/1000:.1f} us/step**", "",
              "| us/step | calls/step | kernel |", "|---|---|---|"]
    for r in rows[:20]:
        name = r["Name"].split("(")[0].replace("|", "\\|")[:90]
        lines.append(f'| {float(r["TotalDurationNs"])/steps/1000:.1f} '
                     f'| {int(r["Calls"])//steps} | `{name}` |')
    open(out_md, "w").write("\n".join(lines) + "\n")
    print("wrote", out_md)


if __name__ == "__main__":
    main()
