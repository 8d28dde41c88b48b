"""Summarize a rocprofv3 rocpd SQLite database into markdown.

The stack's profiling subsystem (the reference has none —
its closest analog is nvidia-smi snapshots): every GPU payload can be run
under `rocprofv3 --kernel-trace --stats -d <dir>` and the resulting
*_results.db condensed into a committed, human-readable profile:

    python -m k3samd.utils.rocpd_summary gpurun_out/prof/triad_results.db
"""

from __future__ import annotations

import argparse
import sqlite3
from pathlib import Path


def shorten(name: str, limit: int = 100) -> str:
    # demangled template monsters -> keep the head
    head = name.split("(", 1)[0]
    return head if len(head) <= limit else head[: limit - 3] + "..."


def summarize(db_path: str | Path) -> str:
    con = sqlite3.connect(str(db_path))
    cur = con.cursor()
    lines = [f"# rocprofv3 kernel summary — {Path(db_path).name}", ""]

    try:
        agents = list(cur.execute(
            "SELECT type, name FROM rocpd_info_agent"))
        gpus = [a[1] for a in agents if "gpu" in str(a[0]).lower()]
        if gpus:
            lines.append(f"agents: {len(gpus)} GPU ({gpus[0]})")
            lines.append("")
    except sqlite3.Error:
        pass

    lines.append("| kernel | calls | total ms | avg us | % GPU time |")
    lines.append("|---|---|---|---|---|")
    try:
        rows = list(cur.execute(
            "SELECT name, total_calls, total_duration, average, percentage "
            "FROM top_kernels ORDER BY percentage DESC LIMIT 15"))
    except sqlite3.Error as e:
        con.close()
        raise SystemExit(
            f"{db_path}: no top_kernels view — was this db produced by "
            f"rocprofv3 --kernel-trace --stats? ({e})")
    for name, calls, total_us, avg_us, pct in rows:
        lines.append(f"| `{shorten(name)}` | {calls} | "
                     f"{total_us / 1e3:.2f} | {avg_us:.1f} | {pct:.1f} |")
    con.close()
    lines.append("")
    return "\n".join(lines)


def main(argv=None) -> int:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("db")
    p.add_argument("-o", "--output", help="write markdown here")
    args = p.parse_args(argv)
    md = summarize(args.db)
    if args.output:
        Path(args.output).write_text(md)
    else:
        print(md)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
