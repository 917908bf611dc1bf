#!/usr/bin/env python3
"""Summarize a rocprofv3 SQLite result db into a kernel-time table
(markdown), optionally restricted to the last --window-ms of the timeline
(steady state, past MIOpen find / warmup)."""
import argparse
import sqlite3
from pathlib import Path


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("--out", default=None)
    ap.add_argument("--window-ms", type=float, default=None)
    ap.add_argument("--top", type=int, default=40)
    args = ap.parse_args()

    conn = sqlite3.connect(args.db)
    cur = conn.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))

    (t0, t1), = cur.execute(f"SELECT MIN(start), MAX(end) FROM {disp}")
    where = ""
    if args.window_ms:
        where = f"WHERE d.start > {t1 - int(args.window_ms * 1e6)}"
    rows = list(
        cur.execute(
            f"""SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6,
                       AVG(d.end-d.start)/1e3
                FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id {where}
                GROUP BY s.display_name ORDER BY 3 DESC"""
        )
    )
    tot = sum(r[2] for r in rows)
    lines = [
        f"# Kernel-time summary: {Path(args.db).name}",
        "",
        f"- timeline span: {(t1 - t0) / 1e9:.2f} s"
        + (f"; window: last {args.window_ms} ms" if args.window_ms else ""),
        f"- busy kernel time in window: {tot:.1f} ms",
        "",
        "| kernel | calls | total ms | avg us | % |",
        "|---|---|---|---|---|",
    ]
    for name, cnt, ms, us in rows[: args.top]:
        lines.append(f"| `{name[:80]}` | {cnt} | {ms:.2f} | {us:.1f} | {100 * ms / tot:.1f} |")
    text = "\n".join(lines) + "\n"
    if args.out:
        Path(args.out).write_text(text)
        print(f"wrote {args.out}")
    else:
        print(text)


if __name__ == "__main__":
    main()
