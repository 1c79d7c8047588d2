"""Summarize a rocprofv3 results.db: per-kernel totals + wall coverage.

Usage: python scripts/profile_summary.py <results.db|dir> [out.md]
"""
import glob
import os
import sqlite3
import sys


def find_db(path):
    if os.path.isfile(path):
        return path
    dbs = glob.glob(os.path.join(path, "**", "*results.db"), recursive=True)
    if not dbs:
        raise SystemExit(f"no results.db under {path}")
    return sorted(dbs)[-1]


def main():
    db_path = find_db(sys.argv[1])
    out = sys.argv[2] if len(sys.argv) > 2 else None
    c = sqlite3.connect(db_path)
    tables = [r[0] for r in c.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tables if t.startswith("rocpd_info_kernel_symbol"))
    rows = list(c.execute(f"""
        SELECT ks.display_name, COUNT(*) n, SUM(k.end-k.start)/1e6 ms,
               AVG(k.end-k.start)/1e3 us,
               MAX(ks.arch_vgpr_count), MAX(ks.accum_vgpr_count),
               MAX(ks.group_segment_size)
        FROM {disp} k JOIN {sym} ks ON k.kernel_id = ks.id
        GROUP BY ks.display_name ORDER BY ms DESC"""))
    span = c.execute(
        f"SELECT (MAX(end)-MIN(start))/1e6 FROM {disp}").fetchone()[0]
    total = sum(r[2] for r in rows)

    lines = [f"# Kernel profile: {os.path.basename(db_path)}", "",
             f"- kernel-busy total: {total:.1f} ms",
             f"- first-to-last-dispatch span: {span:.1f} ms "
             f"(busy fraction {total / span * 100:.1f}%)", "",
             "| time ms | % | calls | avg us | vgpr+acc | lds B | kernel |",
             "|---:|---:|---:|---:|---:|---:|---|"]
    for name, n, ms, us, vgpr, agpr, lds in rows:
        short = name.split("(")[0]
        lines.append(f"| {ms:.1f} | {ms / total * 100:.1f} | {n} | "
                     f"{us:.1f} | {vgpr}+{agpr} | {lds} | `{short[:70]}` |")
    text = "\n".join(lines) + "\n"
    print(text)
    if out:
        with open(out, "w") as f:
            f.write(text)


if __name__ == "__main__":
    main()
