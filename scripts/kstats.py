"""Summarize rocprofv3 kernel_stats csv files passed as globs."""
import csv
import glob
import sys


def main():
    for pat in sys.argv[1:]:
        files = sorted(glob.glob(pat, recursive=True))
        if not files:
            print(f"{pat}: no files")
            continue
        for fn in files:
            rows = list(csv.DictReader(open(fn)))
            if not rows:
                continue
            dur = ("TotalDurationNs" if "TotalDurationNs" in rows[0]
                   else "DurationNs")
            rows.sort(key=lambda r: -float(r.get(dur, 0)))
            tot = sum(float(r.get(dur, 0)) for r in rows)
            print("==", fn, f"total {tot/1e6:.1f} ms")
            for r in rows[:10]:
                d = float(r.get(dur, 0))
                calls = int(r.get("Calls", r.get("Count", 1)))
                print(f"  {d/1e6:8.2f}ms {100*d/max(tot,1):5.1f}% "
                      f"{calls:6d}x  {r.get('Name','')[:64]}")


if __name__ == "__main__":
    main()
