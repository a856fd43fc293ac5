"""Summarise a rocprofv3 kernel trace: per-kernel totals for the whole
run and for the trailing time window (the timed bench steps), plus the
gemscore/conv_mfma/conv_pw native share.

Usage: python tools/kstats.py <kernel_trace.csv> [--window-s 2.0]
"""
import argparse
import csv
import sys
from collections import defaultdict

NATIVE_MARKERS = ("conv_mfma", "conv_pw", "gemscore", "pw_", "halo_",
                  "bn_stats", "bn_apply", "bn_bwd", "maxpool", "avgpool",
                  "sgd_momentum", "bn_finalize")


def summarize(rows, title, out):
    tot = defaultdict(float)
    cnt = defaultdict(int)
    for name, dur in rows:
        tot[name] += dur
        cnt[name] += 1
    total = sum(tot.values())
    native = sum(v for k, v in tot.items() if any(m in k for m in NATIVE_MARKERS))
    out.write(f"\n## {title}: kernel time {total/1e6:.1f} ms, "
              f"native share {100*native/max(total,1):.1f}%\n")
    out.write(f"{'%':>6} {'total_ms':>10} {'calls':>7} {'avg_us':>9}  name\n")
    for name in sorted(tot, key=lambda k: -tot[k])[:40]:
        out.write(
            f"{100*tot[name]/total:6.2f} {tot[name]/1e6:10.2f} "
            f"{cnt[name]:7d} {tot[name]/cnt[name]/1e3:9.1f}  {name[:100]}\n"
        )


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("trace")
    ap.add_argument("--window-s", type=float, default=2.0)
    ap.add_argument("-o", "--out", default=None)
    a = ap.parse_args()
    rows = []
    with open(a.trace) as f:
        r = csv.DictReader(f)
        cols = r.fieldnames
        kname = next(c for c in cols if "Kernel_Name" in c or c == "Name")
        start = next(c for c in cols if "Start" in c)
        end = next(c for c in cols if "End" in c)
        for row in r:
            s, e = float(row[start]), float(row[end])
            rows.append((row[kname], s, e - s))
    out = open(a.out, "w") if a.out else sys.stdout
    summarize([(n, d) for n, _, d in rows], "full run", out)
    tmax = max(s + d for _, s, d in rows)
    w0 = tmax - a.window_s * 1e9
    win = [(n, d) for n, s, d in rows if s >= w0]
    summarize(win, f"trailing {a.window_s}s window", out)
    if a.out:
        out.close()


if __name__ == "__main__":
    main()
