"""Summarise a rocprofv3 rocpd SQLite database (the default output of
ROCm 7.2's rocprofv3): per-kernel totals for the full run and the
trailing window, plus the native (gemscore/conv_mfma/conv_pw) share.

Usage: python tools/kstats_db.py <results.db> [--window-s 3.0] [-o out]
"""
import argparse
import sqlite3
import sys
from collections import defaultdict

NATIVE_MARKERS = ("conv_mfma", "conv_pw", "gemscore", "pw_", "halo_",
                  "bn_stats", "bn_apply", "bn_bwd", "maxpool", "avgpool",
                  "sgd_momentum", "bn_finalize", "addcat")


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
    for name in sorted(tot, key=lambda k: -tot[k])[:45]:
        out.write(
            f"{100*tot[name]/total:6.2f} {tot[name]/1e6:10.2f} "
            f"{cnt[name]:7d} {tot[name]/cnt[name]/1e3:9.1f}  {name[:110]}\n"
        )


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("--window-s", type=float, default=3.0)
    ap.add_argument("-o", "--out", default=None)
    a = ap.parse_args()
    con = sqlite3.connect(a.db)
    cur = con.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' AND "
        "name LIKE 'rocpd_kernel_dispatch%'")]
    rows = []
    for t in tabs:
        suf = t[len("rocpd_kernel_dispatch_"):]
        q = (f"SELECT s.display_name, d.start, d.end - d.start "
             f"FROM {t} d JOIN rocpd_info_kernel_symbol_{suf} s "
             f"ON d.kernel_id = s.id")
        for name, start, dur in cur.execute(q):
            rows.append((name, start, dur))
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
