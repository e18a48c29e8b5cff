"""Aggregate the tail window of a rocprofv3 kernel_trace.csv into per-kernel
totals (steady-state view). Usage: agg_trace.py trace.csv window_ms out.txt"""
import csv
import sys


def main():
    path, window_ms, out = sys.argv[1], float(sys.argv[2]), sys.argv[3]
    rows = []
    with open(path) as f:
        for r in csv.DictReader(f):
            try:
                s = int(r["Start_Timestamp"])
                e = int(r["End_Timestamp"])
            except (KeyError, ValueError):
                continue
            rows.append((s, e, r.get("Kernel_Name", "?")))
    if not rows:
        raise SystemExit("no rows in " + path)
    mx = max(e for _, e, _ in rows)
    w0 = mx - window_ms * 1e6
    agg = {}
    for s, e, name in rows:
        if s >= w0:
            c, t = agg.get(name, (0, 0.0))
            agg[name] = (c + 1, t + (e - s) / 1e6)
    with open(out, "w") as f:
        total = sum(v[1] for v in agg.values())
        ndisp = sum(v[0] for v in agg.values())
        f.write("window %.0f ms: busy %.1f ms, %d dispatches\n" %
                (window_ms, total, ndisp))
        for name, (c, t) in sorted(agg.items(), key=lambda kv: -kv[1][1])[:30]:
            f.write("%8.2f ms %6d  %s\n" % (t, c, name[:100]))


if __name__ == "__main__":
    main()
