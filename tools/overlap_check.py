#!/usr/bin/env python3
"""Quantify prefetch/compute overlap from a rocprofv3 kernel_trace CSV
(BASELINE config 5 evidence): how much of the ddstore gather kernels' wall
time runs concurrently with non-store (train-step) kernels."""
import csv
import sys


def main(path):
    rows = list(csv.DictReader(open(path)))
    store, other = [], []
    for r in rows:
        span = (int(r["Start_Timestamp"]), int(r["End_Timestamp"]))
        (store if "ddstore" in r["Kernel_Name"] else other).append(span)

    def merge(spans):
        spans = sorted(spans)
        out = []
        for s, e in spans:
            if out and s <= out[-1][1]:
                out[-1] = (out[-1][0], max(out[-1][1], e))
            else:
                out.append((s, e))
        return out

    def total(spans):
        return sum(e - s for s, e in spans)

    m_other = merge(other)

    def overlap_one(span, merged):
        s, e = span
        ov = 0
        for ms, me in merged:
            lo, hi = max(s, ms), min(e, me)
            if lo < hi:
                ov += hi - lo
        return ov

    st_total = total(merge(store))
    ov = sum(overlap_one(sp, m_other) for sp in merge(store))
    print(f"store-kernel busy time: {st_total/1e6:.2f} ms "
          f"({len(store)} dispatches)")
    print(f"  of which overlapped with train-step kernels: {ov/1e6:.2f} ms "
          f"({100*ov/max(st_total,1):.1f}%)")
    print(f"train-step kernel busy time: {total(m_other)/1e6:.2f} ms "
          f"({len(other)} dispatches)")
    streams = {r["Stream_Id"] for r in rows if "ddstore" in r["Kernel_Name"]}
    print(f"store kernels ran on stream(s): {sorted(streams)}; "
          f"train streams: {sorted({r['Stream_Id'] for r in rows} - streams)}")


if __name__ == "__main__":
    main(sys.argv[1])
