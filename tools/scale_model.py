#!/usr/bin/env python3
"""Analytic weak-scaling model for the flagship fetch bench -- the expected
shape of SCALE_rNN.json from measured single-GPU rooflines and the xGMI
topology (7 links x ~153 GB/s per GPU, every pair directly connected).

Model per GPU per step (B rows of row_bytes, fraction (N-1)/N remote):
  local bytes:  B*row_bytes/N read at the measured local gather rate
  remote bytes: B*row_bytes*(N-1)/N read over (N-1) links in parallel
                -> bounded by (N-1) * link_bw, capped by the 7-link aggregate
  write bytes:  B*out_bytes to local HBM (concurrent with reads; modeled via
                the measured r+w local rate)
The read and write streams share the kernel, so step time =
  max(remote_time, local_rw_time) with remote overlapping local.

Usage: python tools/scale_model.py [--link-gbps 153] [--local-rw-tbps 4.7]
"""
import argparse


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=262144)
    p.add_argument("--row-bytes", type=int, default=512)
    p.add_argument("--out-bytes", type=int, default=256)   # bf16 out
    p.add_argument("--link-gbps", type=float, default=153.0)
    p.add_argument("--local-rw-tbps", type=float, default=4.7,
                   help="measured local random-row gather r+w rate")
    p.add_argument("--n1-us", type=float, default=44.5,
                   help="measured N=1 step time (anchors overheads)")
    args = p.parse_args()

    B, rb, ob = args.batch, args.row_bytes, args.out_bytes
    local_rate = args.local_rw_tbps * 1e12
    n1_model = B * (rb + ob) / local_rate * 1e6
    overhead_us = max(args.n1_us - n1_model, 0.0)

    print(f"N=1 anchor: model {n1_model:.1f} us + overhead {overhead_us:.1f} us"
          f" = measured {args.n1_us:.1f} us")
    print(f"{'N':>2} {'remote_us':>10} {'local_us':>9} {'step_us':>8} "
          f"{'perGPU_Gs':>10} {'aggregate_Gs':>13} {'weak_eff':>8}")
    n1_sps = B / (args.n1_us * 1e-6)
    for n in (1, 2, 4, 8):
        remote_bytes = B * rb * (n - 1) / n
        nlinks = max(n - 1, 1)
        link_agg = min(nlinks, 7) * args.link_gbps * 1e9
        remote_us = remote_bytes / link_agg * 1e6 if n > 1 else 0.0
        local_bytes = B * rb / n + B * ob
        local_us = local_bytes / local_rate * 1e6
        step = max(remote_us, local_us) + overhead_us
        sps = B / (step * 1e-6)
        print(f"{n:>2} {remote_us:>10.1f} {local_us:>9.1f} {step:>8.1f} "
              f"{sps/1e9:>10.2f} {n*sps/1e9:>13.2f} {sps/n1_sps:>8.1%}")
    print("\nReading SCALE_rNN.json: 'weak_eff' is the driver's efficiency "
          "number; the drop is interconnect physics (remote reads ride "
          "xGMI at ~1/30 of local HBM bandwidth), not software overhead -- "
          "per-GPU samples/s at the modeled level means the IPC/xGMI path "
          "is at its roofline.")


if __name__ == "__main__":
    main()
