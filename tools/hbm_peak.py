"""Measure achievable HBM bandwidth on this box ("per-GPU HBM
peak, measured on the box with a triad kernel and stated").

Three streaming patterns over large fp32 buffers, timed with HIP events:
  copy   y[i] = x[i]            (2 x 4 bytes/elem)
  scale  y[i] = a*x[i]          (2 x 4)
  triad  y[i] = x[i] + a*z[i]   (3 x 4)
Reported GB/s = moved bytes / best-of-reps time. Prints one JSON line.

Usage: python tools/hbm_peak.py [--gib 8] [--reps 20]
"""
import argparse
import json

import torch


def _timed(fn, reps):
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    best = float('inf')
    for _ in range(reps):
        s.record()
        fn()
        e.record()
        torch.cuda.synchronize()
        best = min(best, s.elapsed_time(e))
    return best  # ms


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--gib', type=float, default=8.0,
                    help='size of EACH operand buffer in GiB')
    ap.add_argument('--reps', type=int, default=20)
    args = ap.parse_args()

    n = int(args.gib * (1 << 30) / 4)
    dev = torch.device('cuda:0')
    x = torch.rand(n, device=dev)
    z = torch.rand(n, device=dev)
    y = torch.empty(n, device=dev)
    for _ in range(3):  # warmup
        y.copy_(x)
    torch.cuda.synchronize()

    out = {'device': torch.cuda.get_device_name(0), 'elems': n,
           'buffer_gib': args.gib, 'reps': args.reps}
    ms = _timed(lambda: y.copy_(x), args.reps)
    out['copy_gbs'] = round(2 * 4 * n / (ms * 1e-3) / 1e9, 1)
    ms = _timed(lambda: torch.mul(x, 1.0001, out=y), args.reps)
    out['scale_gbs'] = round(2 * 4 * n / (ms * 1e-3) / 1e9, 1)
    ms = _timed(lambda: torch.add(x, z, alpha=1.0001, out=y), args.reps)
    out['triad_gbs'] = round(3 * 4 * n / (ms * 1e-3) / 1e9, 1)
    out['peak_achievable_gbs'] = max(out['copy_gbs'], out['scale_gbs'],
                                     out['triad_gbs'])
    print(json.dumps(out))


if __name__ == '__main__':
    main()
