#!/usr/bin/env python3
"""Data-loader throughput / memory-leak benchmark CLI.

Behavior contract: reference /root/reference/flaxdiff/data/benchmark_decord.py
(psutil RSS tracking over a long iteration) + training.py --dataset_test
(:175-176, :303-307 — iterate the loader 2000 steps for throughput/leak
checking).

    python -m flaxdiff_amd.data.benchmark_loader --dataset synthetic-64 \
        --batches 500 --batch_size 64
"""
from __future__ import annotations

import argparse
import json
import time


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--dataset", type=str, default="synthetic-64")
    ap.add_argument("--batches", type=int, default=500)
    ap.add_argument("--batch_size", type=int, default=64)
    ap.add_argument("--workers", type=int, default=4)
    ap.add_argument("--report_every", type=int, default=100)
    args = ap.parse_args(argv)

    import psutil
    from .dataloaders import get_dataset

    proc = psutil.Process()
    loader = get_dataset(args.dataset, global_batch_size=args.batch_size,
                         worker_count=args.workers,
                         num_samples=max(args.batches * args.batch_size, 1000))
    it = iter(loader)
    rss0 = proc.memory_info().rss / 1e6
    t0 = time.time()
    rss_track = []
    n = 0
    while n < args.batches:
        try:
            next(it)
        except StopIteration:
            it = iter(loader)
            continue
        n += 1
        if n % args.report_every == 0:
            rss = proc.memory_info().rss / 1e6
            rss_track.append(rss)
            dt = time.time() - t0
            print(f"{n:5d} batches  {n / dt:7.1f} b/s  "
                  f"{n * args.batch_size / dt:9.1f} img/s  RSS {rss:.0f} MB")
    dt = time.time() - t0
    leak = (rss_track[-1] - rss_track[0]) if len(rss_track) >= 2 else 0.0
    print(json.dumps({
        "batches_per_sec": n / dt, "images_per_sec": n * args.batch_size / dt,
        "rss_start_mb": rss0, "rss_end_mb": proc.memory_info().rss / 1e6,
        "rss_drift_mb": leak,
    }))


if __name__ == "__main__":
    main()
