"""Find the best inference batch size by measuring latency/throughput.

Capability parity with the reference's src/batchsizefinder.h (a
latency-model search over batch sizes, scoring latency against
log-throughput; used for sizing dynamic-batching services). Ours measures
on the current device with proper warmup/sync.
"""
import math
import time

import torch


def measure_latency(fn, batch_size, make_batch, iters=5, warmup=2):
    for _ in range(warmup):
        fn(make_batch(batch_size))
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn(make_batch(batch_size))
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def find(fn, make_batch, max_batch_size=1024, latency_target_ms=400.0, verbose=False):
    """Returns the batch size with the best latency/throughput score.

    score = latency/target - log(bs/latency): prefers high throughput until
    latency grows past the target scale (same shape as the reference's
    scoring, batchsizefinder.h:48-50).
    """
    results = {}

    def score(bs):
        if bs not in results:
            lat = measure_latency(fn, bs, make_batch)
            results[bs] = lat
        lat = results[bs]
        return lat * 1000.0 / latency_target_ms - math.log(bs / lat)

    # coarse pass: powers of two
    candidates = []
    bs = 1
    while bs <= max_batch_size:
        candidates.append(bs)
        bs *= 2
    scored = sorted(candidates, key=score)
    best = scored[0]
    # refine around the winner
    lo, hi = max(1, best // 2), min(max_batch_size, best * 2)
    step = max(1, (hi - lo) // 8)
    for bs in range(lo, hi + 1, step):
        score(bs)
    best = min(results, key=lambda b: score(b))
    if verbose:
        for b in sorted(results):
            lat = results[b]
            print(
                "bs %5d  latency %8.3f ms  throughput %10.1f/s  score %8.3f"
                % (b, lat * 1e3, b / lat, score(b))
            )
    return best, results
