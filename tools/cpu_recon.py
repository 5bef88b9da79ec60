#!/usr/bin/env python3
"""cpu_baseline reconciliation probe: why did the
in-round encode-only probe measure 224 GiB/s @64 threads while bench.py's
cpu_baseline reported 58 GiB/s encode+decode at the same thread count?

Controlled sweep on the GPU-box host: {encode-only, encode+decode} x
{batch size} x {threads}, plus a pure memory-bandwidth leg (numpy copy)
to establish the host DRAM ceiling. Prints one JSON line per cell."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np  # noqa: E402

import oracle  # noqa: E402

K, M, C = 8, 3, 1 << 20
GIB = 1 << 30


def set_threads(n):
    try:
        import ctypes
        for lib in (oracle._cpu, ):
            try:
                lib.omp_set_num_threads(ctypes.c_int(n))
                return True
            except AttributeError:
                pass
    except Exception:
        pass
    return False


def mem_bw_probe():
    a = np.empty(1 << 30, dtype=np.uint8)
    a[:] = 1
    b = np.empty_like(a)
    t0 = time.perf_counter()
    reps = 4
    for _ in range(reps):
        np.copyto(b, a)
    dt = time.perf_counter() - t0
    return round(reps * 2 * a.nbytes / GIB / dt, 1)  # read+write GiB/s


def run_cell(stripes, threads, do_decode, secs=6.0):
    set_threads(threads)
    batch = np.empty(stripes * (K + M) * C, dtype=np.uint8)
    if os.environ.get("RECON_TOUCH", "1") == "1":
        oracle.cpu_first_touch(batch)
    batch[:] = np.frombuffer(os.urandom(1 << 20), np.uint8).repeat(
        (batch.nbytes + (1 << 20) - 1) // (1 << 20))[:batch.nbytes]
    present = np.ones(K + M, np.uint8)
    present[[4, 8, 10]] = 0
    oracle.cpu_encode_batch("reed_sol_van", K, M, batch, stripes, C)  # warm
    t0 = time.perf_counter()
    iters = 0
    while time.perf_counter() - t0 < secs:
        oracle.cpu_encode_batch("reed_sol_van", K, M, batch, stripes, C)
        if do_decode:
            oracle.cpu_decode_batch("reed_sol_van", K, M, batch, present,
                                    stripes, C)
        iters += 1
    dt = time.perf_counter() - t0
    legs = 2 if do_decode else 1
    gibs = iters * legs * K * C * stripes / GIB / dt
    print(json.dumps({
        "stripes": stripes, "batch_mib": batch.nbytes >> 20,
        "threads": threads, "decode": do_decode, "iters": iters,
        "gibs": round(gibs, 1)}), flush=True)
    return gibs


def main():
    hw = len(os.sched_getaffinity(0))
    print(json.dumps({"hw_threads": hw, "numpy_copy_gibs": mem_bw_probe()}),
          flush=True)
    for stripes in (186, 16):
        for threads in (32, 64):
            run_cell(stripes, threads, do_decode=False)
            run_cell(stripes, threads, do_decode=True)


if __name__ == "__main__":
    main()
