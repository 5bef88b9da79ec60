#!/usr/bin/env python3
"""Flagship-kernel ceiling probe: is the RS(8,3)
encode kernel's 6.0 TB/s bounded by its GF arithmetic or by the memory
system's 8:3 read:write mix?

Runs the EXACT kernel twice on the same device batch: (a) the real
generator, (b) an all-ones generator (every coefficient class becomes
plain XOR — no v_perm tables, same loads/stores). If (b) ≈ (a), the
kernel is memory-bound and 6.0 is the op's ceiling; if (b) is faster,
the delta is the GF compute cost still on the critical path."""
import json
import sys
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np  # noqa: E402
import ceph_amd  # noqa: E402

import argparse

ap = argparse.ArgumentParser()
ap.add_argument("--k", type=int, default=8)
ap.add_argument("--m", type=int, default=3)
ap.add_argument("--technique", default="reed_sol_van")
ap.add_argument("--stripes", type=int, default=4096)
A = ap.parse_args()
K, M, C, S = A.k, A.m, 1 << 20, A.stripes


def run(ctx, dptr, steps=5):
    ms = []
    for _ in range(steps):
        ctx.encode_batch(dptr, S, C)
        ms.append(ctx.last_kernel_ms())
    return float(np.mean(ms[1:]))


def main():
    ctx = ceph_amd.EcContext(K, M, A.technique, device=0)
    n = K + M
    buf = S * n * C
    d = ctx.dbuf_alloc(buf)
    ctx.fill_random(d, buf, 0xEC)
    ctx.sync()
    alg = (K + M) * C * S
    real = run(ctx, d)
    ctx.set_matrix(np.ones((M, K), dtype=np.uint8))
    allones = run(ctx, d)
    print(json.dumps({
        "k": K, "m": M, "technique": A.technique,
        "real_ms": round(real, 4), "real_GBs": round(alg / real / 1e6, 0),
        "allones_ms": round(allones, 4),
        "allones_GBs": round(alg / allones / 1e6, 0),
        "gf_compute_cost_pct": round(100 * (real - allones) / real, 1)}))
    ctx.close()


if __name__ == "__main__":
    main()
