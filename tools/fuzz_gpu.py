"""Randomized GPU-vs-oracle fuzz sweep: random technique, (k, m), chunk
size (16 B .. 4 MiB, any multiple of 16), erasure mask, zeros-chunks and
accumulated decode patterns, for --seconds wall time. Any mismatch
prints the full reproducer tuple. Exit 0 = clean."""
import argparse
import time

import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=240)
    ap.add_argument("--seed", type=lambda x: int(x, 0), default=0xF0)
    args = ap.parse_args()

    import ceph_amd
    import oracle

    rng = np.random.default_rng(args.seed)
    stop = time.monotonic() + args.seconds
    cases = 0
    # reed_sol_r6_op is a harness-level alias of the isa RS-van matrix at
    # m=2 (tested in test_harness*); the C-ABI exposes the 5 base ids
    techs = ["reed_sol_van", "cauchy", "jerasure_reed_sol_van",
             "jerasure_reed_sol_van_w16", "cauchy_orig", "cauchy_good"]
    BITM = ("cauchy_orig", "cauchy_good")
    while time.monotonic() < stop:
        tech = techs[rng.integers(0, len(techs))]
        if tech in BITM:
            k, m = int(rng.integers(2, 13)), int(rng.integers(1, 5))
            if tech == "cauchy_good" and m == 2:
                m = 3  # m=2 refused (cbest tables unsourceable)
        else:
            k, m = int(rng.integers(2, 21)), int(rng.integers(1, 5))
        n = k + m
        if tech in BITM:
            pkt = int(rng.choice([512, 2048]))
            sw = 8 * pkt
            C = sw * int(rng.integers(1, 9))
        else:
            C = 16 * int(rng.integers(1, 4097))
            pkt = 2048
        repro = (tech, k, m, C, pkt)
        try:
            ctx = ceph_amd.EcContext(k, m, tech, device=0, packetsize=pkt)
        except ceph_amd.EcError as e:
            raise SystemExit(f"create failed {repro}: {e}")
        try:
            data = [None if rng.random() < 0.05 else
                    rng.integers(0, 256, C, dtype=np.uint8)
                    for _ in range(k)]
            full = [np.zeros(C, np.uint8) if d is None else d for d in data]
            if tech in BITM:
                want = oracle.bitmatrix_encode(k, m, full, pkt,
                                               technique=tech)
            elif tech == "jerasure_reed_sol_van_w16":
                want = oracle.encode_w16(k, m, full)
            else:
                want = oracle.encode(tech, k, m, full)
            got = ctx.encode_chunks(data, chunk_bytes=C)
            for j in range(m):
                assert np.array_equal(got[j], want[j]), ("enc", repro, j)
            ne = int(rng.integers(1, m + 1))
            er = sorted(rng.choice(n, size=ne, replace=False).tolist())
            chunks = [d.copy() for d in full] + [p.copy() for p in want]
            ref = full + want
            present = [i not in er for i in range(n)]
            for e in er:
                chunks[e][:] = 0
            ctx.decode_chunks(chunks, present)
            for i in range(n):
                assert np.array_equal(chunks[i], ref[i]), ("dec", repro,
                                                           er, i)
            cases += 1
        finally:
            ctx.close()
    print(f"FUZZ OK: {cases} randomized cases clean")


if __name__ == "__main__":
    main()
