#!/usr/bin/env python3
"""Generates tests/golden/ec_golden.npz from the CPU oracle.

Run once and commit the output. These vectors SELF-pin the oracle (catch
regressions); byte-level parity vs compiled jerasure/isa-l remains an
external spot-check (see oracle/ec_ref.h "PARITY PINNING STATUS").
When a jerasure/ISA-L build or a real Ceph install becomes available,
run `tools/corpus_check.sh <their-corpus-dir>` (one command, maps plugin
and technique names) — and fold their chunks into tests/golden/corpus as
externally-generated KATs, which upgrades the pinning status from
"partial" for every technique it covers."""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__)))))
import oracle  # noqa: E402

out = {}
for tech in ("reed_sol_van", "cauchy", "jerasure_reed_sol_van"):
    for (k, m) in ((2, 1), (8, 3), (10, 4)):
        key = f"{tech}_k{k}m{m}"
        out[f"mat_{key}"] = oracle.matrix(tech, k, m)
        C = 256
        rng = np.random.default_rng(0xEC)
        data = [rng.integers(0, 256, C, dtype=np.uint8) for _ in range(k)]
        out[f"par_{key}"] = np.stack(oracle.encode(tech, k, m, data))

path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                    "ec_golden.npz")
np.savez_compressed(path, **out)
print(f"wrote {path} ({os.path.getsize(path)} bytes)")
