#!/bin/bash
# Round-2 extended robustness: multi-seed fuzz, long soak, plus a
# bitmatrix single-erasure decode A/B (LDS kernel idles half the block
# at n_out=1; the v3 register kernel keeps every lane busy).
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

# A/B in separate processes (the knob is read once per process)
for reg in 0 1; do
  ECX_BITREG=$reg python - <<'PY'
import os, numpy as np, ceph_amd
k, m, C, S = 8, 3, 1 << 20, 4096
ctx = ceph_amd.EcContext(k, m, "cauchy_orig", device=0)
buf = S * (k + m) * C
d = ctx.dbuf_alloc(buf)
ctx.fill_random(d, buf, 0xEC)
ctx.sync()
ctx.encode_batch(d, S, C)
ctx.sync()
mask = ((1 << (k + m)) - 1) & ~(1 << 2)   # single data erasure
ms = []
for _ in range(6):
    ctx.decode_batch(d, S, C, mask)
    ms.append(ctx.last_kernel_ms())
alg = (k + 1) * C * S   # read k survivors + write 1
print({"ECX_BITREG": os.environ.get("ECX_BITREG"),
       "dec1_ms": round(float(np.mean(ms[1:])), 3),
       "dec1_GBs": round(alg / np.mean(ms[1:]) / 1e6, 0)})
ctx.close()
PY
done

timeout 500 python tools/fuzz_gpu.py --seconds 420 --seed 777 2>&1 | tail -1
timeout 500 python tools/soak.py --seconds 400 2>&1 | tail -3
