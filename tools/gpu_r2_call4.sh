#!/bin/bash
# Round-2 GPU call 4: bitmatrix convoy-breaking A/B — 128/64-thread
# blocks (more independent blocks per CU) and start-phase stagger.
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

show() {
  python - "$1" "$2" <<'PY'
import json, sys
try:
    d = json.load(open(sys.argv[2]))
    r = d.get("roofline", {})
    print(sys.argv[1], "enc_ms", r.get("kernel_ms"), "frac", r.get("frac"))
except Exception as e:
    print(sys.argv[1], "FAILED:", e)
PY
}

run() { # name env...
  local name=$1; shift
  env "$@" timeout 200 python bench.py --technique cauchy_orig --steps 5 \
    --warmup 2 --no-cpu-baseline --no-selfcheck \
    > gpurun_out/bm_${name}.json 2> gpurun_out/bm_${name}.err
  show "$name" gpurun_out/bm_${name}.json
}

# parity sanity for the 128-thread + q=128 shape first
ECX_BITT=128 ECX_BITQ=8 python -m pytest tests/test_gpu_parity.py -q \
  -k "bitmatrix" 2>&1 | tail -2

run base
run t128q8    ECX_BITT=128 ECX_BITQ=8
run t128q16   ECX_BITT=128 ECX_BITQ=16
run t64q8     ECX_BITT=64  ECX_BITQ=8
run stag1     ECX_BITSTAGGER=1
run stag2     ECX_BITSTAGGER=2
run stag4     ECX_BITSTAGGER=4
run t128q8s2  ECX_BITT=128 ECX_BITQ=8 ECX_BITSTAGGER=2
run t128q8w16 ECX_BITT=128 ECX_BITQ=8 ECX_BITW=16
