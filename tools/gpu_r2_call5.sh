#!/bin/bash
# Round-2 GPU call 5: DRAM-locality hypothesis — membench strided-read
# probe, packetsize sweep (contiguity without code change), XCD map A/B.
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

hipcc --offload-arch=gfx950 -O3 tools/membench.hip -o gpurun_out/membench \
  2> gpurun_out/membench_build.err
timeout 300 ./gpurun_out/membench | tee gpurun_out/membench.jsonl

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
run() { # name extra-args... (env vars via env)
  local name=$1; shift
  timeout 200 env "$@" python bench.py --technique cauchy_orig --steps 5 \
    --warmup 2 --no-cpu-baseline --no-selfcheck \
    > gpurun_out/bm_${name}.json 2> gpurun_out/bm_${name}.err
  show "$name" gpurun_out/bm_${name}.json
}
runp() { # name pkt
  local name=$1 pkt=$2
  timeout 200 python bench.py --technique cauchy_orig --packetsize $pkt \
    --steps 5 --warmup 2 --no-cpu-baseline --no-selfcheck \
    > gpurun_out/bm_${name}.json 2> gpurun_out/bm_${name}.err
  show "$name" gpurun_out/bm_${name}.json
}

run base2
runp pkt1024 1024
runp pkt512 512
runp pkt256 256
run xcd ECX_BITXCD=1
run xcdw8q16 ECX_BITXCD=1 ECX_BITW=8 ECX_BITQ=16
# parity safety for xcdmap
ECX_BITXCD=1 python -m pytest tests/test_gpu_parity.py -q -k bitmatrix 2>&1 | tail -2
