#!/bin/bash
# Round-2 GPU call 1: full GPU test suite (validates the glds bitmatrix
# load phase, cauchy_good parity, bitmatrix deltas), then a bitmatrix
# kernel A/B sweep (LDS window size x NT) and a cauchy_good bench line.
cd /root/repo
mkdir -p gpurun_out

python -m pytest tests -m gpu -q 2>&1 | tail -5 | tee gpurun_out/pytest_gpu_r2a.log

show() {
  python - "$1" "$2" <<'PY'
import json, sys
try:
    d = json.load(open(sys.argv[2]))
    r = d.get("roofline", {})
    print(sys.argv[1], "enc_ms", r.get("kernel_ms"), "dec_ms",
          r.get("decode_kernel_ms"), "frac", r.get("frac"),
          "value", d.get("value"))
except Exception as e:
    print(sys.argv[1], "FAILED:", e)
PY
}

for bq in 16 32 8; do
  ECX_BITQ=$bq timeout 200 python bench.py --technique cauchy_orig --steps 5 \
    --warmup 2 --no-cpu-baseline --no-selfcheck \
    > gpurun_out/bm_q${bq}.json 2> gpurun_out/bm_q${bq}.err
  show "BITQ=$bq" gpurun_out/bm_q${bq}.json
done

ECX_NT=0 ECX_BITQ=16 timeout 200 python bench.py --technique cauchy_orig \
  --steps 5 --warmup 2 --no-cpu-baseline --no-selfcheck \
  > gpurun_out/bm_nt0.json 2> gpurun_out/bm_nt0.err
show NT0 gpurun_out/bm_nt0.json

timeout 300 python bench.py --technique cauchy_good --steps 5 --warmup 2 \
  --no-cpu-baseline > gpurun_out/bm_good.json 2> gpurun_out/bm_good.err
show GOOD gpurun_out/bm_good.json
