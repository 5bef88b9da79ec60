#!/bin/bash
# Round-2 GPU call 2: bitmatrix wpb (windows-per-block) sweep, PMC profile
# of the best bitmatrix config, cpu_baseline reconciliation probe.
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

# parity safety for the reshaped kernel first (subset: bitmatrix tests)
python -m pytest tests/test_gpu_parity.py -q -k "bitmatrix or cauchy" 2>&1 \
  | tail -3 | tee gpurun_out/pytest_bm.log

show() {
  python - "$1" "$2" <<'PY'
import json, sys
try:
    d = json.load(open(sys.argv[2]))
    r = d.get("roofline", {})
    print(sys.argv[1], "enc_ms", r.get("kernel_ms"), "frac", r.get("frac"),
          "value", d.get("value"))
except Exception as e:
    print(sys.argv[1], "FAILED:", e)
PY
}

for wpb in 1 4 8 16 64; do
  ECX_BITW=$wpb timeout 200 python bench.py --technique cauchy_orig --steps 5 \
    --warmup 2 --no-cpu-baseline --no-selfcheck \
    > gpurun_out/bm_w${wpb}.json 2> gpurun_out/bm_w${wpb}.err
  show "WPB=$wpb" gpurun_out/bm_w${wpb}.json
done

# wpb x window-size cross: larger q with fewer blocks might now win
for bq in 32 48; do
  ECX_BITW=8 ECX_BITQ=$bq timeout 200 python bench.py --technique cauchy_orig \
    --steps 5 --warmup 2 --no-cpu-baseline --no-selfcheck \
    > gpurun_out/bm_w8q${bq}.json 2> gpurun_out/bm_w8q${bq}.err
  show "W8 BITQ=$bq" gpurun_out/bm_w8q${bq}.json
done

# PMC profile of the default config (separate passes; TCC slots are tight)
cd /tmp
for pmc in "SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY" \
           "SQ_LDS_IDX_ACTIVE SQ_LDS_BANK_CONFLICT SQ_WAVES"; do
  rocprofv3 --pmc $pmc -d /root/repo/gpurun_out/prof_bm -o pmc_$(echo $pmc | cut -d' ' -f2) -- \
    timeout 200 python /root/repo/bench.py --technique cauchy_orig --steps 2 \
    --warmup 1 --no-cpu-baseline --no-selfcheck > /dev/null 2>&1
done
cd /root/repo
python - <<'PY'
import csv, glob, collections
for f in sorted(glob.glob("gpurun_out/prof_bm/**/*.csv", recursive=True)):
    agg = collections.defaultdict(float)
    n = collections.defaultdict(int)
    with open(f) as fh:
        for row in csv.DictReader(fh):
            name = row.get("Kernel_Name", "")
            if "bitmatrix" not in name:
                continue
            cname = row.get("Counter_Name")
            val = float(row.get("Counter_Value", 0) or 0)
            agg[cname] += val
            n[cname] += 1
    if agg:
        print(f)
        for k, v in sorted(agg.items()):
            print("   ", k, f"{v:.3e}", "dispatches", n[k])
PY

# cpu_baseline reconciliation (host side of the same box)
timeout 420 python tools/cpu_recon.py 2>&1 | tee gpurun_out/cpu_recon.log
