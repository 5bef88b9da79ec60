#!/bin/bash
# Round-2 GPU call 3: pipelined-bitmatrix A/B (+ parity under PIPE), PMC
# (csv) of the default bitmatrix config, NUMA-fixed cpu_baseline check.
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

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

# parity with the pipelined kernel enabled (correctness before speed)
ECX_BITPIPE=1 python -m pytest tests/test_gpu_parity.py -q \
  -k "bitmatrix or cauchy" 2>&1 | tail -2 | tee gpurun_out/pytest_pipe.log

for wpb in 4 8 16; do
  ECX_BITPIPE=1 ECX_BITW=$wpb timeout 200 python bench.py \
    --technique cauchy_orig --steps 5 --warmup 2 --no-cpu-baseline \
    --no-selfcheck > gpurun_out/bmp_w${wpb}.json 2> gpurun_out/bmp_w${wpb}.err
  show "PIPE WPB=$wpb" gpurun_out/bmp_w${wpb}.json
done
# pipe with bigger window (LDS 2x16K? q=512 => 2x32K+ops, 2 blocks/CU)
ECX_BITPIPE=1 ECX_BITW=8 ECX_BITQ=32 timeout 200 python bench.py \
  --technique cauchy_orig --steps 5 --warmup 2 --no-cpu-baseline \
  --no-selfcheck > gpurun_out/bmp_q32.json 2> gpurun_out/bmp_q32.err
show "PIPE q512" gpurun_out/bmp_q32.json
# non-pipe reference point on the same box
ECX_BITW=8 timeout 200 python bench.py --technique cauchy_orig --steps 5 \
  --warmup 2 --no-cpu-baseline --no-selfcheck \
  > gpurun_out/bm_ref.json 2> gpurun_out/bm_ref.err
show "NOPIPE WPB=8" gpurun_out/bm_ref.json

# PMC (csv this time) for best config so far (non-pipe wpb=8) and pipe
cd /tmp
for cfg in nopipe pipe; do
  [ "$cfg" = pipe ] && export ECX_BITPIPE=1 || export ECX_BITPIPE=0
  for pmc in "SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY" \
             "SQ_LDS_IDX_ACTIVE SQ_LDS_BANK_CONFLICT SQ_WAVES SQ_BUSY_CYCLES"; do
    n=$(echo $pmc | cut -d' ' -f2)
    rocprofv3 --pmc $pmc --output-format csv \
      -d /root/repo/gpurun_out/prof3 -o ${cfg}_${n} -- \
      timeout 200 python /root/repo/bench.py --technique cauchy_orig \
      --steps 2 --warmup 1 --no-cpu-baseline --no-selfcheck --stripes 2048 \
      > /dev/null 2>&1
  done
done
unset ECX_BITPIPE
cd /root/repo
python - <<'PY'
import csv, glob, collections
for f in sorted(glob.glob("gpurun_out/prof3/**/*counter*.csv", recursive=True)):
    agg = collections.defaultdict(float)
    disp = collections.defaultdict(set)
    with open(f) as fh:
        for row in csv.DictReader(fh):
            name = row.get("Kernel_Name", "")
            if "bitmatrix" not in name:
                continue
            agg[row["Counter_Name"]] += float(row["Counter_Value"] or 0)
            disp[row["Counter_Name"]].add(row.get("Dispatch_Id"))
    if agg:
        print(f.split("/")[-1])
        for k, v in sorted(agg.items()):
            print("   ", k, f"{v:.4e}", "disp", len(disp[k]))
PY

# NUMA-fixed CPU baseline: recon with and without first-touch + the real
# bench cpu_baseline path
timeout 300 python tools/cpu_recon.py 2>&1 | tee gpurun_out/cpu_recon2.log
timeout 300 python bench.py --steps 1 --warmup 0 --no-selfcheck \
  > gpurun_out/bench_cpubl.json 2> gpurun_out/bench_cpubl.err
python - <<'PY'
import json
d = json.load(open("gpurun_out/bench_cpubl.json"))
print("cpu_baseline:", json.dumps(d.get("cpu_baseline")))
PY
