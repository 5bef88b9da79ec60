#!/bin/bash
# Round-2 evidence refresh: kernel-trace stats + PMC traffic for the
# redesigned bitmatrix kernel, plus the full bench-line set.
cd /root/repo
mkdir -p gpurun_out/prof_r2
export TMPDIR=/tmp

show() {
  python - "$1" "$2" <<'PY'
import json, sys
try:
    d = json.load(open(sys.argv[2]))
    r = d.get("roofline") or {}
    print(sys.argv[1], "value", d.get("value"), d.get("unit"),
          "enc_ms", r.get("kernel_ms"), "dec_ms", r.get("decode_kernel_ms"),
          "frac", r.get("frac"))
except Exception as e:
    print(sys.argv[1], "FAILED:", e)
PY
}

# canonical bench lines (full: selfcheck + cpu_baseline on the flagship)
timeout 420 python bench.py --steps 20 --warmup 5 \
  > gpurun_out/bench_r02_rs83.json 2> gpurun_out/bench_r02_rs83.err
show rs83 gpurun_out/bench_r02_rs83.json

for t in cauchy_orig cauchy_good; do
  timeout 300 python bench.py --technique $t --steps 10 --warmup 3 \
    --no-cpu-baseline > gpurun_out/bench_r02_$t.json 2> gpurun_out/bench_r02_$t.err
  show $t gpurun_out/bench_r02_$t.json
done
timeout 300 python bench.py --config cauchy104 --steps 10 --warmup 3 \
  --no-cpu-baseline > gpurun_out/bench_r02_cauchy104.json 2> gpurun_out/bench_r02_cauchy104.err
show cauchy104 gpurun_out/bench_r02_cauchy104.json

# rocprofv3 kernel-trace stats over the bitmatrix bench (cross-check the
# hipEvent numbers) — counters in their own separate passes
cd /tmp
rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_r2 -o bmtrace -- \
  timeout 200 python /root/repo/bench.py --technique cauchy_orig --steps 3 \
  --warmup 1 --no-cpu-baseline --no-selfcheck --stripes 2048 \
  > /dev/null 2>&1
rocprofv3 --pmc FETCH_SIZE --output-format csv -d /root/repo/gpurun_out/prof_r2 -o bmfetch -- \
  timeout 200 python /root/repo/bench.py --technique cauchy_orig --steps 2 \
  --warmup 1 --no-cpu-baseline --no-selfcheck --stripes 2048 \
  > /dev/null 2>&1
rocprofv3 --pmc WRITE_SIZE --output-format csv -d /root/repo/gpurun_out/prof_r2 -o bmwrite -- \
  timeout 200 python /root/repo/bench.py --technique cauchy_orig --steps 2 \
  --warmup 1 --no-cpu-baseline --no-selfcheck --stripes 2048 \
  > /dev/null 2>&1
cd /root/repo
for db in gpurun_out/prof_r2/*/bmtrace_results.db gpurun_out/prof_r2/bmtrace_results.db; do
  [ -f "$db" ] && python tools/rocprof_summary.py "$db" 2>/dev/null | head -20
done
python - <<'PY'
import csv, glob, collections
for f in sorted(glob.glob("gpurun_out/prof_r2/**/*counter*.csv", recursive=True)):
    agg = collections.defaultdict(float)
    n = collections.defaultdict(int)
    with open(f) as fh:
        for row in csv.DictReader(fh):
            if "bitmatrix" not in row.get("Kernel_Name", ""):
                continue
            agg[row["Counter_Name"]] += float(row["Counter_Value"] or 0)
            n[row["Counter_Name"]] += 1
    for k, v in agg.items():
        print(f.split("/")[-1], k, f"{v:.6e}", "rows", n[k])
PY
