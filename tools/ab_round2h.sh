#!/bin/bash
# Round-2 eighth GPU pass: (1) ResNet-50 forced-accumulation probe (is
# the fit right to keep global 256?), (2) BN apply-phase HBM-bytes PMC
# (does the re-read hit L2? decides the single-pass BN lever).
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

# 1. ResNet-50: fit's pick (256) vs forced 2048 global via accumulation.
timeout 500 python bench.py --model resnet50-imagenet --steps 8 --warmup 6 \
    > gpurun_out/h_r50_fit.log 2>&1
grep -o '"ms_per_step": [0-9.]*\|"global_batch": [0-9]*\|"samples_per_sec": [0-9.]*' \
    gpurun_out/h_r50_fit.log | tr '\n' ' '; echo
timeout 500 python bench.py --model resnet50-imagenet --steps 8 --warmup 6 \
    --init-batch 2048 --max-batch 2048 --bounds 32,256 \
    > gpurun_out/h_r50_accum.log 2>&1
grep -o '"ms_per_step": [0-9.]*\|"global_batch": [0-9]*\|"samples_per_sec": [0-9.]*\|"accum_steps": [0-9]*' \
    gpurun_out/h_r50_accum.log | tr '\n' ' '; echo

# 2. PMC: HBM fetch/write bytes per kernel on a short flagship run.
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
ADAPTDL_HIPGRAPH=0 timeout 500 rocprofv3 --pmc FETCH_SIZE,WRITE_SIZE \
    -f csv -d gpurun_out/h_pmc -- \
    python bench.py --steps 4 --warmup 4 --eff-probe 2 \
    > gpurun_out/h_pmc.log 2>&1 || true
find gpurun_out/h_pmc -name '*.csv' | head -3
python - <<'PYEOF'
import csv, glob, collections
files = glob.glob("gpurun_out/h_pmc/**/*counter*.csv", recursive=True) or \
        glob.glob("gpurun_out/h_pmc/**/*.csv", recursive=True)
print("pmc files:", files[:3])
agg = collections.defaultdict(lambda: [0.0, 0.0, 0])
for fn in files:
    for r in csv.DictReader(open(fn)):
        name = (r.get("Kernel_Name") or r.get("Kernel-Name") or "?")
        name = name.split("(")[0][:48]
        cname = r.get("Counter_Name") or r.get("Counter-Name") or ""
        try:
            v = float(r.get("Counter_Value") or r.get("Counter-Value") or 0)
        except ValueError:
            continue
        if cname == "FETCH_SIZE":
            agg[name][0] += v; agg[name][2] += 1
        elif cname == "WRITE_SIZE":
            agg[name][1] += v
for name, (f, w, n) in sorted(agg.items(), key=lambda kv: -kv[1][0])[:14]:
    print(f"{name:48s} fetchKB={f:12.0f} writeKB={w:12.0f} n={n}")
PYEOF
echo DONE
