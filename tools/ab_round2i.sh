#!/bin/bash
# Round-2 ninth GPU pass: (1) r50 batch-size throughput probe at valid
# single-microbatch sizes, (2) BN PMC on a minimal dispatch set.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
for bs in 512 1024; do
    timeout 500 python bench.py --model resnet50-imagenet --steps 8 \
        --warmup 6 --init-batch $bs --max-batch $bs --bounds 32,$bs \
        > "gpurun_out/i_r50_bs${bs}.log" 2>&1
    grep -o '"ms_per_step": [0-9.]*\|"global_batch": [0-9]*\|"samples_per_sec": [0-9.]*' \
        "gpurun_out/i_r50_bs${bs}.log" | tr '\n' ' '; echo
done
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 300 rocprofv3 --pmc FETCH_SIZE -f csv -d gpurun_out/i_pmc -- \
    python tools/bn_pmc.py > gpurun_out/i_pmc.log 2>&1 || true
tail -4 gpurun_out/i_pmc.log
find gpurun_out/i_pmc -name '*.csv' | head -3
python - <<'PYEOF'
import csv, glob, collections
files = glob.glob("gpurun_out/i_pmc/**/*.csv", recursive=True)
print("files:", files[:4])
agg = collections.defaultdict(lambda: [0.0, 0])
for fn in files:
    for r in csv.DictReader(open(fn)):
        name = (r.get("Kernel_Name") or "?").split("(")[0][:40]
        cname = r.get("Counter_Name") or ""
        if "FETCH" in cname:
            try:
                agg[name][0] += float(r.get("Counter_Value") or 0)
                agg[name][1] += 1
            except ValueError:
                pass
for name, (f, n) in sorted(agg.items(), key=lambda kv: -kv[1][0])[:10]:
    print(f"{name:42s} fetchKB={f:12.0f} n={n} avgKB={f/max(n,1):10.0f}")
PYEOF
echo DONE
