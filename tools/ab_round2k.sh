#!/bin/bash
# Round-2 eleventh GPU pass: consolidated validation of everything that
# changed since pass g (fp16 stat kernels, r50 bounds, tuple unpacking,
# cast cache everywhere) + transformer graphed A/B retest + smoke.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

timeout 900 python -m pytest tests/ -q -m gpu --tb=short \
    > gpurun_out/k_gputests.log 2>&1
tail -3 gpurun_out/k_gputests.log
ADAPTDL_HIPGRAPH=1 timeout 300 python -m pytest \
    tests/test_gpu_e2e.py -q -k hipgraph -m gpu --tb=short \
    > gpurun_out/k_hipgraph.log 2>&1
tail -2 gpurun_out/k_hipgraph.log

timeout 300 python -c "import __graft_entry__ as g; g.smoke()" \
    > gpurun_out/k_smoke.log 2>&1
tail -3 gpurun_out/k_smoke.log

timeout 260 python bench.py --steps 20 --warmup 12 \
    > gpurun_out/k_bench_default.log 2>&1
grep -o '"ms_per_step": [0-9.]*\|"value": [0-9.]*\|"hipgraph": {[^}]*}' \
    gpurun_out/k_bench_default.log | tr '\n' ' '; echo

for cfg in "" "ADAPTDL_HIPGRAPH=0"; do
    name=${cfg:-graphed}; name=${name%%=*}
    env $cfg timeout 400 python bench.py --model transformer-wt2 \
        --steps 20 --warmup 10 > "gpurun_out/k_tr_${name}.log" 2>&1
    grep -o '"ms_per_step": [0-9.]*\|"global_batch": [0-9]*\|"hipgraph": {[^}]*}' \
        "gpurun_out/k_tr_${name}.log" | tr '\n' ' '; echo
done
echo DONE
