#!/bin/bash
# Round-2 seventh GPU pass: validate the accumulated engine/layer
# changes (weight-cast cache, prev_cycle_valid stats, skip-unused,
# bf16) on hardware and A/B the cast cache in context.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

# 1. Full GPU suite.
timeout 900 python -m pytest tests/ -q -m gpu --tb=short \
    > gpurun_out/g_gputests.log 2>&1
tail -3 gpurun_out/g_gputests.log
ADAPTDL_HIPGRAPH=1 timeout 300 python -m pytest \
    tests/test_gpu_e2e.py -q -k hipgraph -m gpu --tb=short \
    > gpurun_out/g_hipgraph_test.log 2>&1
tail -2 gpurun_out/g_hipgraph_test.log

# 2. Cast-cache A/B: eager and graphed, on vs off.
for cfg in "" "ADAPTDL_NO_WB_CACHE=1" "ADAPTDL_HIPGRAPH=0" \
           "ADAPTDL_HIPGRAPH=0 ADAPTDL_NO_WB_CACHE=1"; do
    name=$(echo "${cfg:-default}" | tr ' =' '__')
    env $cfg timeout 300 python bench.py --steps 20 --warmup 12 \
        > "gpurun_out/g_bench_${name}.log" 2>&1
    grep '"ms_per_step"' "gpurun_out/g_bench_${name}.log" | \
        grep -o '"ms_per_step": [0-9.]*' | tail -1
done

# 3. ResNet-50 re-check with everything current.
timeout 500 python bench.py --model resnet50-imagenet --steps 10 --warmup 8 \
    > gpurun_out/g_bench_r50.log 2>&1
grep -o '"ms_per_step": [0-9.]*\|"value": [0-9.]*' gpurun_out/g_bench_r50.log | tr '\n' ' '
echo DONE
