#!/bin/bash
# Round-2 second GPU pass: (1) hipGraph stepper re-A/B after the
# realign+fresh-pool fixes, (2) first GPU measurements of the new bench
# workloads (BASELINE configs 3/4/5), (3) determinism check of the
# fixed-seed efficiency probe on hardware.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

# 1. hipGraph e2e numerics (failed last round pre-fix; full traceback).
ADAPTDL_HIPGRAPH=1 timeout 300 python -m pytest \
    tests/test_gpu_e2e.py -q -k hipgraph -m gpu --tb=long \
    > gpurun_out/b_hipgraph_test.log 2>&1
tail -5 gpurun_out/b_hipgraph_test.log

# 2. Graph-step CPU-suite sanity on the box (fast) - the bookkeeping
# changed; make sure nothing GPU-environment-specific breaks it.
timeout 300 python -m pytest tests/test_graph_step.py -q \
    > gpurun_out/b_graphstep_cpu.log 2>&1
tail -2 gpurun_out/b_graphstep_cpu.log

# 3. Flagship bench: baseline twice (value-determinism on hardware),
# then hipGraph in-context A/B with the fixed stepper.
for tag in base1 base2; do
    timeout 260 python bench.py --steps 20 --warmup 12 \
        > "gpurun_out/b_bench_${tag}.log" 2>&1
    grep '"metric"' "gpurun_out/b_bench_${tag}.log" | tail -1
done
ADAPTDL_HIPGRAPH=1 timeout 300 python bench.py --steps 20 --warmup 12 \
    > gpurun_out/b_bench_hipgraph.log 2>&1
grep -E '"metric"|desync|capture failed' gpurun_out/b_bench_hipgraph.log | tail -3

# 4. New workloads at N=1 (BASELINE configs 3 and 4, plus the ResNet-50
# ImageNet shape of config 5).
timeout 400 python bench.py --model transformer-wt2 --steps 20 --warmup 10 \
    > gpurun_out/b_bench_transformer.log 2>&1
grep '"metric"' gpurun_out/b_bench_transformer.log | tail -1
timeout 400 python bench.py --model bert-base --steps 20 --warmup 10 \
    > gpurun_out/b_bench_bert.log 2>&1
grep '"metric"' gpurun_out/b_bench_bert.log | tail -1
timeout 500 python bench.py --model resnet50-imagenet --steps 10 --warmup 8 \
    > gpurun_out/b_bench_resnet50.log 2>&1
grep '"metric"' gpurun_out/b_bench_resnet50.log | tail -1

# 5. Kernel traces (csv this time) for the eager-vs-graphed gap
# analysis, summarized with tools/trace_gaps.py.
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
for cfg in "" "ADAPTDL_HIPGRAPH=1"; do
    name=${cfg:-eager}; name=${name%%=*}
    env $cfg timeout 300 rocprofv3 --kernel-trace -f csv \
        -d "gpurun_out/btrace_${name}" -- \
        python bench.py --steps 15 --warmup 10 \
        > "gpurun_out/btrace_${name}.log" 2>&1
    python tools/trace_gaps.py \
        $(find gpurun_out/btrace_${name} -name '*kernel_trace*.csv') \
        > "gpurun_out/b_gaps_${name}.txt" 2>&1 || true
    tail -6 "gpurun_out/b_gaps_${name}.txt"
done
# Kernel-stats summary of the new workloads for profiles/ (one each).
timeout 400 rocprofv3 --stats -f csv -d gpurun_out/bstats_bert -- \
    python bench.py --model bert-base --steps 8 --warmup 6 \
    > gpurun_out/bstats_bert.log 2>&1 || true
timeout 500 rocprofv3 --stats -f csv -d gpurun_out/bstats_resnet50 -- \
    python bench.py --model resnet50-imagenet --steps 6 --warmup 5 \
    > gpurun_out/bstats_resnet50.log 2>&1 || true
echo DONE
