#!/bin/bash
# Round-2 third GPU pass: validate the fixed hipGraph stepper + probe
# isolation + batched Adam GNS on hardware, and re-measure the bench
# suite at the restored headline config.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

# 1. Full GPU test suite (covers the round's rewrites + new kernels).
timeout 900 python -m pytest tests/ -q -m gpu --tb=short \
    > gpurun_out/c_gputests.log 2>&1
tail -4 gpurun_out/c_gputests.log

# 2. hipGraph e2e numerics with the fixed test + stepper.
ADAPTDL_HIPGRAPH=1 timeout 300 python -m pytest \
    tests/test_gpu_e2e.py -q -k hipgraph -m gpu --tb=long \
    > gpurun_out/c_hipgraph_test.log 2>&1
tail -4 gpurun_out/c_hipgraph_test.log

# 3. Flagship bench: twice for value determinism (and the restored
# global_batch=4096 choice), then the hipGraph in-context A/B.
for tag in base1 base2; do
    timeout 260 python bench.py --steps 20 --warmup 12 \
        > "gpurun_out/c_bench_${tag}.log" 2>&1
    grep '"metric"' "gpurun_out/c_bench_${tag}.log" | tail -1
done
ADAPTDL_HIPGRAPH=1 timeout 300 python bench.py --steps 20 --warmup 12 \
    > gpurun_out/c_bench_hipgraph.log 2>&1
grep -E '"metric"|desync|capture failed|Error' \
    gpurun_out/c_bench_hipgraph.log | tail -4

# 4. New workloads with the probe-isolation fix (expect real autoscale).
timeout 400 python bench.py --model transformer-wt2 --steps 20 --warmup 10 \
    > gpurun_out/c_bench_transformer.log 2>&1
grep '"metric"' gpurun_out/c_bench_transformer.log | tail -1
timeout 400 python bench.py --model bert-base --steps 15 --warmup 10 \
    > gpurun_out/c_bench_bert.log 2>&1
grep '"metric"' gpurun_out/c_bench_bert.log | tail -1
timeout 500 python bench.py --model resnet50-imagenet --steps 10 --warmup 8 \
    > gpurun_out/c_bench_resnet50.log 2>&1
grep '"metric"' gpurun_out/c_bench_resnet50.log | tail -1

# 5. Kernel traces: eager vs graphed steady state (tail 0.05 isolates
# the timed region at the end of the run).
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
for cfg in "" "ADAPTDL_HIPGRAPH=1"; do
    name=${cfg:-eager}; name=${name%%=*}
    env $cfg timeout 300 rocprofv3 --kernel-trace -f csv \
        -d "gpurun_out/ctrace_${name}" -- \
        python bench.py --steps 15 --warmup 10 \
        > "gpurun_out/ctrace_${name}.log" 2>&1
    python tools/trace_gaps.py --tail 0.05 \
        $(find gpurun_out/ctrace_${name} -name '*kernel_trace*.csv') \
        > "gpurun_out/c_gaps_${name}.txt" 2>&1 || true
    tail -7 "gpurun_out/c_gaps_${name}.txt"
done
echo DONE
