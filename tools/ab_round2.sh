#!/bin/bash
# Round-2 opener: validate + measure the three pre-built gated paths in
# ONE gpurun call (budget-efficient).  Usage on the GPU box:
#   bash tools/ab_round2.sh
# Writes everything under gpurun_out/ for copy-back.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

# 1. Numerics of the unvalidated experimental kernels (s2 fwd + s2 wrw).
ADAPTDL_EXPERIMENTAL_S2_FWD=1 timeout 180 python -m pytest \
    tests/test_fused_conv.py -q -k "s2_fwd or s2_wrw or w8b" \
    2>&1 | tail -3 | tee gpurun_out/ab_numerics.log

# 2. Per-shape timings vs the library (appends s2 rows).
timeout 180 python tools/conv_time.py > gpurun_out/ab_conv_time.log 2>&1
tail -8 gpurun_out/ab_conv_time.log

# 2b. hipGraph stepper: capture-vs-eager numerics A/B on hardware
# (tests/test_gpu_e2e.py::test_hipgraph_stepper_matches_eager).
ADAPTDL_HIPGRAPH=1 timeout 240 python -m pytest \
    tests/test_gpu_e2e.py -q -k hipgraph -m gpu \
    2>&1 | tail -3 | tee gpurun_out/ab_hipgraph.log

# 3. In-context bench A/B: baseline, then each gated path alone.
for cfg in "" "ADAPTDL_S2_1X1=1" "ADAPTDL_S2_WRW=1" "ADAPTDL_S2_W8B=1" \
           "ADAPTDL_HIPGRAPH=1"; do
    name=${cfg:-baseline}; name=${name%%=*}
    env $cfg timeout 260 python bench.py --steps 20 --warmup 12 \
        > "gpurun_out/ab_bench_${name}.log" 2>&1
    grep '"metric"' "gpurun_out/ab_bench_${name}.log" | tail -1
done
# Keep only measured wins (the bench_r10 rule); see ROADMAP.md item 3.

# 4. Idle-gap evidence for the hipGraph A/B: kernel traces of the
# eager and graphed steady states, summarized with tools/trace_gaps.py
# (copy the summaries into profiles/ when committing).
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
for cfg in "" "ADAPTDL_HIPGRAPH=1"; do
    name=${cfg:-eager}; name=${name%%=*}
    env $cfg timeout 300 rocprofv3 --kernel-trace -f csv \
        -d "gpurun_out/trace_${name}" -- \
        python bench.py --steps 15 --warmup 10 \
        > "gpurun_out/trace_${name}.log" 2>&1
    python tools/trace_gaps.py \
        $(find gpurun_out/trace_${name} -name '*kernel_trace*.csv') \
        > "gpurun_out/gaps_${name}.txt" 2>&1 || true
    tail -5 "gpurun_out/gaps_${name}.txt"
done
