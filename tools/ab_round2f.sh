#!/bin/bash
# Round-2 sixth GPU pass: conv_mm in-context A/B (MIOpen fwd+bwd-data
# is 36% of steady kernel time + 9.6% SubTensorOp fills it drags in),
# and the transformer bench with BPTT accumulation enabled.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

# 0. conv_mm numerics sanity (unchanged code, but re-confirm).
ADAPTDL_EXPERIMENTAL_CONV_MM=1 timeout 300 python -m pytest \
    tests/test_fused_conv.py -q -m gpu -k "mm" --tb=short \
    > gpurun_out/f_mm_numerics.log 2>&1
tail -3 gpurun_out/f_mm_numerics.log

# 1. Flagship bench A/B: default (hipGraph on) vs +conv_mm, and
# eager-vs-eager+conv_mm (isolate the interaction).
timeout 260 python bench.py --steps 20 --warmup 12 \
    > gpurun_out/f_bench_default.log 2>&1
grep '"metric"' gpurun_out/f_bench_default.log | tail -1
ADAPTDL_EXPERIMENTAL_CONV_MM=1 timeout 300 python bench.py --steps 20 \
    --warmup 12 > gpurun_out/f_bench_convmm.log 2>&1
grep -E '"metric"|Error' gpurun_out/f_bench_convmm.log | tail -2
ADAPTDL_HIPGRAPH=0 timeout 260 python bench.py --steps 20 --warmup 12 \
    > gpurun_out/f_bench_eager.log 2>&1
grep '"metric"' gpurun_out/f_bench_eager.log | tail -1
ADAPTDL_HIPGRAPH=0 ADAPTDL_EXPERIMENTAL_CONV_MM=1 timeout 300 \
    python bench.py --steps 20 --warmup 12 \
    > gpurun_out/f_bench_eager_convmm.log 2>&1
grep '"metric"' gpurun_out/f_bench_eager_convmm.log | tail -1

# 2. Transformer with accumulation (expect N=1 batch scaling now).
timeout 400 python bench.py --model transformer-wt2 --steps 20 --warmup 10 \
    > gpurun_out/f_bench_transformer.log 2>&1
grep '"metric"' gpurun_out/f_bench_transformer.log | tail -1

# 3. Kernel stats under conv_mm (SubTensorOp reduction evidence).
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
ADAPTDL_EXPERIMENTAL_CONV_MM=1 ADAPTDL_HIPGRAPH=0 timeout 300 \
    rocprofv3 --kernel-trace -f csv -d gpurun_out/ftrace_convmm -- \
    python bench.py --steps 15 --warmup 10 \
    > gpurun_out/ftrace_convmm.log 2>&1
find gpurun_out/ftrace_convmm -name '*kernel_trace*.csv' | head -1
echo DONE
