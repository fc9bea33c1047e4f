#!/bin/bash
# Round-2 fourth GPU pass: validate the generalized (ImageNet-width)
# wrw kernel, time it vs MIOpen per shape, and A/B the ResNet-50 bench
# with the fused path now auto-engaging.  Also confirm the hipGraph
# default-on bench path.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

# 1. Numerics: all wrw shapes incl. 56/28/14/7.
timeout 420 python -m pytest tests/test_fused_conv.py -q -m gpu \
    -k "wrw and not s2" --tb=short > gpurun_out/d_wrw_numerics.log 2>&1
tail -4 gpurun_out/d_wrw_numerics.log

# 2. Per-shape timing vs MIOpen.
timeout 300 python tools/wrw_time.py > gpurun_out/d_wrw_time.log 2>&1
cat gpurun_out/d_wrw_time.log

# 3. ResNet-50 bench (fused wrw now auto-engages at 56/28/14/7) and a
# default-flag flagship run (hipGraph default-on at N=1).
timeout 500 python bench.py --model resnet50-imagenet --steps 10 --warmup 8 \
    > gpurun_out/d_bench_resnet50.log 2>&1
grep '"metric"' gpurun_out/d_bench_resnet50.log | tail -1
timeout 260 python bench.py --steps 20 --warmup 12 \
    > gpurun_out/d_bench_default.log 2>&1
grep '"metric"' gpurun_out/d_bench_default.log | tail -1

# 4. Kernel-stats of the resnet50 bench for profiles/ (which kernels
# carry the step: fused wrw vs MIOpen).
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 500 rocprofv3 --stats -f csv -d gpurun_out/dstats_r50 -- \
    python bench.py --model resnet50-imagenet --steps 6 --warmup 5 \
    > gpurun_out/dstats_r50.log 2>&1 || true
find gpurun_out/dstats_r50 -name '*stats*.csv' | head -2
echo DONE
