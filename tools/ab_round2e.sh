#!/bin/bash
# Round-2 fifth GPU pass: bf16 stat kernels, wrw big-chunk A/B for the
# 14/7 shapes, ResNet-50 in-context A/B, and the config-5 elastic
# rescale drill end-to-end on 1 GPU.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

# 1. New bf16 kernel numerics + wrw numerics under BOTH chunkings.
timeout 420 python -m pytest tests/test_gpu_kernels.py -q -m gpu \
    --tb=short > gpurun_out/e_kernels.log 2>&1
tail -3 gpurun_out/e_kernels.log
ADAPTDL_WRW_BIGCHUNK=1 timeout 300 python -m pytest \
    tests/test_fused_conv.py -q -m gpu -k "wrw and not s2" --tb=short \
    > gpurun_out/e_wrw_big_numerics.log 2>&1
tail -3 gpurun_out/e_wrw_big_numerics.log

# 2. Per-shape timing: default vs big-chunk (14/7 rows are the A/B).
timeout 300 python tools/wrw_time.py > gpurun_out/e_wrw_time_default.log 2>&1
tail -7 gpurun_out/e_wrw_time_default.log
ADAPTDL_WRW_BIGCHUNK=1 timeout 300 python tools/wrw_time.py \
    > gpurun_out/e_wrw_time_big.log 2>&1
tail -7 gpurun_out/e_wrw_time_big.log

# 3. ResNet-50 bench under the winning config (also default for A/B).
timeout 500 python bench.py --model resnet50-imagenet --steps 10 --warmup 8 \
    > gpurun_out/e_bench_r50_default.log 2>&1
grep '"metric"' gpurun_out/e_bench_r50_default.log | tail -1
ADAPTDL_WRW_BIGCHUNK=1 timeout 500 python bench.py --model resnet50-imagenet \
    --steps 10 --warmup 8 > gpurun_out/e_bench_r50_big.log 2>&1
grep '"metric"' gpurun_out/e_bench_r50_big.log | tail -1

# 4. Config-5 drill: ResNet-50 224 elastic rescale on 1 GPU through the
# controller (SIGTERM -> warm RAM checkpoint -> restart).
timeout 600 python examples/elastic_rescale/main.py --gpus 1 \
    --phase-seconds 25 > gpurun_out/e_rescale_drill.log 2>&1
tail -12 gpurun_out/e_rescale_drill.log
echo DONE
