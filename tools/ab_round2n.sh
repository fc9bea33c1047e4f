#!/bin/bash
# Round-2 final validation pass: full GPU suite, smoke, the driver's
# exact torchrun launch shape at N=1 (RCCL env path never exercised on
# hardware before), and fresh bench numbers for the profiles.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

timeout 900 python -m pytest tests/ -q -m gpu --tb=short \
    > gpurun_out/n_gputests.log 2>&1
tail -3 gpurun_out/n_gputests.log
ADAPTDL_HIPGRAPH=1 timeout 300 python -m pytest \
    tests/test_gpu_e2e.py -q -k hipgraph -m gpu --tb=short \
    > gpurun_out/n_hipgraph.log 2>&1
tail -2 gpurun_out/n_hipgraph.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" \
    > gpurun_out/n_smoke.log 2>&1
tail -2 gpurun_out/n_smoke.log

# Driver launch shape (torchrun, 1 proc, RCCL rendezvous via env://).
timeout 300 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 1 --master-addr 127.0.0.1 --master-port 29531 \
    bench.py --gpus 1 --steps 10 --warmup 8 \
    > gpurun_out/n_torchrun.log 2>&1
grep -o '"ms_per_step": [0-9.]*\|"global_batch": [0-9]*' \
    gpurun_out/n_torchrun.log | tr '\n' ' '; echo

timeout 260 python bench.py --steps 20 --warmup 12 \
    > gpurun_out/n_default.log 2>&1
grep -o '"ms_per_step": [0-9.]*\|"value": [0-9.]*' gpurun_out/n_default.log | tr '\n' ' '; echo
timeout 500 python bench.py --model resnet50-imagenet --steps 8 --warmup 6 \
    > gpurun_out/n_r50.log 2>&1
grep -o '"ms_per_step": [0-9.]*\|"value": [0-9.]*\|"global_batch": [0-9]*' gpurun_out/n_r50.log | tr '\n' ' '; echo
echo DONE
