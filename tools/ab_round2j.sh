#!/bin/bash
# Round-2 tenth GPU pass: measure r50 with the raised atomic bound,
# and BERT/transformer with the stepper enabled (A/B vs eager).
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

timeout 500 python bench.py --model resnet50-imagenet --steps 8 --warmup 6 \
    > gpurun_out/j_r50.log 2>&1
grep -o '"ms_per_step": [0-9.]*\|"global_batch": [0-9]*\|"value": [0-9.]*\|"hipgraph": {[^}]*}' \
    gpurun_out/j_r50.log | tr '\n' ' '; echo

for cfg in "" "ADAPTDL_HIPGRAPH=0"; do
    name=${cfg:-graphed}; name=${name%%=*}
    env $cfg timeout 400 python bench.py --model transformer-wt2 \
        --steps 20 --warmup 10 > "gpurun_out/j_tr_${name}.log" 2>&1
    grep -o '"ms_per_step": [0-9.]*\|"global_batch": [0-9]*\|"hipgraph": {[^}]*}' \
        "gpurun_out/j_tr_${name}.log" | tr '\n' ' '; echo
    env $cfg timeout 500 python bench.py --model bert-base \
        --steps 12 --warmup 8 > "gpurun_out/j_bert_${name}.log" 2>&1
    grep -o '"ms_per_step": [0-9.]*\|"global_batch": [0-9]*\|"hipgraph": {[^}]*}' \
        "gpurun_out/j_bert_${name}.log" | tr '\n' ' '; echo
done
echo DONE
