#!/bin/bash
# xGMI readiness sweep (VERDICT r1 task 8): tune the gradient-bucket
# all-reduce for the MI355X's point-to-point xGMI mesh (7 links/GPU) in
# ONE multi-GPU lease.  Sweeps ADAPTDL_BUCKET_CAP_MB x RCCL channel /
# protocol settings over the flagship bench and records ms_per_step per
# cell; afterwards run the best cell under rocprofv3 and check comm
# overlap with tools/trace_gaps.py (comm_overlap_pct).
#
# Usage:  bash tools/xgmi_sweep.sh [NGPUS] [STEPS] [WARMUP]
# Dry-run (CPU, no GPUs; validates plumbing): DRY_RUN=1 bash tools/xgmi_sweep.sh 2
#
# Rationale for the matrix:
# - Bucket cap: ring all-reduce over xGMI is per-link bound (~153 GB/s
#   per link); caps 8-128 MB bracket the latency/bandwidth tradeoff for
#   ResNet-18-size buckets through BERT-size ones.
# - NCCL_MIN_NCHANNELS: RCCL derives channels from topology; forcing
#   more channels can engage more xGMI links per collective.
# - NCCL_PROTO=Simple vs LL128: LL''s lower latency wins for small
#   buckets, Simple for large.
set -u
cd "$(dirname "$0")/.."
NGPUS=${1:-8}
STEPS=${2:-15}
WARMUP=${3:-10}
OUT=${OUT:-gpurun_out/xgmi_sweep.csv}
mkdir -p "$(dirname "$OUT")"
echo "bucket_cap_mb,rccl_env,ms_per_step,value" > "$OUT"

run_cell() {
    local cap="$1" rcclenv="$2"
    local log
    log=$(mktemp)
    if [ "${DRY_RUN:-0}" = "1" ]; then
        # CPU/gloo plumbing check: 2 ranks, tiny config.
        env ADAPTDL_BUCKET_CAP_MB="$cap" $rcclenv \
            timeout 300 python -m torch.distributed.run --nnodes=1 \
            --nproc-per-node 2 --master-addr 127.0.0.1 \
            --master-port 29517 bench.py --gpus 2 --steps 2 --warmup 5 \
            --init-batch 64 --max-batch 128 --dataset-size 512 \
            --pool 64 > "$log" 2>&1
    else
        env ADAPTDL_BUCKET_CAP_MB="$cap" $rcclenv \
            timeout 300 python -m torch.distributed.run --nnodes=1 \
            --nproc-per-node "$NGPUS" --master-addr 127.0.0.1 \
            --master-port 29517 bench.py --gpus "$NGPUS" \
            --steps "$STEPS" --warmup "$WARMUP" > "$log" 2>&1
    fi
    local line
    line=$(grep '"metric"' "$log" | tail -1)
    local ms val
    ms=$(printf '%s' "$line" | sed -n 's/.*"ms_per_step": \([0-9.]*\).*/\1/p')
    val=$(printf '%s' "$line" | sed -n 's/.*"value": \([0-9.]*\).*/\1/p')
    echo "${cap},\"${rcclenv}\",${ms:-FAIL},${val:-FAIL}" | tee -a "$OUT"
    [ -z "$ms" ] && tail -5 "$log"
    rm -f "$log"
}

if [ "${DRY_RUN:-0}" = "1" ]; then
    # Plumbing check: one cell per axis.
    run_cell 32 ""
    run_cell 8 ""
    echo "dry-run ok"; exit 0
fi

# Phase 1: bucket cap alone (default RCCL settings).
for cap in 8 16 32 64 128; do
    run_cell "$cap" ""
done

# Phase 2: RCCL knobs at the best default-cap (re-run top cap manually
# if phase 1 moved the optimum away from 32).
for rcclenv in "NCCL_MIN_NCHANNELS=28" "NCCL_PROTO=Simple" \
               "NCCL_PROTO=LL128" \
               "NCCL_MIN_NCHANNELS=28 NCCL_PROTO=Simple"; do
    run_cell 32 "$rcclenv"
done

echo "sweep written to $OUT"
