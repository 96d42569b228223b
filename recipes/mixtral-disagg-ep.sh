#!/bin/bash
# BASELINE config #5: Mixtral 8x7B disaggregated P/D with expert
# parallelism inside each TP pool (MoE grouped-GEMM + all-to-all).
MODEL=${MODEL:-mixtral-8x7b}
export MODEL
cd "$(dirname "$0")/.." || exit 1
source recipes/_lib.sh

HIP_VISIBLE_DEVICES=0,1 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29521 \
    -m dynamo_amd.workers --model "$MODEL" --tp-size 2 --moe-ep \
    --worker-type prefill --discovery "$DISC" \
    > "$NS/prefill.log" 2>&1 &
PIDS+=($!)
HIP_VISIBLE_DEVICES=2,3 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29522 \
    -m dynamo_amd.workers --model "$MODEL" --tp-size 2 --moe-ep \
    --worker-type decode --discovery "$DISC" \
    > "$NS/decode.log" 2>&1 &
PIDS+=($!)
wait_marker "$NS/prefill.log" WORKER_READY 600 || exit 1
wait_marker "$NS/decode.log" WORKER_READY 600 || exit 1

python -m dynamo_amd.frontend --discovery "$DISC" --port "$PORT" \
    > "$NS/frontend.log" 2>&1 &
PIDS+=($!)
wait_marker "$NS/frontend.log" FRONTEND_READY 60 || exit 1
sleep 2
maybe_check "${1:-}"
