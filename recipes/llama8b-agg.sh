#!/bin/bash
# BASELINE config #2: Llama-3-8B aggregated on one MI355X.
cd "$(dirname "$0")/.." || exit 1
source recipes/_lib.sh
MODEL=${MODEL:-llama-3-8b}
DEVICE=${DEVICE:-cuda:0}

python -m dynamo_amd.workers --model "$MODEL" --device "$DEVICE" \
    --discovery "$DISC" --max-num-seqs 64 --max-batched-tokens 8192 \
    ${WORKER_ARGS:-} > "$NS/worker.log" 2>&1 &
PIDS+=($!)
wait_marker "$NS/worker.log" WORKER_READY 240 || exit 1

python -m dynamo_amd.frontend --discovery "$DISC" --port "$PORT" \
    > "$NS/frontend.log" 2>&1 &
PIDS+=($!)
wait_marker "$NS/frontend.log" FRONTEND_READY 60 || exit 1
sleep 2
maybe_check "${1:-}"
