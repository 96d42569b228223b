#!/bin/bash
# BASELINE config #4: Llama-3-70B disaggregated, TP4 prefill pool +
# TP4 decode pool (8 GPUs). Each pool is one torchrun group: rank 0
# serves the request plane, followers run the lockstep protocol;
# RCCL all-reduce over xGMI inside each pool.
cd "$(dirname "$0")/.." || exit 1
source recipes/_lib.sh
MODEL=${MODEL:-llama-3-70b}

HIP_VISIBLE_DEVICES=0,1,2,3 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 4 --master-addr 127.0.0.1 --master-port 29511 \
    -m dynamo_amd.workers --model "$MODEL" --tp-size 4 \
    --worker-type prefill --discovery "$DISC" \
    > "$NS/prefill.log" 2>&1 &
PIDS+=($!)
HIP_VISIBLE_DEVICES=4,5,6,7 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 4 --master-addr 127.0.0.1 --master-port 29512 \
    -m dynamo_amd.workers --model "$MODEL" --tp-size 4 \
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
