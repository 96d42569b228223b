#!/bin/bash
# BASELINE config #3: Llama-3-8B disaggregated, 2 prefill + 6 decode
# workers, one GPU each; KV pages move prefill->decode over hipIpc/xGMI.
cd "$(dirname "$0")/.." || exit 1
source recipes/_lib.sh
MODEL=${MODEL:-llama-3-8b}
NP=${NP:-2}
ND=${ND:-6}

g=0
for i in $(seq 1 "$NP"); do
  python -m dynamo_amd.workers --model "$MODEL" --device "cuda:$g" \
      --worker-type prefill --discovery "$DISC" \
      > "$NS/prefill$i.log" 2>&1 &
  PIDS+=($!); g=$((g+1))
done
for i in $(seq 1 "$ND"); do
  python -m dynamo_amd.workers --model "$MODEL" --device "cuda:$g" \
      --worker-type decode --discovery "$DISC" \
      > "$NS/decode$i.log" 2>&1 &
  PIDS+=($!); g=$((g+1))
done
for f in "$NS"/prefill*.log "$NS"/decode*.log; do
  wait_marker "$f" WORKER_READY 240 || exit 1
done

python -m dynamo_amd.frontend --discovery "$DISC" --port "$PORT" \
    > "$NS/frontend.log" 2>&1 &
PIDS+=($!)
wait_marker "$NS/frontend.log" FRONTEND_READY 60 || exit 1
sleep 2
maybe_check "${1:-}"
