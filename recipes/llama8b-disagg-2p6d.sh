#!/bin/bash
# BASELINE config #3: Llama-3-8B disaggregated, 2 prefill + 6 decode
# workers, one GPU each; KV pages move prefill->decode over hipIpc/xGMI.
cd "$(dirname "$0")/.." || exit 1
source recipes/_lib.sh
MODEL=${MODEL:-llama-3-8b}
NP=${NP:-2}
ND=${ND:-6}
# GPUS: space-separated device list, one per worker (default 0..NP+ND-1);
# e.g. GPUS="0 0" NP=1 ND=1 co-locates both workers on one GPU for smoke
GPUS=${GPUS:-$(seq -s" " 0 $((NP + ND - 1)))}
# per-worker HBM fraction (lower it when co-locating workers on one GPU)
MEMFRAC=${MEMFRAC:-0.9}
read -ra GL <<< "$GPUS"

g=0
for i in $(seq 1 "$NP"); do
  : > "$NS/prefill$i.log"   # create before fork (the ready-wait globs)
  python -m dynamo_amd.workers --model "$MODEL" --device "cuda:${GL[$g]}" \
      --worker-type prefill --discovery "$DISC" \
      --gpu-mem-fraction "$MEMFRAC" \
      >> "$NS/prefill$i.log" 2>&1 &
  PIDS+=($!); g=$((g+1))
done
for i in $(seq 1 "$ND"); do
  : > "$NS/decode$i.log"
  python -m dynamo_amd.workers --model "$MODEL" --device "cuda:${GL[$g]}" \
      --worker-type decode --discovery "$DISC" \
      --gpu-mem-fraction "$MEMFRAC" \
      >> "$NS/decode$i.log" 2>&1 &
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
