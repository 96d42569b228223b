#!/bin/bash
# Flagship-load serving benchmark THROUGH the framework stack
# (VERDICT r1 #2): worker CLI + frontend + loadgen at ISL 8192 / conc 16
# on llama-3-70b, 1 GPU. Compare loadgen itl_p50_ms against bench.py's
# direct-engine ms_per_step — the serving stack must hold the same rate.
# Usage (GPU box): bash benchmarks/serve_flagship.sh [OSL] [REQUESTS]
set -u
OSL=${1:-256}
REQS=${2:-32}
OUT=${OUT:-gpurun_out/serve_flagship}
mkdir -p "$OUT"
DISC="file:/tmp/serve_flagship_disc"
rm -rf /tmp/serve_flagship_disc

python -m dynamo_amd.workers --model "${MODEL:-llama-3-70b}" --discovery "$DISC" \
    --max-num-seqs 32 --max-batched-tokens 8192 --max-model-len 16384 \
    --page-size 64 --no-prefix-caching \
    > "$OUT/worker.log" 2>&1 &
WPID=$!
for i in $(seq 1 240); do
  grep -q WORKER_READY "$OUT/worker.log" && break
  kill -0 $WPID 2>/dev/null || { echo "worker died"; tail -20 "$OUT/worker.log"; exit 1; }
  sleep 1
done
grep -q WORKER_READY "$OUT/worker.log" || { echo "worker not ready"; exit 1; }

python -m dynamo_amd.frontend --discovery "$DISC" --port 8031 \
    > "$OUT/frontend.log" 2>&1 &
FPID=$!
for i in $(seq 1 60); do
  grep -q FRONTEND_READY "$OUT/frontend.log" && break
  sleep 0.5
done

sleep 2  # model-card watch settle
timeout 900 python benchmarks/loadgen.py --url http://127.0.0.1:8031 \
    --isl "${ISL:-8192}" --osl "$OSL" --concurrency "${CONC:-16}" --requests "$REQS" \
    --vocab "${VOCAB:-128256}" > "$OUT/loadgen.json" 2> "$OUT/loadgen.err"
RC=$?
# the loadgen url port must match the frontend port
cat "$OUT/loadgen.json"
kill $FPID $WPID 2>/dev/null
wait 2>/dev/null
exit $RC
