# shared recipe helpers
set -u
NS=${NS:-/tmp/dynamo_amd_recipe_$$}
DISC="file:${NS}/disc"
PORT=${DYN_HTTP_PORT:-8000}
PIDS=()
cleanup() { for p in "${PIDS[@]:-}"; do kill "$p" 2>/dev/null; done; wait 2>/dev/null; rm -rf "$NS"; }
trap cleanup EXIT INT TERM
mkdir -p "$NS"

wait_marker() { # file marker timeout_s
  for _ in $(seq 1 $(( ${3:-180} * 2 ))); do
    grep -q "$2" "$1" 2>/dev/null && return 0
    sleep 0.5
  done
  echo "timeout waiting for $2 in $1" >&2; tail -20 "$1" >&2; return 1
}

maybe_check() { # run --check self-test then exit
  if [ "${1:-}" = "--check" ]; then
    curl -fsS "http://127.0.0.1:${PORT}/health" >/dev/null && echo "CHECK OK"
    exit $?
  fi
  echo "serving on http://127.0.0.1:${PORT}/v1  (ctrl-c to stop)"
  wait
}
