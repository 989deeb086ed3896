#!/usr/bin/env bash
# Live smoke test — the CI check the reference runs on every push
# (/root/reference/.github/workflows/ci.yml:146-209): build, start the demo
# gRPC backend + the gateway as real processes, then curl tools/list and
# tools/call against localhost and verify the transcript.
#
# Usage: tools/live_smoke.sh [--frontend asyncio|native]
# Exit 0 = all checks pass.  Runs CPU-only (no GPU required).
set -u
cd "$(dirname "$0")/.."
FRONTEND="${2:-asyncio}"
[ "${1:-}" = "--frontend" ] && FRONTEND="$2"

PY=${PYTHON:-python3}
HTTP_PORT=$((20000 + RANDOM % 20000))
GRPC_PORT=$((20000 + RANDOM % 20000))
LOG=$(mktemp -d)
trap 'kill $BACKEND_PID $GATEWAY_PID 2>/dev/null; wait 2>/dev/null; rm -rf "$LOG"' EXIT

echo "[smoke] starting hello-service backend on :$GRPC_PORT"
$PY - "$GRPC_PORT" > "$LOG/backend.log" 2>&1 <<'EOF' &
import sys, time
from examples.hello_service import serve
server, target = serve(f"127.0.0.1:{sys.argv[1]}")
print("READY", target, flush=True)
while True:
    time.sleep(1)
EOF
BACKEND_PID=$!

echo "[smoke] starting gateway on :$HTTP_PORT (frontend=$FRONTEND)"
$PY -m ggrmcp_amd --grpc-host 127.0.0.1 --grpc-port "$GRPC_PORT" \
    --http-port "$HTTP_PORT" --no-gpu --frontend "$FRONTEND" \
    > "$LOG/gateway.log" 2>&1 &
GATEWAY_PID=$!

fail() { echo "[smoke] FAIL: $1"; echo "--- gateway log ---"; tail -30 "$LOG/gateway.log"; exit 1; }

# wait for /health to go 200 (reflection discovery must complete)
for i in $(seq 1 60); do
  CODE=$(curl -s -o "$LOG/health.json" -w '%{http_code}' "http://127.0.0.1:$HTTP_PORT/health" || true)
  [ "$CODE" = "200" ] && break
  sleep 0.5
done
[ "$CODE" = "200" ] || fail "health never became 200 (last $CODE)"
grep -q '"healthy"' "$LOG/health.json" || fail "health payload: $(cat "$LOG/health.json")"
echo "[smoke] health ok"

# tools/list (ci.yml transcript shape)
curl -s -X POST "http://127.0.0.1:$HTTP_PORT/" -H 'Content-Type: application/json' \
  -d '{"jsonrpc":"2.0","id":1,"method":"tools/list"}' > "$LOG/list.json" || fail "tools/list curl"
grep -q 'hello_helloservice_sayhello' "$LOG/list.json" || fail "tool missing: $(cat "$LOG/list.json")"
echo "[smoke] tools/list ok"

# tools/call round trip (README.md:203-215 transcript)
curl -s -X POST "http://127.0.0.1:$HTTP_PORT/" -H 'Content-Type: application/json' \
  -d '{"jsonrpc":"2.0","id":2,"method":"tools/call","params":{"name":"hello_helloservice_sayhello","arguments":{"name":"World"}}}' \
  > "$LOG/call.json" || fail "tools/call curl"
grep -q 'Hello, World!' "$LOG/call.json" || fail "call result: $(cat "$LOG/call.json")"
grep -q '"isError": *false' "$LOG/call.json" || fail "isError: $(cat "$LOG/call.json")"
echo "[smoke] tools/call ok"

# error path: unknown tool -> -32601
curl -s -X POST "http://127.0.0.1:$HTTP_PORT/" -H 'Content-Type: application/json' \
  -d '{"jsonrpc":"2.0","id":3,"method":"tools/call","params":{"name":"nope","arguments":{}}}' \
  > "$LOG/err.json" || fail "error curl"
grep -q '\-32601' "$LOG/err.json" || fail "error code: $(cat "$LOG/err.json")"
echo "[smoke] error mapping ok"

echo "[smoke] PASS (frontend=$FRONTEND)"
