#!/usr/bin/env bash
# Local e2e: start the three process shapes (apiserver, webhook, controller)
# and run the full-lifecycle test suites against them, mirroring the
# reference's hack/kind-with-registry.sh + e2e flow without a cluster.
set -euo pipefail
cd "$(dirname "$0")/.."

python -m agac.cli apiserver --port 18101 &
API_PID=$!
python -m agac.cli webhook --no-ssl --port 18543 &
WEBHOOK_PID=$!
trap 'kill $API_PID $WEBHOOK_PID 2>/dev/null || true' EXIT

for i in $(seq 50); do
  curl -fsS http://127.0.0.1:18101/healthz >/dev/null 2>&1 && break
  sleep 0.2
done
curl -fsS http://127.0.0.1:18543/healthz

python -m pytest tests/test_http_e2e.py tests/test_full_e2e_scenarios.py \
  tests/test_k8s_wire.py tests/test_cli_processes.py -q "$@"
echo "e2e OK"
