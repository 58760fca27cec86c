#!/usr/bin/env bash
# End-to-end serving check on a GPU box: boot the dual-front-end server on
# the llama-1b synthetic preset, wait for gRPC health SERVING, then exercise
# Generate / GenerateStream / Tokenize / ModelInfo via examples/inference.py
# and /health + /v1/completions over HTTP.
set -u
cd "$(dirname "$0")/.."
OUT="${1:-gpurun_out/e2e.log}"
mkdir -p "$(dirname "$OUT")"

python -m vllm_tgis_adapter_amd --model-name llama-1b --dtype bfloat16 \
    --num-gpu-blocks 2048 --max-num-seqs 16 --grpc-port 8033 --port 8000 \
    > "$OUT.server" 2>&1 &
SRV=$!
trap 'kill $SRV 2>/dev/null' EXIT

ok=0
for i in $(seq 1 120); do
  if python -m vllm_tgis_adapter_amd.healthcheck --timeout 2 >/dev/null 2>&1; then
    ok=1; break
  fi
  sleep 2
done
if [ "$ok" != 1 ]; then
  echo "E2E FAIL: server never became healthy" | tee "$OUT"
  tail -30 "$OUT.server" >> "$OUT"
  exit 1
fi

{
  echo "== healthcheck OK"
  echo "== ModelInfo";  timeout 30 python examples/inference.py --model-info
  echo "== Tokenize";   timeout 30 python examples/inference.py --tokenize --text "hello world from MI355X"
  echo "== Generate";   timeout 60 python examples/inference.py --text "one two three" --max-new-tokens 8
  echo "== GenerateStream"; timeout 60 python examples/inference.py --streaming --text "a b c" --max-new-tokens 8
  echo "== Guided regex";   timeout 60 python examples/inference.py --text "number:" --max-new-tokens 6 --guided-regex "[ab]+"
  echo "== HTTP /health";   timeout 10 python -c "import urllib.request; print(urllib.request.urlopen('http://localhost:8000/health').status)"
  echo "== HTTP /v1/models"; timeout 10 python -c "import urllib.request; print(urllib.request.urlopen('http://localhost:8000/v1/models').read()[:200])"
  echo "== HTTP /metrics (first lines)"; timeout 10 python -c "import urllib.request; print(urllib.request.urlopen('http://localhost:8000/metrics').read()[:300])"
  echo "E2E PASS"
} > "$OUT" 2>&1
rc=$?
kill $SRV 2>/dev/null
exit $rc
