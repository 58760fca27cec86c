#!/usr/bin/env bash
# grpcurl demo against a running server (default plaintext port 8033), the
# counterpart of the reference's examples/inference.sh.
set -euo pipefail
HOST="${1:-localhost:8033}"

# health
grpcurl -plaintext "$HOST" grpc.health.v1.Health/Check

# model info
grpcurl -plaintext -d '{}' "$HOST" fmaas.GenerationService/ModelInfo

# unary generation
grpcurl -plaintext -d '{
  "requests": [{"text": "The answer to life is"}],
  "params": {"stopping": {"max_new_tokens": 16}}
}' "$HOST" fmaas.GenerationService/Generate

# guided decoding: regex-constrained output
grpcurl -plaintext -d '{
  "requests": [{"text": "Give me a number:"}],
  "params": {"stopping": {"max_new_tokens": 8},
             "decoding": {"guided": {"regex": "[0-9]+"}}}
}' "$HOST" fmaas.GenerationService/Generate
