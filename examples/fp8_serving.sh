#!/usr/bin/env bash
# Register + load an fp8 serving variant (halved weight HBM, fp8 KV):
set -euo pipefail
API=${API:-http://localhost:8080}
H="Authorization: Bearer ${KEY:-admin-key}"
curl -sf -H "$H" -X POST $API/api/v1/local-models -d '{
  "name": "llama3-8b-fp8", "preset": "llama3-8b",
  "quantization": "fp8", "kv_cache_dtype": "fp8"}'
curl -sf -H "$H" -X POST $API/api/v1/local-models/llama3-8b-fp8/load
curl -sf -H "$H" $API/v1/chat/completions -d '{
  "model": "llama3-8b-fp8",
  "messages": [{"role": "user", "content": "hello"}]}'
