#!/usr/bin/env bash
# Org runtime walkthrough: positions, bots, streams, message fan-out.
# (reference "helix-org" Bots/Positions/Streams graph)
set -euo pipefail
API=${API:-http://localhost:8080}
KEY=${KEY:-admin-key}
H="Authorization: Bearer $KEY"

OID=$(curl -sf -H "$H" -X POST $API/api/v1/organizations \
      -d '{"name": "acme"}' | python3 -c 'import sys,json;print(json.load(sys.stdin)["id"])')
POS=$(curl -sf -H "$H" -X POST $API/api/v1/organizations/$OID/positions \
      -d '{"name": "support", "system_prompt": "You answer customer questions tersely."}' \
      | python3 -c 'import sys,json;print(json.load(sys.stdin)["id"])')
BOT=$(curl -sf -H "$H" -X POST $API/api/v1/organizations/$OID/bots \
      -d "{\"name\": \"helper\", \"position_id\": \"$POS\"}" \
      | python3 -c 'import sys,json;print(json.load(sys.stdin)["id"])')
STRM=$(curl -sf -H "$H" -X POST $API/api/v1/organizations/$OID/streams \
      -d '{"name": "general"}' | python3 -c 'import sys,json;print(json.load(sys.stdin)["id"])')
curl -sf -H "$H" -X POST $API/api/v1/bots/$BOT/subscribe \
     -d "{\"stream_id\": \"$STRM\"}" > /dev/null
echo "posting to #general; subscribed bots reply:"
curl -sf -H "$H" -X POST $API/api/v1/streams/$STRM/messages \
     -d '{"text": "What is our refund policy?"}' | python3 -m json.tool
