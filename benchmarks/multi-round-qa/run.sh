#!/bin/bash
# Canonical multi-round QA sweep (reference run.sh parity):
#   warmup wave, then NUM_USERS x NUM_ROUNDS at each QPS point.
# Usage: ./run.sh <model> <base_url> [output_prefix]
set -euo pipefail

MODEL="${1:?model name}"
BASE_URL="${2:?base url}"
OUT="${3:-results}"

NUM_USERS=320
NUM_ROUNDS=10
SYSTEM_PROMPT=1000   # tokens
CHAT_HISTORY=20000   # tokens
ANSWER_LEN=100       # tokens
TIME_PER_POINT=100   # seconds

# warmup: touch caches with a large user wave
python3 "$(dirname "$0")/multi_round_qa.py" \
  --base-url "$BASE_URL" --model "$MODEL" \
  --num-users 400 --num-rounds 1 --qps 2.0 \
  --shared-system-prompt "$SYSTEM_PROMPT" \
  --user-history-prompt "$CHAT_HISTORY" \
  --answer-len "$ANSWER_LEN" --time-limit 60 \
  --output "${OUT}_warmup.json" || true

for QPS in 0.1 0.5 0.9 1.3 1.7 2.1 2.5 2.9 3.3 3.7 4.1; do
  echo "=== QPS $QPS ==="
  python3 "$(dirname "$0")/multi_round_qa.py" \
    --base-url "$BASE_URL" --model "$MODEL" \
    --num-users "$NUM_USERS" --num-rounds "$NUM_ROUNDS" --qps "$QPS" \
    --shared-system-prompt "$SYSTEM_PROMPT" \
    --user-history-prompt "$CHAT_HISTORY" \
    --answer-len "$ANSWER_LEN" --time-limit "$TIME_PER_POINT" \
    --output "${OUT}_qps${QPS}.json"
done
