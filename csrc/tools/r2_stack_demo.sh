#!/bin/bash
# Round-2 full-stack demo on one MI355X (BASELINE config #3 shape, scaled
# to a single GPU): 2 Llama-3-8B engine replicas sharing the GPU + a
# remote KV cacheserver (CacheGen serde) + the router with prefix-aware
# routing, driven by the reference-style multi-round-QA harness.
# Output: gpurun_out/r2_stack_demo.json (harness summary + router metrics).
set -u
cd "$(dirname "$0")/../.."
OUT=gpurun_out
mkdir -p $OUT

python -m production_stack_amd.kvpool.cacheserver --port 9411 \
    --capacity-gb 4 > $OUT/r2demo_cacheserver.log 2>&1 &
CS=$!
sleep 1

start_engine() {
  local port=$1 seed=$2
  python -m production_stack_amd.engine.server llama-3-8b \
    --host 127.0.0.1 --port $port --served-model-name llama-3-8b \
    --max-num-seqs 64 --max-num-batched-tokens 2048 \
    --gpu-memory-utilization 0.33 --enable-prefix-caching \
    --async-scheduling --seed $seed \
    --cpu-offload-gb 4 --remote-kv-url 127.0.0.1:9411 \
    > $OUT/r2demo_engine_$port.log 2>&1 &
}
start_engine 8101 1
E1=$!
start_engine 8102 2
E2=$!

for p in 8101 8102; do
  for i in $(seq 1 120); do
    curl -sf http://127.0.0.1:$p/health > /dev/null && break
    sleep 2
  done
done

python -m production_stack_amd.router.app \
  --host 127.0.0.1 --port 8100 \
  --service-discovery static \
  --static-backends http://127.0.0.1:8101,http://127.0.0.1:8102 \
  --static-models llama-3-8b,llama-3-8b \
  --routing-logic prefixaware \
  > $OUT/r2demo_router.log 2>&1 &
R=$!
for i in $(seq 1 30); do
  curl -sf http://127.0.0.1:8100/health > /dev/null && break
  sleep 1
done

python benchmarks/multi-round-qa/multi_round_qa.py \
  --base-url http://127.0.0.1:8100 --model llama-3-8b \
  --num-users 32 --num-rounds 4 --qps 12 \
  --shared-system-prompt 1000 --user-history-prompt 2000 \
  --question-len 100 --answer-len 100 --time-limit 75 \
  --output $OUT/r2_stack_demo.json 2> $OUT/r2demo_harness.log

echo "---- harness summary ----"
tail -5 $OUT/r2demo_harness.log
echo "---- router metrics (selected) ----"
curl -s http://127.0.0.1:8100/metrics | grep -E \
  "healthy_pods|num_requests|ttft|prefill|decode" | head -12
echo "---- engine KV pool ----"
curl -s http://127.0.0.1:8101/metrics | grep -E "offload|remote" | head -6

kill $R $E1 $E2 $CS 2>/dev/null
wait 2>/dev/null
