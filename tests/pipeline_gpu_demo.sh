#!/bin/bash
# Two-stage engine pipeline on ONE MI355X (GPU box tool; BASELINE config #5
# de-risk): broker + two chained ENGINE workers
# (3B translate -> 2B format, stage-2 template interpolating stage-1's
# result) + CLI submit of N jobs + pipeline receive. Both engines share the
# GPU: gpu_memory_utilization caps TOTAL device usage, so the second
# worker's KV pool sizes itself around the first's footprint.
set -u
cd "$(dirname "$0")/.."
N=${1:-200}
MODEL1=${MODEL1:-llama-3.2-3b}
MODEL2=${MODEL2:-tower-plus-2b}
PORT=15679
export LLMQ_BROKER_URL=llmq://127.0.0.1:$PORT
export LLMQ_QUEUE_PREFETCH=256
export LLMQ_GPU_MEMORY_UTILIZATION=0.85
export LLMQ_MAX_NUM_SEQS=64
export LLMQ_MAX_MODEL_LEN=512

python -m llmq_amd broker serve --host 127.0.0.1 --port $PORT &
BROKER=$!
sleep 2

cat > /tmp/pipe2.yaml <<EOF
name: demo
stages:
  - name: translate
    worker: engine
    config:
      model: $MODEL1
      template: "Translate to Dutch: {text}"
      max_tokens: 48
      temperature: 0.0
  - name: format
    worker: engine
    config:
      model: $MODEL2
      template: "Format this: {translate_result}"
      max_tokens: 32
      temperature: 0.0
EOF

python -m llmq_amd worker pipeline /tmp/pipe2.yaml translate &
W1=$!
python -m llmq_amd worker pipeline /tmp/pipe2.yaml format &
W2=$!
sleep 5

python - "$N" <<'PYEOF'
import json, sys
n = int(sys.argv[1])
with open("/tmp/pipe_jobs.jsonl", "w") as f:
    for i in range(n):
        f.write(json.dumps({"id": f"pj-{i}", "text": f"sample sentence number {i}"}) + "\n")
PYEOF

T0=$(date +%s.%N)
python -m llmq_amd submit demo /tmp/pipe_jobs.jsonl --pipeline /tmp/pipe2.yaml
python -m llmq_amd receive --pipeline /tmp/pipe2.yaml --timeout 120 --limit "$N" \
    > /tmp/pipe_results.jsonl
T1=$(date +%s.%N)

python - "$N" <<'PYEOF'
import json, sys
n = int(sys.argv[1])
rows = [json.loads(l) for l in open("/tmp/pipe_results.jsonl")]
assert len(rows) == n, f"expected {n} results, got {len(rows)}"
ids = {r["id"] for r in rows}
assert ids == {f"pj-{i}" for i in range(n)}, "id set mismatch"
bad = [r for r in rows if not r.get("prompt", "").startswith("Format this: ")]
assert not bad, f"{len(bad)} rows missing the stage-2 template interpolation"
print(f"PIPELINE OK: {n}/{n} jobs through translate(3B)->format(2B), "
      f"stage-2 templates interpolated")
PYEOF
RC=$?
echo "elapsed: $(python -c "print(f'{$T1-$T0:.1f}s, {$N/($T1-$T0):.1f} jobs/s')")"

kill $W1 $W2 $BROKER 2>/dev/null
wait 2>/dev/null
exit $RC
