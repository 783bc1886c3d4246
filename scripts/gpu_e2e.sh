#!/usr/bin/env bash
# End-to-end orchestrator validation on a real MI355X box:
#   sky launch (GPU train task via torchrun) + sky jobs + sky serve.
# Writes a summary to gpurun_out/e2e.log. Run via gpurun.
set -x
cd "$(dirname "$0")/.."
export PATH=$PWD/bin:$PATH
export SKY_AMD_HOME=/tmp/sky_e2e_home
mkdir -p gpurun_out
LOG=gpurun_out/e2e.log
: > $LOG

py() { python -m skypilot_amd.cli "$@"; }

cat > /tmp/train1.yaml <<'YAML'
name: e2e-train
resources:
  accelerators: MI355X:1
run: |
  torchrun --standalone --nproc-per-node $SKYPILOT_NUM_GPUS_PER_NODE \
    --master-addr 127.0.0.1 \
    -m skypilot_amd.train.run --model llama-smoke --steps 12 \
    --micro-batch 2 --seq-len 512
YAML

echo "=== sky launch GPU train ===" >> $LOG
timeout 120 python -m skypilot_amd.cli launch /tmp/train1.yaml -c e2e --detach-run >> $LOG 2>&1
for i in $(seq 1 60); do
  ST=$(timeout 30 python -m skypilot_amd.cli queue e2e 2>/dev/null | tail -1)
  echo "$ST" | grep -qE "SUCCEEDED|FAILED" && break
  sleep 3
done
timeout 30 python -m skypilot_amd.cli queue e2e >> $LOG 2>&1
timeout 30 python -m skypilot_amd.cli logs e2e 1 --no-follow 2>/dev/null | tail -5 >> $LOG
echo "=== job-start latency (warm exec) ===" >> $LOG
T0=$(date +%s.%N)
timeout 60 python -m skypilot_amd.cli exec e2e 'echo warm-exec-done' >> $LOG 2>&1
T1=$(date +%s.%N)
python3 -c "print(f'warm exec wall: {$T1 - $T0:.2f}s')" >> $LOG
timeout 60 python -m skypilot_amd.cli down e2e >> $LOG 2>&1

echo "=== sky serve GPU inference ===" >> $LOG
cat > /tmp/serve1.yaml <<'YAML'
name: e2e-serve
service:
  readiness_probe:
    path: /health
    initial_delay_seconds: 240
  replicas: 1
resources:
  accelerators: MI355X:1
run: |
  python -m skypilot_amd.serve.entrypoint --model llama-smoke --port $PORT
YAML
timeout 120 python -m skypilot_amd.cli serve up /tmp/serve1.yaml -n e2esvc >> $LOG 2>&1
EP=$(python - <<'PYEOF'
import json, subprocess
out = subprocess.run(["python", "-m", "skypilot_amd.cli", "serve", "status", "e2esvc"],
                     capture_output=True, text=True, timeout=60).stdout
data = json.loads(out)
print(data[0]["endpoint"] if data else "")
PYEOF
)
echo "endpoint: $EP" >> $LOG
for i in $(seq 1 90); do
  R=$(timeout 10 curl -s $EP/health || true)
  echo "$R" | grep -q '"ok"' && break
  sleep 3
done
timeout 60 curl -s -X POST $EP/v1/completions -H 'Content-Type: application/json' \
  -d '{"prompt": "hello world", "max_tokens": 16}' >> $LOG 2>&1
echo >> $LOG
timeout 120 python -m skypilot_amd.cli serve down e2esvc >> $LOG 2>&1
echo "=== done ===" >> $LOG
tail -40 $LOG
