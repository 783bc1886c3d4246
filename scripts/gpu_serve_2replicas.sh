set -x
cd /root/repo
export PATH=$PWD/bin:$PATH
export SKY_AMD_HOME=/tmp/sky_2rep
mkdir -p gpurun_out
LOG=gpurun_out/serve2.log
: > $LOG
cat > /tmp/svc2.yaml <<'YAML'
name: tworep
resources:
  accelerators: MI355X:0.5
service:
  replicas: 2
  readiness_probe:
    path: /health
    initial_delay_seconds: 240
run: |
  python -m skypilot_amd.serve.entrypoint --model llama-smoke \
    --port $PORT --max-batch 8
YAML
timeout 150 python -m skypilot_amd.cli serve up /tmp/svc2.yaml -n tworep >> $LOG 2>&1
EP=$(grep -o '"endpoint": "[^"]*"' $LOG | head -1 | cut -d'"' -f4)
# wait for 2 READY replicas
for i in $(seq 1 100); do
  N=$(timeout 60 python -m skypilot_amd.cli serve status tworep 2>/dev/null       | grep -c '"status": "READY"')
  [ "$N" -ge 3 ] && break   # service + 2 replicas
  sleep 3
done
echo "endpoint: $EP" >> $LOG
if [ -n "$EP" ]; then
  for i in 1 2 3 4; do
    curl -s -m 30 -X POST $EP/v1/completions \
      -H 'Content-Type: application/json' \
      -d '{"prompt": "hi", "max_tokens": 8}' >> $LOG 2>&1
    echo >> $LOG
  done
  echo "TWO-REPLICA-OK" >> $LOG
else
  echo "NEVER-READY" >> $LOG
  timeout 60 python -m skypilot_amd.cli serve status tworep >> $LOG 2>&1
fi
timeout 120 python -m skypilot_amd.cli serve down tworep >> $LOG 2>&1
rocm-smi --showmeminfo vram 2>/dev/null | head -6 >> $LOG
