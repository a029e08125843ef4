#!/bin/bash
# End-to-end HTTP serving benchmark (reference methodology:
# benchmark_serving against a live OpenAI server). Run on a GPU box:
#   bash scripts/serve_e2e.sh
set -u
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
PORT=3017
python -m parallax_amd.cli serve --model deepseek-r1-distill-llama-8b \
  --port $PORT --max-batch-size 768 --max-model-len 4096 \
  > gpurun_out/serve_e2e_server.log 2>&1 &
SRV=$!
trap 'kill $SRV 2>/dev/null' EXIT

for i in $(seq 1 360); do
  curl -sf "http://127.0.0.1:$PORT/health" >/dev/null 2>&1 && break
  sleep 2
  kill -0 $SRV 2>/dev/null || { echo "server died"; tail -30 gpurun_out/serve_e2e_server.log; exit 1; }
done
echo "server up after ~$((i*2))s"

python -m parallax_amd.benchmark.benchmark_serving \
  --base-url "http://127.0.0.1:$PORT" --backend completions \
  --num-prompts 512 --request-rate 16 --input-len 96 --output-len 128 \
  2>&1 | tail -25 | tee gpurun_out/serve_e2e_rate16.log

python -m parallax_amd.benchmark.benchmark_serving \
  --base-url "http://127.0.0.1:$PORT" --backend completions \
  --num-prompts 512 --request-rate inf --input-len 96 --output-len 128 \
  2>&1 | tail -25 | tee gpurun_out/serve_e2e_rateinf.log

kill $SRV 2>/dev/null
grep -c "tuned gemm\|capturing decode graph" gpurun_out/serve_e2e_server.log || true
