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

summarize() {  # full JSON to file, one-line summary to stdout
  python - "$1" <<'PY'
import json, re, sys
txt = open(sys.argv[1]).read()
m = re.search(r"\{.*\}", txt, re.S)
d = json.loads(m.group(0))
def g(*ks):
    v = d
    for k in ks: v = v.get(k, {})
    return v if not isinstance(v, dict) else None
print(f"out_tok/s={g('output_token_throughput_tps')} req/s={g('request_throughput_rps')} "
      f"ttft_ms p50={g('ttft_ms','median')} p99={g('ttft_ms','p99')} "
      f"tpot_ms p50={g('tpot_ms','median')}")
PY
}

python -m parallax_amd.benchmark.benchmark_serving \
  --base-url "http://127.0.0.1:$PORT" --backend completions \
  --num-prompts 512 --request-rate 16 --input-len 96 --output-len 128 \
  > gpurun_out/serve_e2e_rate16.log 2>&1
echo "rate16: $(summarize gpurun_out/serve_e2e_rate16.log)"

python -m parallax_amd.benchmark.benchmark_serving \
  --base-url "http://127.0.0.1:$PORT" --backend completions \
  --num-prompts 512 --request-rate inf --input-len 96 --output-len 128 \
  > gpurun_out/serve_e2e_rateinf.log 2>&1
echo "rateinf: $(summarize gpurun_out/serve_e2e_rateinf.log)"

kill $SRV 2>/dev/null
grep -c "tuned gemm\|capturing decode graph" gpurun_out/serve_e2e_server.log || true
