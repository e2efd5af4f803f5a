set -x
cd /root/repo
mkdir -p gpurun_out
for R in 16 32 64; do
  HBLS_STREAM_ROUNDS=$R timeout 600 python bench.py --mode stream --steps 3 --warmup 1 > gpurun_out/r2n_stream_R$R.json 2>/dev/null
  echo "R$R rc=$?" | tee -a gpurun_out/r2n_status.txt
done
python - << 'PYEOF'
import json
for R in (16, 32, 64):
    try:
        d = json.load(open(f'gpurun_out/r2n_stream_R{R}.json'))
        print(R, d['value'], 'latency', d['config']['round_latency_ms'])
    except Exception as e:
        print(R, 'err', e)
PYEOF
