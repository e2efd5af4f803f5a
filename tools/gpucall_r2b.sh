set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/r2b_pytest.log 2>&1
echo "pytest rc=$?" | tee gpurun_out/r2b_status.txt
timeout 240 python -c "
from harmony_amd import core
import json
core.init()
out = []
for blocks in (1024, 2048, 4096, 8192):
    for chains in (1, 3, 4):
        r = core._lib.hbls_fpmul_bench_waves(blocks, chains)
        out.append({'blocks64': blocks, 'waves_per_simd': blocks/1024.0, 'chains': chains, 'gmul_s': round(r/1e9, 2)})
        print(json.dumps(out[-1]), flush=True)
json.dump(out, open('gpurun_out/r2b_wavesweep.json','w'), indent=1)
" > gpurun_out/r2b_wavesweep.log 2>&1
echo "sweep rc=$?" | tee -a gpurun_out/r2b_status.txt
timeout 600 python bench.py --mode stream --steps 5 --warmup 1 > gpurun_out/r2b_stream.json 2> gpurun_out/r2b_stream.err
echo "stream rc=$?" | tee -a gpurun_out/r2b_status.txt
tail -3 gpurun_out/r2b_pytest.log
tail -1 gpurun_out/r2b_stream.json
