set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python -m pytest tests/test_stream_gpu.py tests/test_api_gpu.py -x -q > gpurun_out/r2c_pytest.log 2>&1
echo "pytest rc=$?" | tee gpurun_out/r2c_status.txt
timeout 240 python -c "
from harmony_amd import core
import json
core.init()
out = []
for blocks in (1024, 2048, 8192):
    for chains in (3, 5, 6):
        r = core._lib.hbls_fpmul_bench_waves(blocks, chains)
        out.append({'blocks64': blocks, 'waves_per_simd': blocks/1024.0, 'chains': chains, 'gmul_s': round(r/1e9, 2)})
        print(json.dumps(out[-1]), flush=True)
json.dump(out, open('gpurun_out/r2c_wavesweep.json','w'), indent=1)
" > gpurun_out/r2c_wavesweep.log 2>&1
echo "sweep rc=$?" | tee -a gpurun_out/r2c_status.txt
timeout 900 python bench.py --steps 10 --warmup 3 > gpurun_out/r2c_bench.json 2> gpurun_out/r2c_bench.err
echo "bench rc=$?" | tee -a gpurun_out/r2c_status.txt
tail -2 gpurun_out/r2c_pytest.log; tail -c 2000 gpurun_out/r2c_bench.json
