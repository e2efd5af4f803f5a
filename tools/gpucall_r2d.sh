set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/r2d_pytest.log 2>&1
echo "pytest rc=$?" | tee gpurun_out/r2d_status.txt
timeout 300 python tools/count_muls.py > gpurun_out/r2d_counts.log 2>&1
echo "counts rc=$?" | tee -a gpurun_out/r2d_status.txt
timeout 300 python -c "
import time, json, random
from harmony_amd import core
from oracle import pyref as pr
core.init()
res = {}
for n in (4096, 65536):
    sks = b''.join(pr.fr_serialize(pr.synth_sk(i)) for i in range(n))
    pks = core.batch_pk_from_sk(sks, n)
    rng = random.Random(7)
    sc = b''.join(pr.fr_serialize(rng.randrange(pr.R)) for _ in range(n))
    core.msm_g1(pks, sc, n)  # warm
    t0 = time.perf_counter(); core.msm_g1(pks, sc, n); t1 = time.perf_counter()
    res[n] = {'ms': round((t1-t0)*1e3, 2), 'kernel_ns': core.last_kernel_ns()}
    print(n, res[n], flush=True)
json.dump(res, open('gpurun_out/r2d_msm.json', 'w'))
" > gpurun_out/r2d_msm.log 2>&1
echo "msm rc=$?" | tee -a gpurun_out/r2d_status.txt
tail -3 gpurun_out/r2d_pytest.log; cat gpurun_out/r2d_counts.log | tail -20; cat gpurun_out/r2d_msm.log | tail -4
