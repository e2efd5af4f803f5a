set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_gpu_parity.py -x -q -k "msm" > gpurun_out/r2j_pytest.log 2>&1
echo "pytest rc=$?" | tee gpurun_out/r2j_status.txt
timeout 420 python -c "
import ctypes, json, time, random
from harmony_amd import core
from oracle import pyref as pr
core.init()
lib = core._lib
lib.hbls_msm_g1_naive.argtypes = [ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p]
res = {}
for n in (4096, 16384, 65536):
    sks = b''.join(pr.fr_serialize(pr.synth_sk(i)) for i in range(n))
    pks = core.batch_pk_from_sk(sks, n)
    rng = random.Random(7)
    sc = b''.join(pr.fr_serialize(rng.randrange(pr.R)) for _ in range(n))
    out_p = core.msm_g1(pks, sc, n)
    ts = []
    for _ in range(3):
        t0 = time.perf_counter(); core.msm_g1(pks, sc, n); ts.append(time.perf_counter()-t0)
    tp = min(ts)
    o = ctypes.create_string_buffer(48)
    assert lib.hbls_msm_g1_naive(pks, sc, n, o) == 1
    assert o.raw == out_p, 'naive/pippenger mismatch'
    ts = []
    for _ in range(2):
        t0 = time.perf_counter(); lib.hbls_msm_g1_naive(pks, sc, n, o); ts.append(time.perf_counter()-t0)
    tn = min(ts)
    res[n] = {'pippenger_ms': round(tp*1e3,1), 'naive_ms': round(tn*1e3,1), 'speedup': round(tn/tp,1)}
    print(n, res[n], flush=True)
json.dump(res, open('gpurun_out/r2j_msm_ab.json','w'), indent=1)
" > gpurun_out/r2j_msm.log 2>&1
echo "msm rc=$?" | tee -a gpurun_out/r2j_status.txt
tail -2 gpurun_out/r2j_pytest.log; cat gpurun_out/r2j_msm_ab.json 2>/dev/null; tail -3 gpurun_out/r2j_msm.log
