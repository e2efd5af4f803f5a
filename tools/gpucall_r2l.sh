set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/r2l_pytest.log 2>&1
echo "pytest rc=$?" | tee gpurun_out/r2l_status.txt
timeout 420 python -c "
import ctypes, json, time, random, sys
from harmony_amd import core
from oracle import pyref as pr
core.init()
lib = core._lib
lib.hbls_msm_g1_naive.argtypes = [ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t, ctypes.c_char_p]
res = {}
for n in (4096, 65536):
    sks = b''.join(pr.fr_serialize(pr.synth_sk(i)) for i in range(n))
    pks = core.batch_pk_from_sk(sks, n)
    com = core.Committee(pks, n)
    rng = random.Random(7)
    sc = b''.join(pr.fr_serialize(rng.randrange(pr.R)) for _ in range(n))
    o = ctypes.create_string_buffer(48)
    for fn, key in ((lambda: core.msm_g1(pks, sc, n), 'full'),
                    (lambda: com.msm(sc), 'committee'),
                    (lambda: lib.hbls_msm_g1_naive(pks, sc, n, o), 'naive')):
        fn()
        ts = []
        for _ in range(3):
            t0 = time.perf_counter(); fn(); ts.append(time.perf_counter()-t0)
        res.setdefault(n, {})[key+'_ms'] = round(min(ts)*1e3, 1)
    assert com.msm(sc) == core.msm_g1(pks, sc, n) == o.raw
    print(n, res[n], flush=True)
json.dump(res, open('gpurun_out/r2l_msm.json','w'), indent=1)
" > gpurun_out/r2l_msm.log 2>&1
echo "msm rc=$?" | tee -a gpurun_out/r2l_status.txt
timeout 900 python bench.py --steps 10 --warmup 3 > gpurun_out/r2l_bench.json 2> gpurun_out/r2l_bench.err
echo "bench rc=$?" | tee -a gpurun_out/r2l_status.txt
tail -2 gpurun_out/r2l_pytest.log; cat gpurun_out/r2l_msm.json 2>/dev/null; echo; python -c "import json; d=json.load(open('gpurun_out/r2l_bench.json')); print(d['value'], d['roofline']['stages_ms_per_launch'])" 2>/dev/null
