set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/r2i_pytest.log 2>&1
echo "pytest rc=$?" | tee gpurun_out/r2i_status.txt
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
    t0 = time.perf_counter(); core.msm_g1(pks, sc, n); tp = time.perf_counter()-t0
    o = ctypes.create_string_buffer(48)
    assert lib.hbls_msm_g1_naive(pks, sc, n, o) == 1
    assert o.raw == out_p, 'naive/pippenger mismatch'
    t0 = time.perf_counter(); lib.hbls_msm_g1_naive(pks, sc, n, o); tn = time.perf_counter()-t0
    res[n] = {'pippenger_ms': round(tp*1e3,1), 'naive_ms': round(tn*1e3,1), 'speedup': round(tn/tp,1)}
    print(n, res[n], flush=True)
json.dump(res, open('gpurun_out/r2i_msm_ab.json','w'), indent=1)
" > gpurun_out/r2i_msm.log 2>&1
echo "msm rc=$?" | tee -a gpurun_out/r2i_status.txt
timeout 900 python bench.py --steps 10 --warmup 3 > gpurun_out/r2i_bench.json 2> gpurun_out/r2i_bench.err
echo "bench rc=$?" | tee -a gpurun_out/r2i_status.txt
timeout 700 python bench.py --mode stream --steps 5 --warmup 1 > gpurun_out/r2i_stream.json 2> gpurun_out/r2i_stream.err
echo "stream rc=$?" | tee -a gpurun_out/r2i_status.txt
timeout 800 python bench.py --mode config4 --batch 65536 --steps 5 --warmup 1 > gpurun_out/r2i_config4.json 2> gpurun_out/r2i_config4.err
echo "config4 rc=$?" | tee -a gpurun_out/r2i_status.txt
tail -2 gpurun_out/r2i_pytest.log; cat gpurun_out/r2i_msm_ab.json 2>/dev/null
for f in r2i_bench r2i_stream r2i_config4; do echo "== $f"; tail -c 400 gpurun_out/$f.json; echo; done
