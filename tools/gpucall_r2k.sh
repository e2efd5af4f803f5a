set -x
cd /tmp && export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_gpu_parity.py -x -q -k "msm" > gpurun_out/r2k_pytest.log 2>&1
echo "pytest rc=$?" | tee gpurun_out/r2k_status.txt
cat > /tmp/msmprof.py << 'PYEOF'
import ctypes, json, time, random, sys
sys.path.insert(0, '/root/repo')
from harmony_amd import core
from oracle import pyref as pr
core.init()
lib = core._lib
res = {}
for n in (4096, 65536):
    sks = b''.join(pr.fr_serialize(pr.synth_sk(i)) for i in range(n))
    pks = core.batch_pk_from_sk(sks, n)
    com = core.Committee(pks, n)
    rng = random.Random(7)
    sc = b''.join(pr.fr_serialize(rng.randrange(pr.R)) for _ in range(n))
    core.msm_g1(pks, sc, n)
    com.msm(sc)
    t0 = time.perf_counter(); core.msm_g1(pks, sc, n); t1 = time.perf_counter()
    t2 = time.perf_counter(); com.msm(sc); t3 = time.perf_counter()
    res[n] = {'full_ms': round((t1-t0)*1e3,1), 'committee_ms': round((t3-t2)*1e3,1)}
    print(n, res[n], flush=True)
json.dump(res, open('gpurun_out/r2k_msm.json','w'))
PYEOF
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/r2k_prof -- python /tmp/msmprof.py > gpurun_out/r2k_msm.log 2>&1
echo "prof rc=$?" | tee -a gpurun_out/r2k_status.txt
tail -2 gpurun_out/r2k_pytest.log; tail -4 gpurun_out/r2k_msm.log
python3 - << 'PYEOF'
import sqlite3, glob, re
dbs = glob.glob('gpurun_out/r2k_prof/**/*_results.db', recursive=True)
if dbs:
    c = sqlite3.connect(dbs[0])
    tabs = {r[0] for r in c.execute("SELECT name FROM sqlite_master WHERE type='table'")}
    kd = next(t for t in tabs if t.startswith('rocpd_kernel_dispatch'))
    ks = next(t for t in tabs if t.startswith('rocpd_info_kernel_symbol'))
    for name, calls, tot, avg in c.execute(f"""
        SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6, AVG(d.end-d.start)/1e6
        FROM {kd} d JOIN {ks} s ON d.kernel_id = s.id
        WHERE s.display_name LIKE '%msm%' GROUP BY s.display_name ORDER BY 3 DESC"""):
        print(f"{re.sub(r'[(].*','',name):40s} {calls:3d} {tot:9.2f} {avg:8.3f}")
PYEOF
