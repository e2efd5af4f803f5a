set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -x -q -k "not rf" > gpurun_out/r2f_pytest.log 2>&1
echo "pytest rc=$?" | tee gpurun_out/r2f_status.txt
timeout 600 python -c "
import os, json, time, random
from harmony_amd import core
from oracle import pyref as pr
core.init()
n = 4096
sks = b''.join(pr.fr_serialize(pr.synth_sk(i)) for i in range(n))
pks = core.batch_pk_from_sk(sks, n)
com = core.Committee(pks, n)
bm1 = bytes([0xFF]) * (n // 8)
msg = pr.construct_commit_payload(5, pr.synth_msg(5), 6)
sk_sum = sum(pr.synth_sk(i) for i in range(n)) % pr.R
sig = core.sign_hash(pr.fr_serialize(sk_sum), msg)
res = {}
for items in (16, 8):
    os.environ['HBLS_COOP_ITEMS'] = str(items)
    # env read is cached per process...: use the C getenv each call? coop_items caches static.
    # -> fork per setting
    pid = os.fork()
    if pid == 0:
        out = {}
        for batch in (16, 512, 2048, 4096, 8192):
            bms, sigs, msgs = bm1*batch, sig*batch, msg*batch
            r = com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
            assert all(x == 1 for x in r)
            ts = []
            for _ in range(3):
                t0 = time.perf_counter()
                com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
                ts.append(time.perf_counter()-t0)
            out[batch] = round(min(ts)*1e3, 1)
        # single-verify latency
        t0 = time.perf_counter(); core.verify_hash(pks[:48], core.sign_hash(sks[:32], msg), msg); t1 = time.perf_counter()
        out['single_ms'] = round((t1-t0)*1e3, 1)
        json.dump(out, open(f'gpurun_out/r2f_coop{items}.json','w'))
        os._exit(0)
    os.waitpid(pid, 0)
print(open('gpurun_out/r2f_coop16.json').read())
print(open('gpurun_out/r2f_coop8.json').read())
" > gpurun_out/r2f_coopab.log 2>&1
echo "coopab rc=$?" | tee -a gpurun_out/r2f_status.txt
timeout 600 python bench.py --mode stream --steps 5 --warmup 1 > gpurun_out/r2f_stream.json 2> gpurun_out/r2f_stream.err
echo "stream rc=$?" | tee -a gpurun_out/r2f_status.txt
tail -2 gpurun_out/r2f_pytest.log; tail -4 gpurun_out/r2f_coopab.log; tail -c 600 gpurun_out/r2f_stream.json
