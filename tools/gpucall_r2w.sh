set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r2w_pytest.log 2>&1
echo "tests=$?" | tee gpurun_out/r2w_status.txt
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> gpurun_out/r2w_pytest.log 2>&1
echo "smoke=$?" | tee -a gpurun_out/r2w_status.txt
timeout 700 python bench.py --steps 5 --warmup 2 > gpurun_out/r2w_bench.json 2>/dev/null
echo "bench=$?" | tee -a gpurun_out/r2w_status.txt
# 8-minute mixed burn-in incl. the async-check stream loop
timeout 600 python -c "
import time, json
from harmony_amd import core
from harmony_amd.stream import MultiStreamVerifier
from oracle import pyref as pr
core.init()
t_end = time.time() + 480
n = 4096
sks = b''.join(pr.fr_serialize(pr.synth_sk(i)) for i in range(n))
pks = core.batch_pk_from_sk(sks, n)
com = core.Committee(pks, n)
bm1 = bytes([0xFF]) * (n // 8)
msg = pr.construct_commit_payload(5, pr.synth_msg(5), 6)
sk_sum = sum(pr.synth_sk(i) for i in range(n)) % pr.R
sig = core.sign_hash(pr.fr_serialize(sk_sum), msg)
batch = 65536
bms, sigs, msgs = bm1*batch, sig*batch, msg*batch
npks = core.batch_pk_from_sk(b''.join(pr.fr_serialize(pr.synth_sk(i)) for i in range(256)), 256)
payloads = [pr.construct_commit_payload(r, pr.synth_msg(400+r), r) for r in range(16)]
msv = MultiStreamVerifier(npks, 256, payloads, window=100)
sks256 = [pr.fr_serialize(pr.synth_sk(i)) for i in range(256)]
vsigs = [core.batch_sign(b''.join(sks256), payloads[r]*256, len(payloads[r]), 256) for r in range(16)]
votes = [(r, i, vsigs[r][96*i:96*(i+1)]) for i in range(256) for r in range(16)]
cycles = 0
while time.time() < t_end:
    r = com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
    assert all(x == 1 for x in r)
    assert msv.final_check_collect()
    msv.reset_rounds(payloads)
    for lo in range(0, len(votes), 4096):
        msv.process(votes[lo:lo+4096])
    msv.final_check_submit()
    cycles += 1
assert msv.final_check_collect()
json.dump({'cycles': cycles, 'all_verified': True}, open('gpurun_out/r2w_burnin.json','w'))
print('burn-in PASS', cycles)
" > gpurun_out/r2w_burnin.log 2>&1
echo "burnin=$?" | tee -a gpurun_out/r2w_status.txt
tail -3 gpurun_out/r2w_pytest.log
python -c "import json; d=json.load(open('gpurun_out/r2w_bench.json')); print('bench', d['value'])" 2>/dev/null
tail -1 gpurun_out/r2w_burnin.log
