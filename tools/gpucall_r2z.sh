set -x
cd /root/repo
mkdir -p gpurun_out
timeout 1800 python -c "
import time, json
from harmony_amd import core
from harmony_amd.stream import MultiStreamVerifier
from oracle import pyref as pr
core.init()
t_end = time.time() + 1560
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
import random
rng = random.Random(9)
sc = b''.join(pr.fr_serialize(rng.randrange(pr.R)) for _ in range(n))
cycles = 0
while time.time() < t_end:
    r = com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
    assert all(x == 1 for x in r)
    assert msv.final_check_collect()
    msv.reset_rounds(payloads)
    for lo in range(0, len(votes), 4096):
        msv.process(votes[lo:lo+4096])
    msv.final_check_submit()
    com.msm(sc)
    cycles += 1
assert msv.final_check_collect()
json.dump({'seconds': 1560, 'cycles': cycles, 'all_verified': True},
          open('gpurun_out/r2z_burnin.json','w'))
print('extended burn-in PASS', cycles, 'cycles')
" > gpurun_out/r2z_burnin.log 2>&1
echo "burnin=$?"
tail -2 gpurun_out/r2z_burnin.log
