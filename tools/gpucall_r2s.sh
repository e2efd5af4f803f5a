set -x
cd /root/repo
mkdir -p gpurun_out
# sustained mixed burn-in: ~14 min of alternating production workloads
timeout 1020 python -c "
import time, json, random
from harmony_amd import core
from harmony_amd.stream import MultiStreamVerifier
from oracle import capi, pyref as pr
core.init()
t_end = time.time() + 840
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
rng = random.Random(1)
sc = b''.join(pr.fr_serialize(rng.randrange(pr.R)) for _ in range(n))
cycles = 0
while time.time() < t_end:
    r = com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
    assert all(x == 1 for x in r)
    msv.reset_rounds(payloads)
    for lo in range(0, len(votes), 4096):
        msv.process(votes[lo:lo+4096])
    assert msv.final_check_all()
    com.msm(sc)
    cycles += 1
    print('cycle', cycles, 'ok', flush=True)
json.dump({'cycles': cycles, 'all_verified': True}, open('gpurun_out/r2s_burnin.json','w'))
print('burn-in PASS', cycles, 'cycles')
" > gpurun_out/r2s_burnin.log 2>&1
echo "burnin rc=$?" | tee gpurun_out/r2s_status.txt
tail -3 gpurun_out/r2s_burnin.log
