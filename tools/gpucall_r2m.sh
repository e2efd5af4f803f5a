set -x
cd /root/repo
mkdir -p gpurun_out
for so in libhbls.so libhbls_cs.so; do
  HBLS_SO=/root/repo/harmony_amd/$so timeout 420 python -c "
import os, json, time
from harmony_amd import core
from oracle import pyref as pr
core.init()
n, batch = 4096, 131072
sks = b''.join(pr.fr_serialize(pr.synth_sk(i)) for i in range(n))
pks = core.batch_pk_from_sk(sks, n)
com = core.Committee(pks, n)
bm1 = bytes([0xFF]) * (n // 8)
msg = pr.construct_commit_payload(5, pr.synth_msg(5), 6)
sk_sum = sum(pr.synth_sk(i) for i in range(n)) % pr.R
sig = core.sign_hash(pr.fr_serialize(sk_sum), msg)
bms, sigs, msgs = bm1*batch, sig*batch, msg*batch
core.set_coop_threshold(0)
r = com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
assert all(x == 1 for x in r)
out = {}
for _ in range(3):
    com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
    st = [core._lib.hbls_last_stage_ns(i)/1e6 for i in range(4)]
    for k, v in zip(('mask','hash','decompress','verify'), st):
        out.setdefault(k, []).append(round(v,1))
best = {k: min(v) for k, v in out.items()}
so = os.environ['HBLS_SO'].split('/')[-1]
print(so, best, flush=True)
json.dump(best, open(f'gpurun_out/r2m_{so}.json','w'))
" >> gpurun_out/r2m_ab.log 2>&1
  echo "$so rc=$?" | tee -a gpurun_out/r2m_status.txt
done
cat gpurun_out/r2m_ab.log
