set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/r2h_pytest.log 2>&1
echo "pytest rc=$?" | tee gpurun_out/r2h_status.txt
timeout 600 python -c "
import ctypes, json, time
from harmony_amd import core
from oracle import pyref as pr
core.init()
core._lib.hbls_set_verify_rf.argtypes = [ctypes.c_int]
n, batch = 4096, 131072
sks = b''.join(pr.fr_serialize(pr.synth_sk(i)) for i in range(n))
pks = core.batch_pk_from_sk(sks, n)
com = core.Committee(pks, n)
bm1 = bytes([0xFF]) * (n // 8)
msg = pr.construct_commit_payload(5, pr.synth_msg(5), 6)
sk_sum = sum(pr.synth_sk(i) for i in range(n)) % pr.R
sig = core.sign_hash(pr.fr_serialize(sk_sum), msg)
bms, sigs, msgs = bm1*batch, sig*batch, msg*batch
res = {}
core.set_coop_threshold(0)
for mode in (0, 4, 0):
    core._lib.hbls_set_verify_rf(mode)
    r = com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
    assert all(x == 1 for x in r)
    t = []
    for _ in range(3):
        com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
        t.append(core._lib.hbls_last_stage_ns(3)/1e6)
    res.setdefault(mode, []).append(round(min(t),1))
    print(mode, min(t), flush=True)
core._lib.hbls_set_verify_rf(-1); core.set_coop_threshold(-1)
json.dump(res, open('gpurun_out/r2h_2p_ab.json','w'))
" > gpurun_out/r2h_2p.log 2>&1
echo "2p rc=$?" | tee -a gpurun_out/r2h_status.txt
tail -3 gpurun_out/r2h_pytest.log; tail -5 gpurun_out/r2h_2p.log
