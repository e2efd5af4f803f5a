set -x
cd /root/repo
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_gpu_parity.py -x -q -k "rf" > gpurun_out/r2p2_pytest.log 2>&1
echo "pytest rc=$?" | tee gpurun_out/r2p2_status.txt
timeout 600 python -c "
import ctypes, json
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
core.set_coop_threshold(0)
res = {}
for mode in (0, 5, 0, 5):
    core._lib.hbls_set_verify_rf(mode)
    r = com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
    assert all(x == 1 for x in r), mode
    t = []
    for _ in range(2):
        com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
        t.append(round(core._lib.hbls_last_stage_ns(3)/1e6, 1))
    res.setdefault(mode, []).extend(t)
    print(mode, t, flush=True)
core._lib.hbls_set_verify_rf(-1); core.set_coop_threshold(-1)
json.dump({k: min(v) for k, v in res.items()}, open('gpurun_out/r2p2_fp6_ab.json','w'))
" > gpurun_out/r2p2_ab.log 2>&1
echo "ab rc=$?" | tee -a gpurun_out/r2p2_status.txt
tail -2 gpurun_out/r2p2_pytest.log; tail -6 gpurun_out/r2p2_ab.log
