set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python -m pytest tests/test_gpu_parity.py -x -q -k "rf or verify or msm" > gpurun_out/r2e_pytest.log 2>&1
echo "pytest rc=$?" | tee gpurun_out/r2e_status.txt
timeout 900 python -c "
import ctypes, json, time, random
from harmony_amd import core
from oracle import pyref as pr
core.init()
core._lib.hbls_set_verify_rf.argtypes = [ctypes.c_int]
n, batch = 4096, 65536
sks = b''.join(pr.fr_serialize(pr.synth_sk(i)) for i in range(n))
pks = core.batch_pk_from_sk(sks, n)
com = core.Committee(pks, n)
rng = random.Random(42)
import numpy as np
bits = np.random.default_rng(42).integers(0, 256, (batch, n), dtype=np.uint8) < 230
bms = np.packbits(bits, axis=1, bitorder='little').tobytes()
# one signer-sum signature per item is expensive to build; reuse a single
# (bm, sig, msg) replicated: same work per item, accept-all
bm1 = bms[:n//8]
idx = [i for i in range(n) if bm1[i>>3] >> (i&7) & 1]
msg = pr.construct_commit_payload(5, pr.synth_msg(5), 6)
sk_sum = sum(pr.synth_sk(i) for i in idx) % pr.R
sig = core.sign_hash(pr.fr_serialize(sk_sum), msg)
bms_r = bm1 * batch
sigs_r = sig * batch
msgs_r = msg * batch
res = {}
core.set_coop_threshold(0)
for mode in (0, 1, 2, 3):
    core._lib.hbls_set_verify_rf(mode)
    r = com.batch_agg_verify(bms_r, sigs_r, msgs_r, len(msg), batch)
    assert all(x == 1 for x in r), (mode, r[:5])
    t = []
    for _ in range(3):
        com.batch_agg_verify(bms_r, sigs_r, msgs_r, len(msg), batch)
        stages = [core._lib.hbls_last_stage_ns(i)/1e6 for i in range(4)]
        t.append(stages)
    best = min(t, key=lambda s: s[3])
    res[mode] = {'stages_ms': [round(x,1) for x in best], 'verify_ms': round(best[3],1)}
    print(mode, res[mode], flush=True)
core._lib.hbls_set_verify_rf(-1)
core.set_coop_threshold(-1)
json.dump(res, open('gpurun_out/r2e_rf_ab.json','w'), indent=1)
" > gpurun_out/r2e_rf_ab.log 2>&1
echo "ab rc=$?" | tee -a gpurun_out/r2e_status.txt
tail -3 gpurun_out/r2e_pytest.log; tail -8 gpurun_out/r2e_rf_ab.log
