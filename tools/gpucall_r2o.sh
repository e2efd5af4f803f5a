set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/r2o_pytest.log 2>&1
echo "pytest rc=$?" | tee gpurun_out/r2o_status.txt
for mode in on off; do
  env $([ $mode = off ] && echo HBLS_NO_BATCH_AFFINE=1) timeout 420 python -c "
import os, json
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
r = com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
assert all(x == 1 for x in r)
out = []
for _ in range(3):
    com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
    out.append(round(core._lib.hbls_last_stage_ns(3)/1e6, 1))
mode = 'off' if os.environ.get('HBLS_NO_BATCH_AFFINE') else 'on'
print('affine', mode, 'verify_ms', min(out), flush=True)
json.dump({'verify_ms': min(out)}, open(f'gpurun_out/r2o_affine_{mode}.json','w'))
" >> gpurun_out/r2o_ab.log 2>&1
  echo "$mode rc=$?" | tee -a gpurun_out/r2o_status.txt
done
timeout 900 python bench.py --steps 10 --warmup 3 > gpurun_out/r2o_bench.json 2> gpurun_out/r2o_bench.err
echo "bench rc=$?" | tee -a gpurun_out/r2o_status.txt
tail -2 gpurun_out/r2o_pytest.log; cat gpurun_out/r2o_ab.log
python -c "import json; d=json.load(open('gpurun_out/r2o_bench.json')); print(d['value'], d['roofline']['stages_ms_per_launch'])" 2>/dev/null
