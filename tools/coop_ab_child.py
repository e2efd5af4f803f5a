"""one coop-items A/B point (fresh process; HBLS_COOP_ITEMS env set by parent)"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from harmony_amd import core           # noqa: E402
from oracle import pyref as pr         # noqa: E402

core.init()
n = 4096
sks = b"".join(pr.fr_serialize(pr.synth_sk(i)) for i in range(n))
pks = core.batch_pk_from_sk(sks, n)
com = core.Committee(pks, n)
bm1 = bytes([0xFF]) * (n // 8)
msg = pr.construct_commit_payload(5, pr.synth_msg(5), 6)
sk_sum = sum(pr.synth_sk(i) for i in range(n)) % pr.R
sig = core.sign_hash(pr.fr_serialize(sk_sum), msg)
out = {}
for batch in (16, 512, 2048, 4096, 8192):
    bms, sigs, msgs = bm1 * batch, sig * batch, msg * batch
    r = com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
    assert all(x == 1 for x in r)
    ts = []
    for _ in range(3):
        t0 = time.perf_counter()
        com.batch_agg_verify(bms, sigs, msgs, len(msg), batch)
        ts.append(time.perf_counter() - t0)
    out[str(batch)] = round(min(ts) * 1e3, 1)
t0 = time.perf_counter()
core.verify_hash(pks[:48], core.sign_hash(sks[:32], msg), msg)
out["single_ms"] = round((time.perf_counter() - t0) * 1e3, 1)
items = os.environ.get("HBLS_COOP_ITEMS", "auto")
json.dump(out, open(f"gpurun_out/r2g_coop{items}.json", "w"))
print(items, out)
