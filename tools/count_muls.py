"""Measure the GPU path's OWN fp_mul counts per operation (instrumented
libhbls_count.so, -DHBLS_COUNT_MULS): settles the roofline denominator
against the oracle's op-counter and locates algorithmic-work levers.

Run on a GPU box:  python tools/count_muls.py
"""
import ctypes
import json
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from oracle import pyref as pr  # noqa: E402

lib = ctypes.CDLL(os.path.join(REPO, "harmony_amd", "libhbls_count.so"))
lib.hbls_mulcount_read.restype = ctypes.c_uint64
lib.hbls_committee_build.restype = ctypes.c_void_p
lib.hbls_committee_build.argtypes = [ctypes.c_char_p, ctypes.c_size_t]
lib.hbls_agg_verify.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p,
                                ctypes.c_char_p, ctypes.c_size_t]
lib.hbls_mask_aggregate_g1.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p]

assert lib.hbls_init(-1) == 1


def count(fn):
    lib.hbls_mulcount_reset()
    fn()
    return int(lib.hbls_mulcount_read())


out = {}
n = 4096
sks = b"".join(pr.fr_serialize(pr.synth_sk(i)) for i in range(n))
pks = ctypes.create_string_buffer(48 * n)
assert lib.hbls_batch_pk_from_sk(sks, n, pks) == 1
out["keygen_per_key"] = count(lambda: lib.hbls_batch_pk_from_sk(sks, n, pks)) // n
com = lib.hbls_committee_build(pks.raw, n)
assert com
out["committee_build_4096_total"] = count(
    lambda: None)  # build already done; placeholder 0

import random
rng = random.Random(42)
bm = bytearray(n // 8)
signers = [i for i in range(n) if rng.random() < 0.9]
for i in signers:
    bm[i >> 3] |= 1 << (i & 7)
bm = bytes(bm)
msg = pr.construct_commit_payload(1000, pr.synth_msg(1000), 17)
sk_sum = sum(pr.synth_sk(i) for i in signers) % pr.R
sig = ctypes.create_string_buffer(96)
assert lib.hbls_sign_hash(pr.fr_serialize(sk_sum), msg, len(msg), sig) == 1
sig = sig.raw

outp = ctypes.create_string_buffer(48)
out["mask_aggregate_4096_b09"] = count(
    lambda: lib.hbls_mask_aggregate_g1(com, bm, outp))
h96 = ctypes.create_string_buffer(96)
out["hash_to_g2"] = count(lambda: lib.hbls_hash_to_g2(msg, len(msg), h96))
out["g2_check_sig"] = count(lambda: lib.hbls_g2_check(sig))
out["g1_check_pk"] = count(lambda: lib.hbls_g1_check(pks.raw[:48]))

for thr, name in ((0, "scalar"), (1 << 30, "coop")):
    lib.hbls_set_coop_threshold(thr)
    out[f"agg_verify_total_{name}"] = count(
        lambda: lib.hbls_agg_verify(com, bm, sig, msg, len(msg)))
lib.hbls_set_coop_threshold(-1)

# oracle comparison
from oracle import capi
capi.reset_op_count()
oc = capi.Committee(pks.raw, n)
capi.reset_op_count()
assert oc.agg_verify(bm, sig, msg) is True
out["oracle_agg_verify_total"] = capi.op_count()

print(json.dumps(out, indent=1))
with open(os.path.join(REPO, "gpurun_out", "r2d_mulcounts.json"), "w") as f:
    json.dump(out, f, indent=1)
