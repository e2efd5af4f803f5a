#!/usr/bin/env python3
"""Generate tests/golden/oracle_kats.json — seeded known-answer vectors from
the CPU oracle, cross-checked against the independent pure-Python restatement
before committing (the reference ships no signature/hash-to-G2
vectors, so the build pins its own).

  python3 oracle/gen_kats.py
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from oracle import capi, pyref as pr  # noqa: E402


def main():
    out = {"comment": "seeded KATs from oracle/hbls_oracle.c, cross-checked vs oracle/pyref.py; "
                      "sk_i = SHA256('hbls-sk'||LE32(i)) mod r; msgs = commit payloads over "
                      "keccak('blk'||LE64(j)) block hashes (SURVEY.md §8d)"}
    sks = [pr.fr_serialize(pr.synth_sk(i)) for i in range(8)]
    out["sk_pk"] = []
    for i, sk in enumerate(sks):
        pk = capi.pk_from_sk(sk)
        assert pk == pr.g1_serialize(pr.get_public_key(pr.synth_sk(i)))
        out["sk_pk"].append({"sk": sk.hex(), "pk": pk.hex()})

    out["hash_to_g2"] = []
    for j, mlen in [(0, 32), (1, 40), (2, 48), (3, 48)]:
        msg = pr.construct_commit_payload(j, pr.synth_msg(j), j + 1)[:mlen]
        fast = capi.hash_to_g2(msg)
        capi.set_g2_cofactor_mode(False)
        full = capi.hash_to_g2(msg)
        capi.set_g2_cofactor_mode(True)
        assert fast == pr.g2_serialize(pr.hash_to_g2(msg, fast_cofactor=True))
        assert full == pr.g2_serialize(pr.hash_to_g2(msg, fast_cofactor=False))
        out["hash_to_g2"].append({"msg": msg.hex(), "fast": fast.hex(), "full_h2": full.hex()})

    out["signatures"] = []
    for i in range(4):
        msg = pr.construct_commit_payload(100 + i, pr.synth_msg(100 + i), i)
        sig = capi.sign_hash(sks[i], msg)
        assert sig == pr.g2_serialize(pr.sign_hash(pr.synth_sk(i), msg))
        out["signatures"].append({"sk": sks[i].hex(), "msg": msg.hex(), "sig": sig.hex()})

    # aggregate case: 5 signers of one payload + the masked key sum
    msg = pr.construct_commit_payload(7, pr.synth_msg(7), 9)
    signers = [0, 2, 3, 5, 6]
    agg_sig = capi.aggregate_sigs([capi.sign_hash(sks[i], msg) for i in signers])
    pks = [capi.pk_from_sk(s) for s in sks]
    comm = capi.Committee(b"".join(pks), len(pks))
    bm = bytearray(1)
    for i in signers:
        bm[0] |= 1 << i
    agg_pk = comm.mask_aggregate(bytes(bm))
    assert capi.verify_hash(agg_pk, agg_sig, msg)
    out["aggregate"] = {"msg": msg.hex(), "bitmap": bytes(bm).hex(),
                        "signers": signers, "agg_pk": agg_pk.hex(),
                        "agg_sig": agg_sig.hex(), "committee": [p.hex() for p in pks]}

    out["keccak256"] = [{"in": d.hex(), "out": capi.keccak256(d).hex()}
                        for d in (b"", b"abc", b"blk" + (7).to_bytes(8, "little"))]

    path = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                        "tests", "golden", "oracle_kats.json")
    with open(path, "w") as f:
        json.dump(out, f, indent=1)
    print("wrote", path)


if __name__ == "__main__":
    main()
