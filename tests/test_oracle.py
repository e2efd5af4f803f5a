"""C oracle vs pyref differential tests + semantic ports of the reference's
own round-trip tests (crypto/bls/mask_test.go, consensus/quorum/quorom_test.go,
consensus/signature/signature_test.go)."""
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from oracle import pyref as pr


def sk_bytes(i):
    return pr.fr_serialize(pr.synth_sk(i))


@pytest.fixture(scope="module")
def keys(oracle_lib):
    n = 16
    sks = [sk_bytes(i) for i in range(n)]
    pks = [oracle_lib.pk_from_sk(s) for s in sks]
    return sks, pks


# ---------------------------------------------------------------- hash-to-G2
@pytest.mark.parametrize("mlen", [32, 40, 48])
def test_hash_to_g2_c_vs_pyref(oracle_lib, mlen):
    for j in range(3):
        msg = (pr.synth_msg(j) * 2)[:mlen]
        c = oracle_lib.hash_to_g2(msg)
        p = pr.g2_serialize(pr.hash_to_g2(msg))
        assert c == p


def test_hash_to_g2_cofactor_modes(oracle_lib):
    msg = pr.synth_msg(42)
    fast = oracle_lib.hash_to_g2(msg)
    oracle_lib.set_g2_cofactor_mode(False)
    try:
        full = oracle_lib.hash_to_g2(msg)
    finally:
        oracle_lib.set_g2_cofactor_mode(True)
    assert fast == pr.g2_serialize(pr.hash_to_g2(msg, fast_cofactor=True))
    assert full == pr.g2_serialize(pr.hash_to_g2(msg, fast_cofactor=False))
    assert fast != full  # the two mcl modes give different subgroup points


# ---------------------------------------------------------------- sign/verify
def test_sign_bitexact_and_verify(oracle_lib, keys):
    sks, pks = keys
    msg = pr.synth_msg(0)
    sig = oracle_lib.sign_hash(sks[0], msg)
    assert sig == pr.g2_serialize(pr.sign_hash(pr.synth_sk(0), msg))
    assert oracle_lib.verify_hash(pks[0], sig, msg)
    assert not oracle_lib.verify_hash(pks[0], sig, pr.synth_msg(1))
    assert not oracle_lib.verify_hash(pks[1], sig, msg)


def test_verify_commit_payload_lengths(oracle_lib, keys):
    """40B (pre-staking) and 48B (staking) commit payloads
    (consensus/signature/signature_test.go semantics)."""
    sks, pks = keys
    h = pr.synth_msg(9)
    p40 = pr.construct_commit_payload(123, h, 7, staking=False)
    p48 = pr.construct_commit_payload(123, h, 7, staking=True)
    assert len(p40) == 40 and len(p48) == 48
    assert p48[:40] == p40
    for payload in (p40, p48):
        sig = oracle_lib.sign_hash(sks[2], payload)
        assert oracle_lib.verify_hash(pks[2], sig, payload)
    assert oracle_lib.sign_hash(sks[2], p40) != oracle_lib.sign_hash(sks[2], p48)


def test_identity_edge_cases(oracle_lib):
    """herumi: zero pub + zero sig accepts; one-sided zero rejects."""
    msg = pr.synth_msg(0)
    z48, z96 = bytes(48), bytes(96)
    assert oracle_lib.verify_hash(z48, z96, msg)
    sig = oracle_lib.sign_hash(sk_bytes(0), msg)
    assert not oracle_lib.verify_hash(z48, sig, msg)
    pk = oracle_lib.pk_from_sk(sk_bytes(0))
    assert not oracle_lib.verify_hash(pk, z96, msg)


def test_deserialize_rejects(oracle_lib):
    # x >= p must be rejected: p's top limb starts 0x1a01..., so x with byte47=0x7f is >= p
    bad = bytearray(48)
    bad[47] = 0x7F
    assert not oracle_lib.g1_check(bytes(bad))
    # not on curve / not in subgroup: flip a byte of a valid key
    pk = bytearray(oracle_lib.pk_from_sk(sk_bytes(0)))
    pk[0] ^= 1
    # either not-on-curve or subgroup reject; must not accept
    assert not oracle_lib.g1_check(bytes(pk)) or True  # may accidentally be valid? must check:
    # stronger: a curve point NOT in the subgroup must be rejected
    from oracle.pyref import P, B1, fp_sqrt, g1_in_subgroup, g1_serialize
    x = 2
    while True:
        y = fp_sqrt((x * x * x + B1) % P)
        if y is not None and not g1_in_subgroup((x, y)):
            ser = g1_serialize((x, y))
            break
        x += 1
    assert not oracle_lib.g1_check(ser)


def test_sk_reject_ge_r(oracle_lib):
    bad = (pr.R).to_bytes(32, "little")
    with pytest.raises(ValueError):
        oracle_lib.pk_from_sk(bad)


# ---------------------------------------------------------------- mask / aggregate
def test_mask_set_clear_incremental(oracle_lib, keys):
    """Mask.SetMask add/sub deltas (crypto/bls/mask.go:113-134): flipping bits
    on and off keeps AggregatePublic == sum of set keys."""
    sks, pks = keys
    n = len(pks)
    comm_pks = b"".join(pks)
    import random
    rng = random.Random(42)
    comm = oracle_lib.Committee(comm_pks, n)
    ref = {i: pr.g1_deserialize(pks[i]) for i in range(n)}
    for _ in range(4):
        bits = [rng.randint(0, 1) for _ in range(n)]
        bm = bytearray((n + 7) // 8)
        acc = None
        for i, b in enumerate(bits):
            if b:
                bm[i >> 3] |= 1 << (i & 7)
                acc = pr.g1_add(acc, ref[i])
        got = comm.mask_aggregate(bytes(bm))
        assert got == pr.g1_serialize(acc)


def test_add_sub_roundtrip(oracle_lib, keys):
    """PublicKey.Add then Sub returns the original (mask.go SetBit enable/disable)."""
    _, pks = keys
    s = oracle_lib.g1_add(pks[0], pks[1])
    back = oracle_lib.g1_add(s, pks[1], sub=True)
    assert back == pks[0]


def test_aggregate_sig_multiplicity(oracle_lib, keys):
    """quorom_test.go:503-552 semantics: a duplicated signature makes the sum
    count it twice — it must NOT verify against the once-counted key set."""
    sks, pks = keys
    msg = pr.synth_msg(5)
    sigs = [oracle_lib.sign_hash(s, msg) for s in sks[:4]]
    agg = oracle_lib.aggregate_sigs(sigs)
    agg_pk = pks[0]
    for p in pks[1:4]:
        agg_pk = oracle_lib.g1_add(agg_pk, p)
    assert oracle_lib.verify_hash(agg_pk, agg, msg)
    agg_dup = oracle_lib.aggregate_sigs(sigs + [sigs[0]])
    assert not oracle_lib.verify_hash(agg_pk, agg_dup, msg)
    # but verifies if the key is also counted twice
    agg_pk2 = oracle_lib.g1_add(agg_pk, pks[0])
    assert oracle_lib.verify_hash(agg_pk2, agg_dup, msg)


def test_agg_verify_end_to_end(oracle_lib, keys):
    """DecodeSigBitmap -> SetMask -> VerifyHash loop (internal/chain/sig.go:37-49
    + engine.go:606-642) on a small committee."""
    sks, pks = keys
    n = len(pks)
    comm = oracle_lib.Committee(b"".join(pks), n)
    msg = pr.construct_commit_payload(55, pr.synth_msg(2), 3)
    bm = bytearray((n + 7) // 8)
    signers = [0, 3, 5, 7, 11, 12]
    for i in signers:
        bm[i >> 3] |= 1 << (i & 7)
    agg = oracle_lib.aggregate_sigs([oracle_lib.sign_hash(sks[i], msg) for i in signers])
    assert comm.agg_verify(bytes(bm), agg, msg)
    # wrong mask -> reject
    bm[0] ^= 2
    assert not comm.agg_verify(bytes(bm), agg, msg)


def test_batch_agg_verify(oracle_lib, keys):
    sks, pks = keys
    n = len(pks)
    comm = oracle_lib.Committee(b"".join(pks), n)
    bmlen = (n + 7) // 8
    batch = 4
    bitmaps, sigs, msgs = b"", b"", b""
    expect = []
    for j in range(batch):
        msg = pr.construct_commit_payload(j, pr.synth_msg(j), j)
        bm = bytearray(bmlen)
        signers = [i for i in range(n) if (i * 7 + j) % 3 != 0]
        for i in signers:
            bm[i >> 3] |= 1 << (i & 7)
        agg = oracle_lib.aggregate_sigs([oracle_lib.sign_hash(sks[i], msg) for i in signers])
        if j == 2:
            agg = oracle_lib.aggregate_sigs([oracle_lib.sign_hash(sks[i], msg) for i in signers[1:]])
            expect.append(0)
        else:
            expect.append(1)
        bitmaps += bytes(bm)
        sigs += agg
        msgs += msg
    res = comm.batch_agg_verify(bitmaps, sigs, msgs, 48, batch)
    assert res == expect


# ---------------------------------------------------------------- codec
def test_parse_commit_sig_and_bitmap():
    """internal/chain/sig.go:22-35: payload = 96B sig || bitmap."""
    sig = bytes(range(96))
    bm = b"\xf0\x0d"
    payload = sig + bm
    assert payload[:96] == sig and payload[96:] == bm


def test_keccak_vs_pyref(oracle_lib):
    for data in [b"", b"abc", b"x" * 200, pr.synth_msg(1)]:
        assert oracle_lib.keccak256(data) == pr.keccak256(data)
