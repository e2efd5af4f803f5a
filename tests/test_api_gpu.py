"""GPU tests of the Go-API mirrors (harmony_amd.bls / quorum) — these read
like the reference's own tests (quorom_test.go:421-552, mask_test.go) with
the crypto running through libhbls.so."""
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

pytestmark = pytest.mark.gpu

from oracle import pyref as pr  # noqa: E402


def _gpu_available():
    try:
        from harmony_amd import core
        return core.device_count() > 0
    except Exception:
        return False


if not os.environ.get("HBLS_FORCE_GPU_TESTS"):
    pytestmark = [pytest.mark.gpu,
                  pytest.mark.skipif(not _gpu_available(), reason="no AMD GPU")]


@pytest.fixture(scope="module")
def env():
    from harmony_amd import bls
    sks = [bls.SecretKey(pr.fr_serialize(pr.synth_sk(i))) for i in range(8)]
    wrappers = [bls.PublicKeyWrapper(sk.get_public_key()) for sk in sks]
    return bls, sks, wrappers


def test_sign_verify_roundtrip_api(env):
    bls, sks, ws = env
    msg = pr.construct_commit_payload(1, pr.synth_msg(1), 1)
    sig = sks[0].sign_hash(msg)
    assert sig.verify_hash(ws[0].Object, msg)
    assert not sig.verify_hash(ws[1].Object, msg)


def test_aggregate_four_sigs(env):
    """quorom_test.go:421-472: 4-signer aggregate verifies against the
    summed public key."""
    bls, sks, ws = env
    msg = pr.synth_msg(2)
    sigs = [sk.sign_hash(msg) for sk in sks[:4]]
    agg = bls.aggregate_sig(sigs)
    pub = bls.PublicKey(ws[0].Bytes)
    for w in ws[1:4]:
        pub.add(w.Object)
    assert agg.verify_hash(pub, msg)
    # remove one signer's key -> reject (Sub path, mask.go:126-130)
    pub.sub(ws[3].Object)
    assert not agg.verify_hash(pub, msg)


def test_invalid_aggregate_multiplicity_api(env):
    """quorom_test.go:503-552 TestInvalidAggregateSig: duplicated signature
    must fail against the once-counted key sum."""
    bls, sks, ws = env
    msg = pr.synth_msg(3)
    sigs = [sk.sign_hash(msg) for sk in sks[:3]]
    agg = bls.aggregate_sig(sigs + [sks[0].sign_hash(msg)])
    pub = bls.PublicKey(ws[0].Bytes)
    pub.add(ws[1].Object).add(ws[2].Object)
    assert not agg.verify_hash(pub, msg)
    pub.add(ws[0].Object)      # count key 0 twice
    assert agg.verify_hash(pub, msg)


def test_mask_aggregate_public_api(env):
    """mask.go SetMask/SetBit + AggregatePublic through the GPU table."""
    bls, sks, ws = env
    m = bls.Mask(ws)
    m.set_bit(1, True)
    m.set_bit(4, True)
    m.set_key(ws[6].Bytes, True)
    agg = m.AggregatePublic
    expect = bls.PublicKey(ws[1].Bytes)
    expect.add(ws[4].Object).add(ws[6].Object)
    assert agg == expect
    # verify an aggregate signature against the mask
    msg = pr.synth_msg(4)
    sig = bls.aggregate_sig([sks[i].sign_hash(msg) for i in (1, 4, 6)])
    assert sig.verify_hash(m.AggregatePublic, msg)
    assert m.agg_verify(m.mask(), sig, msg)


def test_decider_flow_api(env):
    """AddNewVote -> quorum -> AggregateVotes -> verify_seal
    (quorum.go:354-394 + one-node-staked-vote.go:58-133 shape)."""
    bls, sks, ws = env
    from harmony_amd.quorum import COMMIT, Decider
    d = Decider(ws)
    block_hash = pr.synth_msg(9)
    payload = pr.construct_commit_payload(9, block_hash, 2)
    for i in range(6):
        d.add_new_vote(COMMIT, [ws[i]], sks[i].sign_hash(payload),
                       block_hash, 9, verify_payload=payload)
    assert d.is_quorum_achieved(COMMIT)
    agg = d.aggregate_votes(COMMIT)
    bm = bytearray(1)
    for i in range(6):
        bm[0] |= 1 << i
    assert d.verify_seal(bytes(bm), agg, payload)
    # bad vote rejected at AddNewVote
    with pytest.raises(ValueError):
        d.add_new_vote(COMMIT, [ws[7]], sks[6].sign_hash(payload),
                       block_hash, 9, verify_payload=payload)
    # double vote rejected
    with pytest.raises(ValueError):
        d.add_new_vote(COMMIT, [ws[0]], sks[0].sign_hash(payload),
                       block_hash, 9, verify_payload=payload)


def test_vrf_primitive_api(env):
    """crypto/vrf/bls/bls_vrf.go:63-101: VRF = SignHash/VerifyHash over a
    SHA-256 digest — same primitive, 32B message."""
    import hashlib
    bls, sks, ws = env
    digest = hashlib.sha256(b"vrf-alpha").digest()
    proof = sks[2].sign_hash(digest)
    assert proof.verify_hash(ws[2].Object, digest)
    assert not proof.verify_hash(ws[2].Object, hashlib.sha256(b"other").digest())


def test_vrf_evaluate_proof_to_hash(env):
    """crypto/vrf/bls/bls_vrf.go:62-101 mirror: Evaluate
    produces (beta, pi); ProofToHash accepts pi and returns beta; tampered
    alpha or pi rejects; beta/pi match the oracle's CPU derivation."""
    import hashlib
    from harmony_amd import vrf
    from oracle import capi
    bls, sks, ws = env
    signer = vrf.new_vrf_signer(sks[0])
    verifier = vrf.new_vrf_verifier(ws[0].Object)
    alpha = b"vrf-alpha-0"
    beta, pi = signer.evaluate(alpha)
    assert verifier.proof_to_hash(alpha, pi) == beta
    # oracle parity: pi = SignHash(sk, sha256(alpha)), beta = sha256(pi)
    pi_ref = capi.sign_hash(sks[0].serialize(), hashlib.sha256(alpha).digest())
    assert pi == pi_ref
    assert beta == hashlib.sha256(pi_ref).digest()
    # wrong alpha rejects
    with pytest.raises(vrf.ErrInvalidVRF):
        verifier.proof_to_hash(b"vrf-alpha-1", pi)
    # wrong verifier key rejects
    with pytest.raises(vrf.ErrInvalidVRF):
        vrf.new_vrf_verifier(ws[1].Object).proof_to_hash(alpha, pi)
    # empty / undecodable proofs reject like the reference's ErrInvalidVRF
    with pytest.raises(vrf.ErrInvalidVRF):
        verifier.proof_to_hash(alpha, b"")
    with pytest.raises(vrf.ErrInvalidVRF):
        verifier.proof_to_hash(alpha, b"\xff" * 96)


def test_committee_bitmap_length_validation(env):
    """a short/long bitmap must raise before crossing the ABI
    (the FFI memcpys ceil(n/8) bytes — OOB read otherwise)."""
    from harmony_amd import core
    bls, sks, ws = env
    c = core.Committee(b"".join(w.Bytes for w in ws), len(ws))
    msg = b"m" * 40
    with pytest.raises(ValueError):
        c.agg_verify(b"\xff\x00", b"\x00" * 96, msg)     # 8 keys -> 1 byte
    with pytest.raises(ValueError):
        c.mask_aggregate(b"")
    with pytest.raises(ValueError):
        c.batch_agg_verify(b"\xff", b"\x00" * 192, msg * 2, 40, 2)
    with pytest.raises(ValueError):
        c.mask_partials(b"\xff\xff\xff", 2)
