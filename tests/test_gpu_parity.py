"""GPU parity tests: HIP product path vs CPU oracle, bit-exact on serialized
outputs and accept/reject-exact on verifies.  All tests here need an MI355X.

Seeded inputs per SURVEY.md §8d; sizes small enough that the oracle side
finishes in seconds, plus size-independent identities at config sizes."""
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
def core():
    from harmony_amd import core as c
    c.init(-1 if c.device_count() == 0 else 0)
    return c


@pytest.fixture(scope="module")
def capi(oracle_lib):
    return oracle_lib


def sk_bytes(i):
    return pr.fr_serialize(pr.synth_sk(i))


@pytest.fixture(scope="module")
def keys16(core, capi):
    n = 16
    sks = b"".join(sk_bytes(i) for i in range(n))
    pks = core.batch_pk_from_sk(sks, n)
    return sks, pks, n


# ---------------------------------------------------------------- primitives
def test_pk_from_sk_parity(core, capi, keys16):
    sks, pks, n = keys16
    for i in range(n):
        assert pks[48 * i:48 * i + 48] == capi.pk_from_sk(sks[32 * i:32 * i + 32])


def test_golden_vectors_on_gpu(core):
    import json
    here = os.path.dirname(os.path.abspath(__file__))
    vec = json.load(open(os.path.join(here, "golden", "sk_pk.json")))
    sks = b"".join(bytes.fromhex(e["sk"]) for e in vec)
    pks = core.batch_pk_from_sk(sks, len(vec))
    for i, e in enumerate(vec):
        assert pks[48 * i:48 * i + 48].hex() == e["pk"]


@pytest.mark.parametrize("mlen", [32, 40, 48])
def test_hash_to_g2_parity(core, capi, mlen):
    batch = 8
    msgs = b"".join((pr.synth_msg(j) * 2)[:mlen] for j in range(batch))
    got = core.batch_hash_to_g2(msgs, mlen, batch)
    for j in range(batch):
        exp = capi.hash_to_g2(msgs[mlen * j:mlen * (j + 1)])
        assert got[96 * j:96 * (j + 1)] == exp


def test_hash_to_g2_cofactor_modes(core, capi):
    msg = pr.synth_msg(11)
    for fast in (True, False):
        core.set_g2_cofactor_mode(fast)
        capi.set_g2_cofactor_mode(fast)
        try:
            assert core.hash_to_g2(msg) == capi.hash_to_g2(msg)
        finally:
            core.set_g2_cofactor_mode(True)
            capi.set_g2_cofactor_mode(True)


def test_sign_parity(core, capi, keys16):
    sks, _, n = keys16
    msg = pr.construct_commit_payload(9, pr.synth_msg(1), 4)
    sigs = core.batch_sign(sks, msg * n, len(msg), n)
    for i in range(n):
        assert sigs[96 * i:96 * i + 96] == capi.sign_hash(sks[32 * i:32 * i + 32], msg)


def test_g1_g2_add_parity(core, capi, keys16):
    _, pks, n = keys16
    a, b = pks[:48], pks[48:96]
    assert core.g1_add(a, b) == capi.g1_add(a, b)
    assert core.g1_sub(a, b) == capi.g1_add(a, b, sub=True)
    msg = pr.synth_msg(2)
    s1 = capi.sign_hash(sk_bytes(0), msg)
    s2 = capi.sign_hash(sk_bytes(1), msg)
    assert core.g2_add(s1, s2) == capi.g2_add(s1, s2)
    # identity (zero-value struct) handling
    assert core.g1_add(b"\x00" * 48, a) == a
    assert core.g2_add(b"\x00" * 96, s1) == s1


def test_deserialize_rejects_gpu(core):
    bad = bytearray(48)
    bad[47] = 0x7F          # x >= p
    assert not core.g1_check(bytes(bad))
    # out-of-subgroup curve point must be rejected
    x = 2
    while True:
        y = pr.fp_sqrt((x * x * x + pr.B1) % pr.P)
        if y is not None and not pr.g1_in_subgroup((x, y)):
            ser = pr.g1_serialize((x, y))
            break
        x += 1
    assert not core.g1_check(ser)
    # valid genesis key accepted
    import json
    here = os.path.dirname(os.path.abspath(__file__))
    g = json.load(open(os.path.join(here, "golden", "genesis_pubkeys.json")))[0]
    assert core.g1_check(bytes.fromhex(g))


# ---------------------------------------------------------------- mask + aggregate verify
@pytest.mark.parametrize("n", [16, 256])
def test_mask_aggregate_parity(core, capi, n):
    import random
    rng = random.Random(1234 + n)
    sks = b"".join(sk_bytes(i) for i in range(n))
    pks = core.batch_pk_from_sk(sks, n)
    gc = core.Committee(pks, n)
    oc = capi.Committee(pks, n)
    for trial in range(3):
        bm = bytes(rng.getrandbits(8) for _ in range((n + 7) // 8))
        assert gc.mask_aggregate(bm) == oc.mask_aggregate(bm)
    # empty and full masks
    assert gc.mask_aggregate(bytes((n + 7) // 8)) == oc.mask_aggregate(bytes((n + 7) // 8))
    full = bytes([0xFF] * (n // 8)) if n % 8 == 0 else None
    if full:
        assert gc.mask_aggregate(full) == oc.mask_aggregate(full)


def test_mask_windowed_edges(core, capi):
    """Windowed mask path (n >= 2048): ragged tail-byte masking and the
    infinity-flag branch — a key and its negation in the same 8-key group
    sum to the point at infinity, which the affine window table cannot
    represent (hbls_dev.hip k_mask_aggregate_w / winf)."""
    import random
    n = 2053                       # >= 2048 gate, ragged final byte (5 bits)
    rng = random.Random(99)
    sks = b"".join(sk_bytes(i) for i in range(n))
    pks = bytearray(core.batch_pk_from_sk(sks, n))
    # keys 8 and 9 (same byte group): make 9 the negation of 8 by flipping
    # the herumi y-parity bit (bit 7 of the last byte)
    pks[9 * 48:10 * 48] = pks[8 * 48:9 * 48]
    pks[10 * 48 - 1] ^= 0x80
    pks = bytes(pks)
    gc = core.Committee(pks, n)
    oc = capi.Committee(pks, n)
    nb = (n + 7) // 8
    # exactly the negation pair: its group's window entry is infinity
    bm = bytearray(nb)
    bm[1] = 0x03                   # bits 8 and 9
    cases = [bytes(bm), bytes(nb)]
    # full mask (ragged tail: bits 2053..2055 must stay clear)
    full = bytearray([0xFF] * nb)
    full[-1] = (1 << (n % 8)) - 1
    cases.append(bytes(full))
    # negation pair + random sparse and dense masks
    for density in (0.1, 0.9):
        r = bytearray(nb)
        for i in range(n):
            if rng.random() < density:
                r[i >> 3] |= 1 << (i & 7)
        r[1] |= 0x03
        r[-1] &= (1 << (n % 8)) - 1
        cases.append(bytes(r))
    for bm in cases:
        assert gc.mask_aggregate(bm) == oc.mask_aggregate(bm)


def test_agg_verify_parity(core, capi, keys16):
    sks, pks, n = keys16
    gc = core.Committee(pks, n)
    oc = capi.Committee(pks, n)
    msg = pr.construct_commit_payload(55, pr.synth_msg(2), 3)
    signers = [0, 3, 5, 7, 11, 12]
    bm = bytearray((n + 7) // 8)
    for i in signers:
        bm[i >> 3] |= 1 << (i & 7)
    agg = capi.aggregate_sigs([capi.sign_hash(sks[32 * i:32 * i + 32], msg) for i in signers])
    assert gc.agg_verify(bytes(bm), agg, msg) is True
    assert oc.agg_verify(bytes(bm), agg, msg) is True
    bad = bytearray(bm)
    bad[0] ^= 2
    assert gc.agg_verify(bytes(bad), agg, msg) is False
    # empty mask + zero sig accepts (herumi identity edge)
    assert gc.agg_verify(bytes((n + 7) // 8), b"\x00" * 96, msg) is True
    # empty mask + real sig rejects
    assert gc.agg_verify(bytes((n + 7) // 8), agg, msg) is False


def test_batch_agg_verify_mixed(core, capi, keys16):
    sks, pks, n = keys16
    gc = core.Committee(pks, n)
    bmlen = (n + 7) // 8
    batch = 6
    bitmaps, sigs, msgs, expect = b"", b"", b"", []
    for j in range(batch):
        msg = pr.construct_commit_payload(j, pr.synth_msg(j), j)
        signers = [i for i in range(n) if (i + j) % 3 != 0]
        bm = bytearray(bmlen)
        for i in signers:
            bm[i >> 3] |= 1 << (i & 7)
        agg = capi.aggregate_sigs([capi.sign_hash(sks[32 * i:32 * i + 32], msg)
                                   for i in (signers if j != 4 else signers[1:])])
        expect.append(0 if j == 4 else 1)
        bitmaps += bytes(bm)
        sigs += agg
        msgs += msg
    assert gc.batch_agg_verify(bitmaps, sigs, msgs, 48, batch) == expect


def test_batch_verify_votes(core, capi, keys16):
    sks, pks, n = keys16
    gc = core.Committee(pks, n)
    msg = pr.construct_commit_payload(1, pr.synth_msg(0), 1)
    idx = [3, 7, 9]
    sigs = b"".join(capi.sign_hash(sks[32 * i:32 * i + 32], msg) for i in idx)
    res = gc.batch_verify_votes(idx, sigs, msg * len(idx), len(msg))
    assert res == [1, 1, 1]
    # swap one signature -> reject that item only
    sigs2 = sigs[:96] + capi.sign_hash(sks[32 * 8:32 * 9], msg) + sigs[192:]
    assert gc.batch_verify_votes(idx, sigs2, msg * 3, len(msg)) == [1, 0, 1]


def test_g2_subgroup_method_equivalence(core, capi):
    """psi-criterion membership == full [r]Q membership, on an in-subgroup
    signature and on an out-of-subgroup curve point."""
    import ctypes
    out = (ctypes.c_int32 * 2)()
    sig = capi.sign_hash(sk_bytes(0), pr.synth_msg(0))
    core._lib.hbls_g2_subgroup_methods(sig, out)
    assert list(out) == [1, 1]
    # out-of-subgroup point on E'(Fp2): sweep x = (c, 1) until y^2 is a QR
    c = 1
    while True:
        x = (c, 1)
        y2 = pr.f2_add(pr.f2_mul(pr.f2_sqr(x), x), pr.B2)
        y = pr.f2_sqrt(y2)
        if y is not None and pr.g2_mul((x, y), pr.R) is not None:
            bad = pr.g2_serialize((x, y))
            break
        c += 1
    core._lib.hbls_g2_subgroup_methods(bad, out)
    assert list(out) == [0, 0]
    assert not core.g2_check(bad)


def test_msm_parity_small(core):
    n = 4
    pts = [pr.get_public_key(pr.synth_sk(i)) for i in range(n)]
    scalars = [pr.synth_sk(100 + i) for i in range(n)]
    exp = None
    for p, s in zip(pts, scalars):
        exp = pr.g1_add(exp, pr.g1_mul(p, s))
    points = b"".join(pr.g1_serialize(p) for p in pts)
    sc = b"".join(pr.fr_serialize(s) for s in scalars)
    assert core.msm_g1(points, sc, n) == pr.g1_serialize(exp)


def test_keccak_batch_parity(core, capi):
    batch, mlen = 32, 136 + 9   # cross a rate boundary
    msgs = b"".join((pr.synth_msg(j) * 5)[:mlen] for j in range(batch))
    got = core.batch_keccak256(msgs, mlen, batch)
    for j in range(batch):
        assert got[32 * j:32 * j + 32] == capi.keccak256(msgs[mlen * j:mlen * (j + 1)])


def test_config4_sharded_partials(core, capi):
    """config-4 shape on one GPU: committee split into two slices; slice A's
    partial sums travel serialized (the RCCL payload) and are folded into
    slice B's verify — result must equal the full-committee verify."""
    n = 32
    sks = b"".join(sk_bytes(i) for i in range(n))
    pks = core.batch_pk_from_sk(sks, n)
    half = n // 2
    cA = core.Committee(pks[:48 * half], half)
    cB = core.Committee(pks[48 * half:], half)
    full = core.Committee(pks, n)
    batch = 3
    bmlen = (n + 7) // 8
    bmsA, bmsB, bms, sigs, msgs = b"", b"", b"", b"", b""
    for j in range(batch):
        msg = pr.construct_commit_payload(j, pr.synth_msg(j), j)
        signers = [i for i in range(n) if (i * 5 + j) % 4 != 0]
        bm = bytearray(bmlen)
        for i in signers:
            bm[i >> 3] |= 1 << (i & 7)
        sk_sum = sum(pr.synth_sk(i) for i in signers) % pr.R
        sig = capi.sign_hash(pr.fr_serialize(sk_sum), msg)
        bms += bytes(bm)
        bmsA += bytes(bm[:half // 8])
        bmsB += bytes(bm[half // 8:])
        sigs += sig
        msgs += msg
    partialsA = cA.mask_partials(bmsA, batch)
    res = cB.batch_agg_verify_partials(bmsB, partialsA, 1, sigs, msgs, 48, batch)
    assert res == [1, 1, 1]
    assert full.batch_agg_verify(bms, sigs, msgs, 48, batch) == [1, 1, 1]
    # corrupt one partial -> that item must reject
    bad = bytearray(partialsA)
    bad[0:48] = capi.pk_from_sk(sk_bytes(31))
    res2 = cB.batch_agg_verify_partials(bmsB, bytes(bad), 1, sigs, msgs, 48, batch)
    assert res2[0] == 0 and res2[1:] == [1, 1]
    # MALFORMED partial (x >= p, not a point encoding) -> bad input, not a
    # local-slice-only verify result (dpok must gate the result)
    mal = bytearray(partialsA)
    mal[0:48] = pr.fp_to_le48(pr.P)      # x == p: fp decode fails
    res3 = cB.batch_agg_verify_partials(bmsB, bytes(mal), 1, sigs, msgs, 48, batch)
    assert res3[0] == core.HBLS_ERR_BADINPUT and res3[1:] == [1, 1]


def test_deserialize_fuzz_agreement(core, capi):
    """accept/reject parity on arbitrary byte strings: GPU and oracle must
    agree exactly on every malformed/mutated candidate (G1 and G2)."""
    import random
    rng = random.Random(7)
    pk = capi.pk_from_sk(sk_bytes(0))
    sig = capi.sign_hash(sk_bytes(0), pr.synth_msg(0))
    cands48 = [bytes(rng.randrange(256) for _ in range(48)) for _ in range(24)]
    cands96 = [bytes(rng.randrange(256) for _ in range(96)) for _ in range(16)]
    for _ in range(8):
        b = bytearray(pk)
        b[rng.randrange(48)] ^= 1 << rng.randrange(8)
        cands48.append(bytes(b))
        c = bytearray(sig)
        c[rng.randrange(96)] ^= 1 << rng.randrange(8)
        cands96.append(bytes(c))
    for cand in cands48:
        assert core.g1_check(cand) == capi.g1_check(cand), cand.hex()
    for cand in cands96:
        assert core.g2_check(cand) == capi.g2_check(cand), cand.hex()


def test_config4_full_size_65536(core):
    """config-4 committee size on one GPU: slice-partials additivity and a
    full aggregate-verify at n=65536 (size-independent identities; the
    oracle only checks the small-slice path elsewhere)."""
    n = 65536
    half = n // 2
    sks_sum_all = 0
    sks = []
    for i in range(n):
        s = pr.synth_sk(i)
        sks.append(pr.fr_serialize(s))
        sks_sum_all += s
    pks = core.batch_pk_from_sk(b"".join(sks), n)
    full = core.Committee(pks, n)
    cA = core.Committee(pks[:48 * half], half)
    cB = core.Committee(pks[48 * half:], half)
    # full mask: partial(A) + partial(B) == full-committee aggregate
    bm_full = bytes([0xFF] * (n // 8))
    bm_half = bytes([0xFF] * (half // 8))
    pa = cA.mask_partials(bm_half, 1)
    pb = cB.mask_partials(bm_half, 1)
    assert core.g1_add(pa, pb) == full.mask_aggregate(bm_full)
    # aggregate-verify with every key signing (sig = (sum sk)*H(m))
    msg = pr.construct_commit_payload(5, pr.synth_msg(5), 6)
    agg = core.sign_hash(pr.fr_serialize(sks_sum_all % pr.R), msg)
    assert full.agg_verify(bm_full, agg, msg) is True
    bad = bytearray(bm_full)
    bad[100] ^= 0x10
    assert full.agg_verify(bytes(bad), agg, msg) is False


# --------------------------------------------------- config-2 size (4096) identities
@pytest.fixture(scope="module")
def committee4096(core):
    n = 4096
    sks = b"".join(sk_bytes(i) for i in range(n))
    pks = core.batch_pk_from_sk(sks, n)
    return sks, pks, core.Committee(pks, n)


def test_config2_full_size(core, capi, committee4096):
    """4096-key aggregate-verify: oracle-exact masked sum + accept/reject,
    plus the additivity identity agg(A) + agg(B) == agg(A|B) for disjoint
    masks (size-independent property at the full config size)."""
    sks, pks, gc = committee4096
    n = 4096
    import random
    rng = random.Random(42)
    bm = bytearray(n // 8)
    signers = [i for i in range(n) if rng.random() < 0.9]   # Bernoulli(0.9), seed 42
    for i in signers:
        bm[i >> 3] |= 1 << (i & 7)
    oc = capi.Committee(pks, n)
    assert gc.mask_aggregate(bytes(bm)) == oc.mask_aggregate(bytes(bm))
    # disjoint-mask additivity entirely on GPU
    bm_a = bytes(b & 0x0F for b in bm)
    bm_b = bytes(b & 0xF0 for b in bm)
    s = core.g1_add(gc.mask_aggregate(bm_a), gc.mask_aggregate(bm_b))
    assert s == gc.mask_aggregate(bytes(bm))
    # one full verify, accept + reject.  The aggregate signature of the signer
    # set equals (sum sk_i)*H(m) — computed via one sign call (bilinearity),
    # so the oracle side stays fast; GPU batch_sign is spot-checked separately.
    msg = pr.construct_commit_payload(1000, pr.synth_msg(1000), 17)
    sk_sum = sum(pr.synth_sk(i) for i in signers) % pr.R
    agg = capi.sign_hash(pr.fr_serialize(sk_sum), msg)
    some = signers[:4]
    sigs = core.batch_sign(b"".join(sks[32 * i:32 * i + 32] for i in some),
                           msg * len(some), len(msg), len(some))
    for j, i in enumerate(some):
        assert sigs[96 * j:96 * j + 96] == capi.sign_hash(sks[32 * i:32 * i + 32], msg)
    assert gc.agg_verify(bytes(bm), agg, msg) is True
    assert oc.agg_verify(bytes(bm), agg, msg) is True
    bad = bytearray(bm)
    bad[5] ^= 1 << 3
    assert gc.agg_verify(bytes(bad), agg, msg) is False


def test_coop_scalar_kernel_agreement(core, capi, keys16):
    """both verify kernels (wave-cooperative and thread-per-item) must return
    identical verdicts on a mixed accept/reject batch."""
    sks, pks, n = keys16
    gc = core.Committee(pks, n)
    bmlen = (n + 7) // 8
    batch = 5
    bitmaps, sigs, msgs = b"", b"", b""
    for j in range(batch):
        msg = pr.construct_commit_payload(j, pr.synth_msg(j + 50), j)
        signers = [i for i in range(n) if (i + j) % 2 == 0]
        bm = bytearray(bmlen)
        for i in signers:
            bm[i >> 3] |= 1 << (i & 7)
        use = signers if j != 3 else signers[:-1]   # item 3 rejects
        sk_sum = sum(pr.synth_sk(i) for i in use) % pr.R
        sigs += capi.sign_hash(pr.fr_serialize(sk_sum), msg)
        bitmaps += bytes(bm)
        msgs += msg
    core.set_coop_threshold(10**9)
    try:
        coop = gc.batch_agg_verify(bitmaps, sigs, msgs, 48, batch)
        core.set_coop_threshold(0)
        scal = gc.batch_agg_verify(bitmaps, sigs, msgs, 48, batch)
    finally:
        core.set_coop_threshold(-1)
    assert coop == scal == [1, 1, 1, 0, 1]


def test_deserialize_fuzz_extended(core, capi):
    """larger accept/reject fuzz sweep (mutations of valid points, random
    bytes, flag-bit flips) — GPU and oracle must agree on every candidate."""
    import random
    rng = random.Random(20260915)
    pk = capi.pk_from_sk(sk_bytes(3))
    sig = capi.sign_hash(sk_bytes(3), pr.synth_msg(3))
    c48, c96 = [], []
    for _ in range(120):
        c48.append(bytes(rng.randrange(256) for _ in range(48)))
        b = bytearray(pk)
        for _ in range(rng.randrange(1, 3)):
            b[rng.randrange(48)] ^= 1 << rng.randrange(8)
        c48.append(bytes(b))
    for _ in range(60):
        c96.append(bytes(rng.randrange(256) for _ in range(96)))
        b = bytearray(sig)
        for _ in range(rng.randrange(1, 3)):
            b[rng.randrange(96)] ^= 1 << rng.randrange(8)
        c96.append(bytes(b))
    # flag-bit-only flips (top bit of last byte): parity selection path
    b = bytearray(pk); b[47] ^= 0x80; c48.append(bytes(b))
    b = bytearray(sig); b[95] ^= 0x80; c96.append(bytes(b))
    mism48 = [c.hex() for c in c48 if core.g1_check(c) != capi.g1_check(c)]
    mism96 = [c.hex() for c in c96 if core.g2_check(c) != capi.g2_check(c)]
    assert mism48 == [] and mism96 == []


def test_msm_pippenger_4096(core, capi):
    """Pippenger bucket MSM parity at n=4096 random scalars.
    Points are pk_i = sk_i*G, so sum_i s_i*P_i = (sum_i s_i*sk_i mod r)*G —
    an exact python-int expectation with one C scalar mult.  Includes
    duplicate points with equal digits (doubling branch), a zero scalar,
    the identity point, and near-r scalars."""
    import random
    rng = random.Random(1234)
    n = 4096
    sks = b"".join(sk_bytes(i) for i in range(n))
    pks = core.batch_pk_from_sk(sks, n)
    pts = [pks[48 * i:48 * (i + 1)] for i in range(n)]
    scalars = [rng.randrange(pr.R) for _ in range(n)]
    scalars[0] = 0                      # zero scalar contributes nothing
    scalars[1] = pr.R - 1               # top digits exercised
    pts[2] = pts[3]                     # duplicate point
    scalars[2] = scalars[3]             # ... with identical digits (dbl path)
    pts[4] = b"\x00" * 48               # identity point, nonzero scalar
    sk_ints = [pr.synth_sk(i) for i in range(n)]
    acc = 0
    for i in range(n):
        if pts[i] != b"\x00" * 48:
            j = i
            if i == 2:
                j = 3               # pts[2] was replaced by pts[3]
            acc = (acc + scalars[i] * sk_ints[j]) % pr.R
    expected = capi.pk_from_sk(pr.fr_serialize(acc))
    got = core.msm_g1(b"".join(pts), b"".join(pr.fr_serialize(s) for s in scalars), n)
    assert got == expected
    # malformed inputs reject
    import pytest as _pytest
    bad_pts = bytearray(b"".join(pts))
    bad_pts[0:48] = pr.fp_to_le48(pr.P)
    with _pytest.raises(ValueError):
        core.msm_g1(bytes(bad_pts), b"".join(pr.fr_serialize(s) for s in scalars), n)
    bad_sc = bytearray(b"".join(pr.fr_serialize(s) for s in scalars))
    bad_sc[0:32] = b"\xff" * 32         # scalar >= r
    with _pytest.raises(ValueError):
        core.msm_g1(b"".join(pts), bytes(bad_sc), n)


def test_verify_rf_parity(core, capi, keys16):
    """Alternative verify kernels: the split-leg pairing (mode 4, always
    built) and — when the opt-in -DHBLS_RF build is present — the
    register-file kernels, accept/reject identical to the round-1 kernel
    incl. identity edges and wrong-signer rejects."""
    sks, pks, n = keys16
    gc = core.Committee(pks, n)
    msg = pr.construct_commit_payload(77, pr.synth_msg(7), 9)
    signers = [1, 2, 4, 8, 9, 13]
    bm = bytearray((n + 7) // 8)
    for i in signers:
        bm[i >> 3] |= 1 << (i & 7)
    agg = capi.aggregate_sigs([capi.sign_hash(sks[32 * i:32 * i + 32], msg)
                               for i in signers])
    bad = bytearray(bm)
    bad[0] ^= 1
    core._lib.hbls_set_verify_rf.argtypes = [__import__("ctypes").c_int]
    modes = (0, 4, 5) if not core._lib.hbls_has_rf() else (0, 1, 2, 3, 4, 5)
    try:
        for mode in modes:
            core._lib.hbls_set_verify_rf(mode)
            core.set_coop_threshold(0)        # force the scalar (rf) kernel
            try:
                assert gc.agg_verify(bytes(bm), agg, msg) is True, mode
                assert gc.agg_verify(bytes(bad), agg, msg) is False, mode
                # herumi identity edge through the rf path
                assert gc.agg_verify(bytes((n + 7) // 8), b"\x00" * 96, msg) is True, mode
                assert gc.agg_verify(bytes((n + 7) // 8), agg, msg) is False, mode
            finally:
                core.set_coop_threshold(-1)
    finally:
        core._lib.hbls_set_verify_rf(-1)


def test_msm_committee_parity(core, capi, committee4096):
    """hbls_msm_g1_committee: Pippenger core over the resident validated
    table — same result as the serialized-points entry and the python-int
    expectation (points are pk_i = sk_i*G)."""
    import random
    sks, pks, gc = committee4096
    n = 4096
    rng = random.Random(77)
    scalars = [rng.randrange(pr.R) for _ in range(n)]
    sc = b"".join(pr.fr_serialize(s) for s in scalars)
    got = gc.msm(sc)
    assert got == core.msm_g1(pks, sc, n)
    sk_ints = [pr.synth_sk(i) for i in range(n)]
    acc = sum(s * k for s, k in zip(scalars, sk_ints)) % pr.R
    assert got == capi.pk_from_sk(pr.fr_serialize(acc))
    with pytest.raises(ValueError):
        gc.msm(sc[:-32])
