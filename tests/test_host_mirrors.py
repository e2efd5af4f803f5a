"""CPU tests of the harmony_amd host-logic mirrors (no GPU: only the pure
bookkeeping legs — bitmaps, dedup, quorum policy, payload/codec).  These port
the semantics of the reference's own tests (crypto/bls/mask_test.go,
consensus/quorum/quorom_test.go, consensus/signature/signature_test.go)."""
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oracle import pyref as pr  # noqa: E402


def _wrappers(n):
    """PublicKeyWrapper-shaped objects with distinct serialized keys —
    real curve points are unnecessary for bit bookkeeping."""
    from harmony_amd import bls

    class W:
        def __init__(self, b):
            self.Bytes = b
            self.Object = None
    assert bls.PublicKeySizeInBytes == 48
    return [W(bytes([i]) + bytes(47)) for i in range(n)]


def test_mask_bit_bookkeeping():
    """mask_test.go semantics: SetBit/SetMask/Count/IndexEnabled (no crypto)."""
    from harmony_amd.bls import Mask
    ws = _wrappers(10)
    m = Mask(ws)
    assert m.length() == 2 and m.count_total() == 10 and m.count_enabled() == 0
    m.set_bit(3, True)
    m.set_bit(9, True)
    assert m.count_enabled() == 2
    assert m.index_enabled(3) and m.index_enabled(9) and not m.index_enabled(0)
    assert m.key_enabled(ws[3].Bytes)
    m.set_bit(3, False)
    assert m.count_enabled() == 1
    m.set_mask(b"\xff\x03")
    assert m.count_enabled() == 10
    with pytest.raises(ValueError):
        m.set_mask(b"\xff")          # wrong length
    with pytest.raises(ValueError):
        m.set_bit(10, True)          # out of range
    m.clear()
    assert m.count_enabled() == 0
    assert m.get_signed_pub_keys_from_bitmap(b"\x05\x00") == [ws[0], ws[2]]


def test_aggregate_masks_and_separate():
    from harmony_amd import bls
    assert bls.aggregate_masks(b"\x0f\x10", b"\xf0\x01") == b"\xff\x11"
    with pytest.raises(ValueError):
        bls.aggregate_masks(b"\x00", b"\x00\x00")
    sig, bm = bls.separate_sig_and_mask(bytes(range(96)) + b"\xab\xcd")
    assert sig == bytes(range(96)) and bm == b"\xab\xcd"
    with pytest.raises(ValueError):
        bls.separate_sig_and_mask(b"\x00" * 95)


def test_quorum_policy_uniform():
    """one-node-one-vote.go SuperMajorityVote: 2f+1 = n*2/3 + 1."""
    from harmony_amd.quorum import Decider
    d = Decider(_wrappers(10))
    assert d.two_thirds_count() == 7
    bm = bytearray(2)
    for i in range(6):
        bm[i >> 3] |= 1 << (i & 7)
    assert not d.is_quorum_achieved_by_mask(bytes(bm))
    bm[0] |= 1 << 6
    assert d.is_quorum_achieved_by_mask(bytes(bm))


def test_quorum_policy_staked():
    """one-node-staked-vote.go: quorum = stake share > 2/3."""
    from harmony_amd.quorum import Decider
    d = Decider(_wrappers(4), stakes=[10, 10, 10, 70])
    # key 3 alone has 70% > 2/3
    assert d.is_quorum_achieved_by_mask(b"\x08")
    # keys 0-2 have 30%
    assert not d.is_quorum_achieved_by_mask(b"\x07")


def test_ballot_box_dedup():
    """quorum.go submitVote double-sign rejection."""
    from harmony_amd.quorum import BallotBox, PREPARE
    bb = BallotBox()
    k0, k1 = b"\x00" * 48, b"\x01" + b"\x00" * 47
    bb.submit_vote(PREPARE, [k0], b"h" * 32, b"s" * 96, 1)
    with pytest.raises(ValueError):
        bb.submit_vote(PREPARE, [k0], b"h" * 32, b"s" * 96, 1)
    bb.submit_vote(PREPARE, [k1], b"h" * 32, b"t" * 96, 1)
    assert bb.signers_count(PREPARE) == 2
    assert len(bb.read_all_ballots(PREPARE)) == 2
    bb.reset([PREPARE])
    assert bb.signers_count(PREPARE) == 0


def test_commit_payload_python_mirror():
    """signature_test.go: 40B pre-staking / 48B staking payloads."""
    h = pr.synth_msg(1)
    p40 = pr.construct_commit_payload(0x1122334455667788, h, 7, staking=False)
    p48 = pr.construct_commit_payload(0x1122334455667788, h, 7, staking=True)
    assert p40 == bytes.fromhex("8877665544332211") + h
    assert p48 == p40 + (7).to_bytes(8, "little")


def test_core_codec_helpers():
    """hbls_construct_commit_payload / hbls_parse_commit_sig_bitmap are host
    byte code in libhbls.so: loadable and byte-exact without a GPU."""
    from harmony_amd import core
    from oracle import pyref as pr
    h = pr.synth_msg(4)
    assert core.construct_commit_payload(9, h, 3, staking=True) == \
        pr.construct_commit_payload(9, h, 3, staking=True)
    assert core.construct_commit_payload(9, h, 3, staking=False) == \
        pr.construct_commit_payload(9, h, 3, staking=False)
    sig = bytes(range(96))
    bm = b"\xde\xad\xbe"
    s2, b2 = core.parse_commit_sig_bitmap(sig + bm)
    assert s2 == sig and b2 == bm
    import pytest as _pytest
    with _pytest.raises(ValueError):
        core.parse_commit_sig_bitmap(b"\x00" * 95)


def test_abi_symbols_exported():
    """every entry point declared in include/hbls.h resolves in libhbls.so
    (no compute — loadability check, runs without a GPU)."""
    import ctypes
    import re
    from harmony_amd import core
    hdr = open(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "include", "hbls.h")).read()
    names = re.findall(r"^\s*(?:int|void|double|size_t|uint64_t|const char \*|hbls_committee_t \*)\s*(hbls_\w+)\s*\(",
                       hdr, re.M)
    assert len(names) >= 25
    missing = [n for n in set(names) if not hasattr(core._lib, n)]
    assert missing == []


def test_mask_set_mask_clears_padding():
    """the reference's SetMask only flips per-key bits so Bitmap
    padding stays zero (mask.go:113-134) — set_mask must clear bits >= n."""
    from harmony_amd.bls import Mask
    ws = _wrappers(10)
    m = Mask(ws)
    m.set_mask(b"\xff\xff")          # padding bits 10..15 set by the caller
    assert m.mask() == b"\xff\x03"   # cleared to the 10 real keys
    assert m.count_enabled() == 10
    m2 = Mask(_wrappers(16))         # n % 8 == 0: nothing to clear
    m2.set_mask(b"\xff\xff")
    assert m2.mask() == b"\xff\xff"


def test_quorum_bitmap_length_and_padding():
    """verify_seal path must reject mismatched bitmap lengths
    (like Mask.SetMask, mask.go:113-118) and never count padding bits."""
    from harmony_amd.quorum import Decider
    d = Decider(_wrappers(10))
    with pytest.raises(ValueError):
        d.is_quorum_achieved_by_mask(b"\xff")            # short
    with pytest.raises(ValueError):
        d.is_quorum_achieved_by_mask(b"\xff\x03\x00")    # long
    # 6 real votes + all 6 padding bits set: padding must NOT reach quorum (7)
    assert not d.is_quorum_achieved_by_mask(b"\x3f\xfc")
    # 7 real votes -> quorum
    assert d.is_quorum_achieved_by_mask(b"\x7f\x00")
    # staked policy validates length too
    ds = Decider(_wrappers(4), stakes=[10, 10, 10, 70])
    with pytest.raises(ValueError):
        ds.is_quorum_achieved_by_mask(b"\x08\x00")
