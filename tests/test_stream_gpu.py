"""config-5 streaming harness on GPU: per-message verify + incremental
aggregate + windowed pairing checks, parity with the oracle at the end."""
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


def test_stream_round(oracle_lib):
    from harmony_amd import core
    from harmony_amd.stream import StreamVerifier
    n = 64
    sks = [pr.fr_serialize(pr.synth_sk(i)) for i in range(n)]
    pks = core.batch_pk_from_sk(b"".join(sks), n)
    payload = pr.construct_commit_payload(77, pr.synth_msg(77), 5)
    sv = StreamVerifier(pks, n, payload, window=16)

    blob_len = 512
    import random
    rng = random.Random(7)
    order = list(range(n)) + [3, 5]          # two duplicate votes
    rng.shuffle(order)
    bad_at = order[10]                        # one corrupted signature

    sigs_all = core.batch_sign(b"".join(sks), payload * n, len(payload), n)
    for start in range(0, len(order), 8):
        chunk = order[start:start + 8]
        sigs = b""
        for i in chunk:
            s = sigs_all[96 * i:96 * (i + 1)]
            if i == bad_at:
                # substitute wrong-message signature (flip a hash byte —
                # the payload TAIL is LE64(viewID) and may already be zero)
                s = oracle_lib.sign_hash(sks[i], bytes([payload[0] ^ 0xFF]) + payload[1:])
            sigs += s
        blobs = b"".join((pr.synth_msg(i) * 20)[:blob_len] for i in chunk)
        sv.process_batch(chunk, sigs, blobs, blob_len)

    # the corrupted vote and the duplicates must not be in the aggregate
    assert sv.accepted == n - 1
    assert sv.rejected == 3          # 1 bad sig + 2 duplicates
    assert sv.final_check() is True
    # oracle cross-check of the final aggregate
    oc = oracle_lib.Committee(pks, n)
    assert oc.agg_verify(bytes(sv.bitmap), sv.agg_sig, payload) is True


def test_stream_digest_check(oracle_lib):
    """the sender-auth digest leg must be a real check — a vote
    whose blob does not hash to the digest the sender committed to is
    rejected before the signature check (checks.go:20-39 analog)."""
    from harmony_amd import core
    from harmony_amd.stream import StreamVerifier
    n = 16
    sks = [pr.fr_serialize(pr.synth_sk(i)) for i in range(n)]
    pks = core.batch_pk_from_sk(b"".join(sks), n)
    payload = pr.construct_commit_payload(5, pr.synth_msg(5), 1)
    sv = StreamVerifier(pks, n, payload, window=10 ** 9)
    blob_len = 256
    chunk = list(range(n))
    sigs = core.batch_sign(b"".join(sks), payload * n, len(payload), n)
    blobs = b"".join((pr.synth_msg(i) * 10)[:blob_len] for i in chunk)
    # expected digests from the oracle keccak; corrupt blob #4 after hashing
    expected = b"".join(oracle_lib.keccak256(
        blobs[blob_len * j:blob_len * (j + 1)]) for j in range(n))
    tampered = bytearray(blobs)
    tampered[blob_len * 4] ^= 0xFF
    res = sv.process_batch(chunk, sigs, bytes(tampered), blob_len,
                           expected_digests=expected)
    assert res[4] == 0 and all(r == 1 for j, r in enumerate(res) if j != 4)
    assert sv.accepted == n - 1 and sv.rejected == 1
    assert sv.final_check() is True


def test_device_stream_context(oracle_lib):
    """core.Stream (hbls_stream_*): multi-round device-resident verify +
    dedup + accumulate, cross-checked against the oracle round by round."""
    from harmony_amd import core
    n, R = 32, 3
    sks = [pr.fr_serialize(pr.synth_sk(i)) for i in range(n)]
    pks = core.batch_pk_from_sk(b"".join(sks), n)
    committee = core.Committee(pks, n)
    payloads = [pr.construct_commit_payload(r, pr.synth_msg(100 + r), r + 1)
                for r in range(R)]
    st = core.Stream(committee, R)
    st.set_rounds(list(range(R)), b"".join(payloads), len(payloads[0]))

    # votes: every key votes in round 0; half in round 1; round 2 stays empty.
    # include: one invalid sig, one duplicate in-batch, one duplicate
    # across ticks.
    votes = []   # (round, key)
    for i in range(n):
        votes.append((0, i))
    for i in range(0, n, 2):
        votes.append((1, i))
    sig_of = {}
    for r, i in set(votes):
        sig_of[(r, i)] = oracle_lib.sign_hash(sks[i], payloads[r])
    batch1 = votes[:20]
    sigs1 = b"".join(sig_of[v] for v in batch1)
    # corrupt vote #3's signature (valid point, wrong message)
    bad = oracle_lib.sign_hash(sks[batch1[3][1]], payloads[2])
    sigs1 = sigs1[:96 * 3] + bad + sigs1[96 * 4:]
    res1 = st.process([v[1] for v in batch1], [v[0] for v in batch1], sigs1)
    assert res1[3] == 0 and all(r == 1 for j, r in enumerate(res1) if j != 3)

    batch2 = votes[20:] + [batch1[5], votes[21]]     # cross-tick + in-batch dup
    sigs2 = b"".join(sig_of[v] for v in batch2)
    res2 = st.process([v[1] for v in batch2], [v[0] for v in batch2], sigs2)
    n2 = len(batch2)
    assert res2[n2 - 2] == 2                          # cross-tick duplicate
    assert sorted([res2[1], res2[n2 - 1]]) == [1, 2]  # in-batch dup: one wins
    assert all(r in (1, 2) for r in res2)

    # round checks: 0 and 1 verify; 2 is empty (identity vs empty mask = true
    # herumi edge); after re-opening round 2 with a payload it stays true
    assert st.check([0, 1, 2]) == [True, True, True]

    # vote #3 (the corrupted one) retried with the right sig -> accepted now
    k3 = batch1[3]
    res3 = st.process([k3[1]], [k3[0]], sig_of[k3])
    assert res3 == [1]
    assert st.check([0]) == [True]
    # oracle cross-check of each round's exported bitmap + aggregate
    oc = oracle_lib.Committee(pks, n)
    for r in range(2):
        bm, agg = st.get(r)
        voters = set(i for (rr, i) in votes if rr == r)
        assert bm == bytes(
            sum(1 << (i & 7) for i in voters if i >> 3 == b) for b in range(4))
        assert oc.agg_verify(bm, agg, payloads[r]) is True
    # unknown round slot raises
    with pytest.raises(ValueError):
        st.process([0], [R + 5], sig_of[(0, 0)])
    with pytest.raises(ValueError):
        st.check([R])


def test_stream_ragged_committee(oracle_lib):
    """n=33: the device bitmap is 32-bit words with a ragged tail — dedup
    across the word boundary (keys 31/32) and byte-exact export."""
    from harmony_amd import core
    n = 33
    sks = [pr.fr_serialize(pr.synth_sk(i)) for i in range(n)]
    pks = core.batch_pk_from_sk(b"".join(sks), n)
    st = core.Stream(core.Committee(pks, n), 2)
    payloads = [pr.construct_commit_payload(r, pr.synth_msg(300 + r), r)
                for r in range(2)]
    st.set_rounds([0, 1], b"".join(payloads), len(payloads[0]))
    votes = [(r, i) for r in range(2) for i in (31, 32, 0, 32)]   # dups incl. key 32
    sigs = b"".join(oracle_lib.sign_hash(sks[i], payloads[r]) for r, i in votes)
    res = st.process([v[1] for v in votes], [v[0] for v in votes], sigs)
    for r in range(2):
        sub = res[4 * r:4 * r + 4]
        assert sorted(sub) == [1, 1, 1, 2], res      # one dup of key 32 per round
    assert st.check([0, 1]) == [True, True]
    oc = oracle_lib.Committee(pks, n)
    for r in range(2):
        bm, agg = st.get(r)
        assert len(bm) == 5
        assert bm == bytes([0x01, 0x00, 0x00, 0x80, 0x01])   # keys 0, 31, 32
        assert oc.agg_verify(bm, agg, payloads[r]) is True


def test_stream_async_check(oracle_lib):
    """hbls_stream_check_submit/poll: the snapshot-based side-stream check
    returns the same verdicts as the synchronous check while later ticks
    proceed; double-submit rejects; empty poll is a no-op."""
    from harmony_amd import core
    n = 32
    sks = [pr.fr_serialize(pr.synth_sk(i)) for i in range(n)]
    pks = core.batch_pk_from_sk(b"".join(sks), n)
    st = core.Stream(core.Committee(pks, n), 2)
    payloads = [pr.construct_commit_payload(r, pr.synth_msg(500 + r), r)
                for r in range(2)]
    st.set_rounds([0, 1], b"".join(payloads), len(payloads[0]))
    votes = [(r, i) for r in range(2) for i in range(0, n, 2)]
    sigs = b"".join(oracle_lib.sign_hash(sks[i], payloads[r]) for r, i in votes)
    assert all(x == 1 for x in st.process([v[1] for v in votes],
                                          [v[0] for v in votes], sigs))
    assert st.check_poll() == {}                 # nothing pending
    st.check_submit([0, 1])
    with pytest.raises(ValueError):
        st.check_submit([0])                     # one in flight at a time
    # overlap: more ticks while the check runs on the side stream
    votes2 = [(r, i) for r in range(2) for i in range(1, n, 2)]
    sigs2 = b"".join(oracle_lib.sign_hash(sks[i], payloads[r]) for r, i in votes2)
    assert all(x == 1 for x in st.process([v[1] for v in votes2],
                                          [v[0] for v in votes2], sigs2))
    # the snapshot verdict covers the state at submit time (half the votes)
    assert st.check_poll() == {0: True, 1: True}
    # synchronous check of the final state agrees
    assert st.check([0, 1]) == [True, True]
