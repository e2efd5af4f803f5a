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
