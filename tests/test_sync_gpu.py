"""sync-path batch seal verification: the catch-up loop's
per-block VerifyHeaderSignature (stagedstreamsync/sig_verify.go:32-59) batched
through one hbls_batch_seal_verify call from raw commitSigAndBitmap blobs,
plus the sender-auth digest leg (§8f-2: checks.go:20-39 — Keccak256 of the
message, VerifyHash over the 32B digest)."""
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


def test_sync_batch_seal_verify(oracle_lib):
    """A window of blocks: header j+1 carries commitSigAndBitmap for block j
    (worker.go:595-607 SetLastCommitSignature/Bitmap); sync verifies the
    whole window in one batched call."""
    from harmony_amd import core
    n = 32
    bmlen = (n + 7) // 8
    sks = [pr.fr_serialize(pr.synth_sk(i)) for i in range(n)]
    pks = core.batch_pk_from_sk(b"".join(sks), n)
    comm = core.Committee(pks, n)

    window = 12
    blobs, msgs = b"", b""
    for j in range(window):
        block_hash = oracle_lib.keccak256(b"blk" + j.to_bytes(8, "little"))
        payload = pr.construct_commit_payload(j, block_hash, j + 1)
        signers = [i for i in range(n) if (i * 3 + j) % 4 != 0]
        bm = bytearray(bmlen)
        for i in signers:
            bm[i >> 3] |= 1 << (i & 7)
        sk_sum = sum(pr.synth_sk(i) for i in signers) % pr.R
        sig = oracle_lib.sign_hash(pr.fr_serialize(sk_sum), payload)
        blobs += sig + bytes(bm)           # commitSigAndBitmap wire blob
        msgs += payload
    res = comm.batch_seal_verify(blobs, 96 + bmlen, msgs, 48, window)
    assert res == [1] * window
    # corrupt one blob's bitmap -> only that block rejects
    bad = bytearray(blobs)
    bad[(96 + bmlen) * 5 + 96] ^= 0x01
    res2 = comm.batch_seal_verify(bytes(bad), 96 + bmlen, msgs, 48, window)
    assert res2[5] == 0 and [r for i, r in enumerate(res2) if i != 5] == [1] * (window - 1)
    # wrong blob length rejected
    with pytest.raises(ValueError):
        comm.batch_seal_verify(blobs, 96 + bmlen + 1, msgs, 48, window)


def test_sender_auth_digest_leg(oracle_lib):
    """checks.go:20-39 analog: Keccak256(message blob) on GPU, then per-sender
    VerifyHash over the 32B digest."""
    from harmony_amd import core
    n = 8
    sks = [pr.fr_serialize(pr.synth_sk(i)) for i in range(n)]
    pks = core.batch_pk_from_sk(b"".join(sks), n)
    comm = core.Committee(pks, n)
    blob_len = 300
    blobs = b"".join((pr.synth_msg(j) * 12)[:blob_len] for j in range(4))
    digests = core.batch_keccak256(blobs, blob_len, 4)
    idx = [0, 2, 5, 7]
    sigs = b"".join(oracle_lib.sign_hash(sks[i], digests[32 * j:32 * (j + 1)])
                    for j, i in enumerate(idx))
    res = comm.batch_verify_votes(idx, sigs, digests, 32)
    assert res == [1, 1, 1, 1]
    # signature by the wrong sender index rejects
    res2 = comm.batch_verify_votes([1, 2, 5, 7], sigs, digests, 32)
    assert res2[0] == 0
