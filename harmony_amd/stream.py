"""harmony_amd.stream — config-5 streaming FBFT round harness
(BASELINE.json configs[4]): a stream of Prepare/Commit vote messages,
each a protobuf-like blob -> Keccak-256 digest -> per-message signature
verify against the sender's committee key -> incremental aggregate
(mask bit + G2 sum), with a batched pairing check of the running aggregate
every `window` messages.

Mirrors the leader's per-message hot loop (consensus/leader.go:221-309:
parse -> ConstructCommitPayload -> Sign.Deserialize -> VerifyHash ->
AddNewVote -> commitBitmap.SetKeysAtomic) re-shaped around the batch GPU
API: messages are collected into micro-batches and verified in one
hbls_batch_verify_votes launch; aggregation happens on the (tiny) mask
bitmap + one aggregate check per window via hbls_agg_verify.
"""
import os
import sys

from . import core

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


class StreamVerifier:
    def __init__(self, pks_cat: bytes, n: int, payload: bytes, window: int = 100):
        """pks_cat: committee table; payload: the commit payload all votes of
        this round sign (leader.go:250); window: aggregate-check period."""
        self.committee = core.Committee(pks_cat, n)
        self.n = n
        self.payload = payload
        self.window = window
        self.bitmap = bytearray((n + 7) // 8)
        self.agg_sig = b"\x00" * 96
        self.accepted = 0
        self.rejected = 0
        self.window_checks = 0
        self._since_check = 0

    def process_batch(self, key_idx, sigs_cat: bytes, blobs_cat: bytes, blob_len: int,
                      expected_digests: bytes = None):
        """One micro-batch of votes: blobs are hashed on-GPU (sender-auth
        digest path, checks.go:20-39 analog), signatures verified per sender
        key in one launch, accepted votes folded into the aggregate.

        expected_digests: batch*32 bytes of Keccak-256 digests the senders
        committed to (the hash the message signature covers in
        consensus_service.go:115-119); a vote whose blob hashes differently
        is rejected before the signature check."""
        batch = len(key_idx)
        # keccak digests of the raw message blobs (crypto/hash/hash.go:9-15)
        digests = core.batch_keccak256(blobs_cat, blob_len, batch)
        digest_ok = [True] * batch
        if expected_digests is not None:
            if len(expected_digests) != 32 * batch:
                raise ValueError("expected_digests must be batch*32 bytes")
            digest_ok = [digests[32 * j:32 * (j + 1)] ==
                         expected_digests[32 * j:32 * (j + 1)] for j in range(batch)]
        # per-vote verify of the commit payload signature
        msgs = self.payload * batch
        res = self.committee.batch_verify_votes(key_idx, sigs_cat, msgs, len(self.payload))
        res = [r if digest_ok[j] else 0 for j, r in enumerate(res)]
        fresh = []
        for j, ok in enumerate(res):
            i = key_idx[j]
            if ok == 1 and not (self.bitmap[i >> 3] >> (i & 7)) & 1:
                self.bitmap[i >> 3] |= 1 << (i & 7)
                fresh.append(sigs_cat[96 * j:96 * (j + 1)])
                self.accepted += 1
            else:
                self.rejected += 1
        if fresh:
            # one batched sum + one add (same group result as per-vote adds)
            batch_sum = core.g2_aggregate(b"".join(fresh), len(fresh))
            self.agg_sig = core.g2_add(self.agg_sig, batch_sum)
        self._since_check += batch
        if self._since_check >= self.window:
            self._since_check = 0
            self.window_checks += 1
            ok = self.committee.agg_verify(bytes(self.bitmap), self.agg_sig, self.payload)
            if not ok:
                raise RuntimeError("streaming aggregate diverged from mask")
        return res

    def final_check(self) -> bool:
        return self.committee.agg_verify(bytes(self.bitmap), self.agg_sig, self.payload)


class MultiStreamVerifier:
    """R concurrent FBFT rounds (consecutive blocks / multiple shards) sharing
    one committee: pending votes from ALL rounds are verified in ONE
    batch_verify_votes launch, then folded into each round's aggregate.
    This is the production shape of the vote pipeline — per-item latency is
    amortized across rounds in flight, so throughput tracks the batch rate
    instead of the single-pairing latency."""

    def __init__(self, pks_cat: bytes, n: int, payloads, window: int = 100):
        self.committee = core.Committee(pks_cat, n)
        self.n = n
        self.rounds = [StreamVerifier.__new__(StreamVerifier) for _ in payloads]
        for sv, payload in zip(self.rounds, payloads):
            sv.committee = self.committee
            sv.n = n
            sv.payload = payload
            sv.window = window
            sv.bitmap = bytearray((n + 7) // 8)
            sv.agg_sig = b"\x00" * 96
            sv.accepted = sv.rejected = sv.window_checks = 0
            sv._since_check = 0
        if len(set(len(p) for p in payloads)) > 1:
            raise ValueError("payloads must share a length for the batched launch")
        self.mlen = len(payloads[0])

    def process(self, votes):
        """votes: list of (round_idx, key_idx, sig96 bytes).  One combined
        verify launch; bookkeeping per round."""
        batch = len(votes)
        key_idx = [v[1] for v in votes]
        sigs = b"".join(v[2] for v in votes)
        msgs = b"".join(self.rounds[v[0]].payload for v in votes)
        res = self.committee.batch_verify_votes(key_idx, sigs, msgs, self.mlen)
        fresh = {}
        for (r, i, sig), ok in zip(votes, res):
            sv = self.rounds[r]
            if ok == 1 and not (sv.bitmap[i >> 3] >> (i & 7)) & 1:
                sv.bitmap[i >> 3] |= 1 << (i & 7)
                fresh.setdefault(r, []).append(sig)
                sv.accepted += 1
            else:
                sv.rejected += 1
        for r, sigs_r in fresh.items():
            sv = self.rounds[r]
            s = core.g2_aggregate(b"".join(sigs_r), len(sigs_r))
            sv.agg_sig = core.g2_add(sv.agg_sig, s)
        return res

    def final_check_all(self) -> bool:
        """one batched aggregate-verify across every round in flight"""
        bitmaps = b"".join(bytes(sv.bitmap) for sv in self.rounds)
        sigs = b"".join(sv.agg_sig for sv in self.rounds)
        msgs = b"".join(sv.payload for sv in self.rounds)
        res = self.committee.batch_agg_verify(bitmaps, sigs, msgs, self.mlen,
                                              len(self.rounds))
        return all(r == 1 for r in res)
