"""harmony_amd.quorum — mirror of the reference's quorum decider host logic
(consensus/quorum/quorum.go + one-node-staked-vote.go + votepower/roster.go),
re-built around the batch GPU API.

This is host bookkeeping (maps, stake decimals); the crypto legs (vote
signature verification, mask aggregation, aggregate verification) go through
harmony_amd.core / the committee table on the GPU.
"""
from fractions import Fraction

from . import bls, core

# phases (consensus/quorum/quorum.go Phase)
PREPARE, COMMIT, VIEWCHANGE = "Prepare", "Commit", "ViewChange"


class Ballot:
    """votepower.Ballot (consensus/votepower/roster.go:27-33)"""

    def __init__(self, signer_pubkeys, block_header_hash, signature: bytes, height: int):
        self.SignerPubKeys = list(signer_pubkeys)   # list of 48B serialized keys
        self.BlockHeaderHash = block_header_hash
        self.Signature = signature
        self.Height = height


class BallotBox:
    """cIdentities ballot box (quorum.go:152-162, submitVote dedup :354-394)"""

    def __init__(self):
        self._votes = {p: {} for p in (PREPARE, COMMIT, VIEWCHANGE)}

    def submit_vote(self, phase, pub_keys, block_hash, sig_bytes: bytes, height: int):
        """quorum.go submitVote: reject double-vote by the same key."""
        for pk in pub_keys:
            if pk in self._votes[phase]:
                raise ValueError(f"duplicate vote by key {pk.hex()[:16]}")
        ballot = Ballot(pub_keys, block_hash, sig_bytes, height)
        for pk in pub_keys:
            self._votes[phase][pk] = ballot
        return ballot

    def read_all_ballots(self, phase):
        seen, out = set(), []
        for b in self._votes[phase].values():
            if id(b) not in seen:
                seen.add(id(b))
                out.append(b)
        return out

    def signers_count(self, phase) -> int:
        return len(self._votes[phase])

    def reset(self, phases):
        for p in phases:
            self._votes[p] = {}


class Decider:
    """One decider combining the uniform (one-node-one-vote.go, 2f+1) and
    stake-weighted (one-node-staked-vote.go, >2/3 stake) policies.

    members: list[PublicKeyWrapper]; stakes: optional list of int/Fraction
    (None -> uniform voting, mirroring SuperMajorityVote)."""

    def __init__(self, members, stakes=None):
        self.members = list(members)
        self.index = {w.Bytes: i for i, w in enumerate(self.members)}
        self.ballots = BallotBox()
        self.stakes = None
        if stakes is not None:
            total = sum(Fraction(s) for s in stakes)
            self.stakes = [Fraction(s) / total for s in stakes]
        self._committee_cache = None   # built lazily: host-only logic needs no GPU

    @property
    def _committee(self):
        if self._committee_cache is None:
            self._committee_cache = core.Committee(
                b"".join(w.Bytes for w in self.members), len(self.members))
        return self._committee_cache

    # -- policy (one-node-one-vote.go:42-58 / one-node-staked-vote.go:175-188)
    def two_thirds_count(self) -> int:
        # SuperMajorityVote: n*2/3 + 1 voters
        return (len(self.members) * 2) // 3 + 1

    def _mask_power(self, bitmap: bytes) -> Fraction:
        acc = Fraction(0)
        for i in range(len(self.members)):
            if bitmap[i >> 3] & (1 << (i & 7)):
                acc += self.stakes[i] if self.stakes else Fraction(1, len(self.members))
        return acc

    def is_quorum_achieved_by_mask(self, bitmap: bytes) -> bool:
        """IsQuorumAchievedByMask (one-node-staked-vote.go:175-188 /
        one-node-one-vote.go:75-86).  The reference builds the mask through
        Mask.SetMask, which errors on a length mismatch (mask.go:113-118), and
        its padding bits are always zero — enforce both here so an arbitrary
        caller bitmap cannot inflate the count via padding/extra bytes."""
        if len(bitmap) != (len(self.members) + 7) >> 3:
            raise ValueError(
                f"mismatching bitmap lengths expected {(len(self.members) + 7) >> 3} "
                f"got {len(bitmap)}")
        if self.stakes is None:
            n = sum(1 for i in range(len(self.members))
                    if bitmap[i >> 3] & (1 << (i & 7)))
            return n >= self.two_thirds_count()
        return self._mask_power(bitmap) > Fraction(2, 3)

    def is_quorum_achieved(self, phase) -> bool:
        if self.stakes is None:
            return self.ballots.signers_count(phase) >= self.two_thirds_count()
        acc = Fraction(0)
        for pk in self.ballots._votes[phase]:
            acc += self.stakes[self.index[pk]]
        return acc > Fraction(2, 3)

    # -- votes (AddNewVote, one-node-staked-vote.go:58-133)
    def add_new_vote(self, phase, pub_wrappers, sig: "bls.Sign", block_hash: bytes,
                     height: int, verify_payload: bytes = None):
        """Verifies the (possibly multi-key) vote signature on the GPU, then
        books it.  verify_payload defaults to block_hash (PREPARE); COMMIT
        votes pass the commit payload (leader.go:257-301 flow)."""
        payload = verify_payload if verify_payload is not None else block_hash
        agg_pub = bls.PublicKey()
        for w in pub_wrappers:
            agg_pub.add(w.Object)
        if not sig.verify_hash(agg_pub, payload):
            raise ValueError("vote signature verification failed")
        return self.ballots.submit_vote(
            phase, [w.Bytes for w in pub_wrappers], block_hash, sig.serialize(), height)

    def aggregate_votes(self, phase) -> "bls.Sign":
        """AggregateVotes (quorum.go:164-196): dedup ballots, sum signatures."""
        sigs = [bls.Sign.deserialize(b.Signature)
                for b in self.ballots.read_all_ballots(phase)]
        return bls.aggregate_sig(sigs)

    def participants_count(self) -> int:
        return len(self.members)

    # -- the aggregate-verify leg used by validators & the chain engine
    def verify_seal(self, bitmap: bytes, agg_sig: "bls.Sign", payload: bytes) -> bool:
        """verifySignature (internal/chain/engine.go:619-642): quorum by mask,
        then pairing check of the aggregate against the masked key sum."""
        if not self.is_quorum_achieved_by_mask(bitmap):
            return False
        return self._committee.agg_verify(bitmap, agg_sig.serialize(), payload)

    def batch_verify_seals(self, bitmaps: bytes, sigs: bytes, payloads: bytes,
                           mlen: int, batch: int):
        """sync-path batch seal verification (stagedstreamsync/sig_verify.go)"""
        return self._committee.batch_agg_verify(bitmaps, sigs, payloads, mlen, batch)
