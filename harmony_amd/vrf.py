"""harmony_amd.vrf — mirror of the reference's BLS-based VRF
(crypto/vrf/bls/bls_vrf.go, package blsvrf).

The VRF is a thin wrapper over the same SignHash/VerifyHash primitive the
consensus hot path uses:

  Evaluate(sk, alpha)  = (beta, pi)  with pi = SignHash(sk, SHA256(alpha)),
                         beta = SHA256(Serialize(pi))
                         (bls_vrf.go:62-77)
  ProofToHash(pk, alpha, pi) = SHA256(pi) after VerifyHash(pk, pi,
                         SHA256(alpha)) accepts (bls_vrf.go:79-101)

All curve arithmetic goes through the GPU C-ABI (harmony_amd.bls / core);
the SHA-256 shell is host-side, exactly as in the reference (Go crypto/sha256
around the cgo calls).
"""
import hashlib

from . import bls


class ErrInvalidVRF(ValueError):
    """bls_vrf.go:14 — the VRF does not validate."""


class PrivateKey:
    """blsvrf.PrivateKey (bls_vrf.go:22-25): wraps a bls SecretKey."""

    def __init__(self, secret_key: "bls.SecretKey"):
        self.SecretKey = secret_key

    def public(self) -> "bls.PublicKey":
        """Public (bls_vrf.go:31-34)"""
        return self.SecretKey.get_public_key()

    def evaluate(self, alpha: bytes):
        """Evaluate (bls_vrf.go:62-77): returns (beta, pi)."""
        msg_hash = hashlib.sha256(alpha).digest()
        pi = self.SecretKey.sign_hash(msg_hash)
        if pi is None:
            return b"\x00" * 32, None
        pi_ser = pi.serialize()
        beta = hashlib.sha256(pi_ser).digest()
        return beta, pi_ser


class PublicKey:
    """blsvrf.PublicKey (bls_vrf.go:17-20): wraps a bls PublicKey."""

    def __init__(self, public_key: "bls.PublicKey"):
        self.PublicKey = public_key

    def proof_to_hash(self, alpha: bytes, pi: bytes) -> bytes:
        """ProofToHash (bls_vrf.go:79-101): verify pi over SHA256(alpha),
        return SHA256(pi).  Raises ErrInvalidVRF like the reference returns
        ErrInvalidVRF."""
        if len(pi) == 0:
            raise ErrInvalidVRF("invalid VRF proof")
        try:
            sig = bls.Sign.deserialize(pi)
        except ValueError as e:
            raise ErrInvalidVRF(str(e))
        msg_hash = hashlib.sha256(alpha).digest()
        if not sig.verify_hash(self.PublicKey, msg_hash):
            raise ErrInvalidVRF("invalid VRF proof")
        return hashlib.sha256(pi).digest()


def new_vrf_signer(secret_key: "bls.SecretKey") -> PrivateKey:
    """NewVRFSigner (bls_vrf.go:51-54)"""
    return PrivateKey(secret_key)


def new_vrf_verifier(public_key: "bls.PublicKey") -> PublicKey:
    """NewVRFVerifier (bls_vrf.go:46-49)"""
    return PublicKey(public_key)
