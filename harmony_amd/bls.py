"""harmony_amd.bls — mirror of the reference's Go `crypto/bls` surface
(crypto/bls/bls.go + mask.go) over the HIP C-ABI (harmony_amd.core).

Class and method names follow the Go API so the parity tests read like the
reference's own tests (SURVEY.md §8b drop-in surface).  All curve arithmetic
happens on the GPU through libhbls.so; this layer is host bookkeeping only.
"""
from . import core

PublicKeySizeInBytes = 48      # crypto/bls/bls.go:18
BLSSignatureSizeInBytes = 96   # crypto/bls/bls.go:19


class SecretKey:
    """bls_core.SecretKey mirror (herumi Fr, 32B little-endian)."""

    def __init__(self, data: bytes = b"\x00" * 32):
        if len(data) != 32:
            raise ValueError("SecretKey must be 32 bytes")
        self._b = bytes(data)

    @classmethod
    def from_hex(cls, h: str) -> "SecretKey":
        return cls(bytes.fromhex(h))

    def serialize(self) -> bytes:
        return self._b

    def serialize_to_hex_str(self) -> str:
        return self._b.hex()

    def get_public_key(self) -> "PublicKey":
        return PublicKey(core.pk_from_sk(self._b))

    def sign_hash(self, msg: bytes) -> "Sign":
        return Sign(core.sign_hash(self._b, msg))


class PublicKey:
    """bls_core.PublicKey mirror (G1, 48B compressed herumi LE).
    The zero value (48 zero bytes) is the group identity, matching Go's
    `&bls_core.PublicKey{}` usage (leader.go:165,279; quorum.go:168-195)."""

    def __init__(self, data: bytes = b"\x00" * 48):
        if len(data) != 48:
            raise ValueError("PublicKey must be 48 bytes")
        self._b = bytes(data)

    @classmethod
    def deserialize(cls, data: bytes) -> "PublicKey":
        if data != b"\x00" * 48 and not core.g1_check(data):
            raise ValueError("invalid public key")
        return cls(data)

    @classmethod
    def deserialize_hex_str(cls, h: str) -> "PublicKey":
        return cls.deserialize(bytes.fromhex(h))

    def serialize(self) -> bytes:
        return self._b

    def serialize_to_hex_str(self) -> str:
        return self._b.hex()

    def add(self, other: "PublicKey") -> "PublicKey":
        """in-place += (Go pointer-receiver semantics)"""
        self._b = core.g1_add(self._b, other._b)
        return self

    def sub(self, other: "PublicKey") -> "PublicKey":
        self._b = core.g1_sub(self._b, other._b)
        return self

    def is_equal(self, other: "PublicKey") -> bool:
        return self._b == other._b

    def __eq__(self, other):
        return isinstance(other, PublicKey) and self._b == other._b

    def __hash__(self):
        return hash(self._b)


class Sign:
    """bls_core.Sign mirror (G2, 96B compressed herumi LE); zero value = identity."""

    def __init__(self, data: bytes = b"\x00" * 96):
        if len(data) != 96:
            raise ValueError("Sign must be 96 bytes")
        self._b = bytes(data)

    @classmethod
    def deserialize(cls, data: bytes) -> "Sign":
        if data != b"\x00" * 96 and not core.g2_check(data):
            raise ValueError("invalid signature")
        return cls(data)

    @classmethod
    def deserialize_hex_str(cls, h: str) -> "Sign":
        return cls.deserialize(bytes.fromhex(h))

    def serialize(self) -> bytes:
        return self._b

    def serialize_to_hex_str(self) -> str:
        return self._b.hex()

    def add(self, other: "Sign") -> "Sign":
        self._b = core.g2_add(self._b, other._b)
        return self

    def verify_hash(self, pub: PublicKey, msg: bytes) -> bool:
        return core.verify_hash(pub._b, self._b, msg)


def aggregate_sig(sigs) -> Sign:
    """bls.AggregateSig (crypto/bls/mask.go:57-64): sum of signatures."""
    acc = Sign()
    for s in sigs:
        acc.add(s)
    return acc


class PublicKeyWrapper:
    """crypto/bls/bls.go:30-33: serialized + deserialized forms together."""

    def __init__(self, pub: PublicKey):
        self.Bytes = pub.serialize()
        self.Object = pub

    @classmethod
    def from_hex(cls, h: str) -> "PublicKeyWrapper":
        return cls(PublicKey.deserialize_hex_str(h))

    def hex(self) -> str:
        return self.Bytes.hex()


class Mask:
    """crypto/bls/mask.go:66-262 mirror.

    The reference maintains AggregatePublic incrementally (one cgo G1 add per
    flipped bit).  Here the aggregate is recomputed by the masked-sum kernel
    against the device-resident committee table on demand — the GROUP RESULT
    is identical (EC addition is associative/commutative), which the parity
    tests check bit-exactly against the oracle's sequential loop."""

    def __init__(self, publics):
        self.Publics = list(publics)            # list[PublicKeyWrapper]
        self.PublicsIndex = {w.Bytes: i for i, w in enumerate(self.Publics)}
        self.Bitmap = bytearray(self.length())
        self._committee_cache = None            # lazy: bit bookkeeping needs no GPU
        self._agg_cache = None

    @property
    def _committee(self):
        if self._committee_cache is None:
            self._committee_cache = core.Committee(
                b"".join(w.Bytes for w in self.Publics), len(self.Publics))
        return self._committee_cache

    def length(self) -> int:
        return (len(self.Publics) + 7) >> 3

    def mask(self) -> bytes:
        return bytes(self.Bitmap)

    def set_mask(self, mask: bytes):
        if len(mask) != self.length():
            raise ValueError(
                f"mismatching bitmap lengths expected {self.length()} got {len(mask)}")
        self.Bitmap = bytearray(mask)
        # The reference's SetMask only flips per-key bits, so its Bitmap padding
        # stays zero (mask.go:113-134); clear bits >= len(Publics) so mask()
        # round-trips and raw-bitmap comparisons match.
        tail = len(self.Publics) & 7
        if tail and self.Bitmap:
            self.Bitmap[-1] &= (1 << tail) - 1
        self._agg_cache = None

    def set_bit(self, i: int, enable: bool):
        if i >= len(self.Publics):
            raise ValueError("index out of range")
        byt, msk = i >> 3, 1 << (i & 7)
        old = bool(self.Bitmap[byt] & msk)
        if old != enable:
            self.Bitmap[byt] ^= msk
            self._agg_cache = None

    def set_key(self, pub_bytes: bytes, enable: bool):
        i = self.PublicsIndex.get(pub_bytes)
        if i is None:
            raise ValueError("key not found")
        self.set_bit(i, enable)

    def set_keys_atomic(self, wrappers, enable: bool):
        idx = []
        for w in wrappers:
            i = self.PublicsIndex.get(w.Bytes)
            if i is None:
                raise ValueError("key not found")
            idx.append(i)
        for i in idx:
            self.set_bit(i, enable)

    @property
    def AggregatePublic(self) -> PublicKey:
        if self._agg_cache is None:
            self._agg_cache = PublicKey(self._committee.mask_aggregate(bytes(self.Bitmap)))
        return self._agg_cache

    def index_enabled(self, i: int) -> bool:
        if i >= len(self.Publics):
            raise ValueError("index out of range")
        return bool(self.Bitmap[i >> 3] & (1 << (i & 7)))

    def key_enabled(self, pub_bytes: bytes) -> bool:
        i = self.PublicsIndex.get(pub_bytes)
        if i is None:
            raise ValueError("key not found")
        return self.index_enabled(i)

    def count_enabled(self) -> int:
        return sum(1 for i in range(len(self.Publics)) if self.index_enabled(i))

    def count_total(self) -> int:
        return len(self.Publics)

    def get_pub_key_from_mask(self, flag: bool):
        return [w for i, w in enumerate(self.Publics) if self.index_enabled(i) == flag]

    def get_signed_pub_keys_from_bitmap(self, bitmap: bytes):
        if len(bitmap) != self.length():
            raise ValueError("mismatching bitmap lengths")
        return [w for i, w in enumerate(self.Publics)
                if bitmap[i >> 3] & (1 << (i & 7))]

    def clear(self):
        self.Bitmap = bytearray(self.length())
        self._agg_cache = None

    # batch entry: verify an aggregate signature against this mask's committee
    def agg_verify(self, bitmap: bytes, sig: Sign, msg: bytes) -> bool:
        return self._committee.agg_verify(bitmap, sig.serialize(), msg)


def aggregate_masks(a: bytes, b: bytes) -> bytes:
    """bls.AggregateMasks (mask.go:266-276)"""
    if len(a) != len(b):
        raise ValueError("mismatching Bitmap lengths")
    return bytes(x | y for x, y in zip(a, b))


def separate_sig_and_mask(commit_sigs: bytes):
    """bls.SeparateSigAndMask (crypto/bls/bls.go:122-136)"""
    if len(commit_sigs) < BLSSignatureSizeInBytes:
        raise ValueError("no mask data found in commit sigs")
    return (commit_sigs[:BLSSignatureSizeInBytes],
            commit_sigs[BLSSignatureSizeInBytes:])
