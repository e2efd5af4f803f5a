"""ctypes wrapper for the CPU oracle (oracle/libhbls_oracle.so).

TEST INFRASTRUCTURE + CPU BASELINE ONLY — see oracle/pyref.py header."""
import ctypes
import os

_dir = os.path.dirname(os.path.abspath(__file__))
_lib = ctypes.CDLL(os.path.join(_dir, "libhbls_oracle.so"))

_lib.oracle_pk_from_sk.restype = ctypes.c_int
_lib.oracle_sign_hash.restype = ctypes.c_int
_lib.oracle_verify_hash.restype = ctypes.c_int
_lib.oracle_hash_to_g2.restype = ctypes.c_int
_lib.oracle_g1_add_ser.restype = ctypes.c_int
_lib.oracle_g2_add_ser.restype = ctypes.c_int
_lib.oracle_g1_deserialize_check.restype = ctypes.c_int
_lib.oracle_g2_deserialize_check.restype = ctypes.c_int
_lib.oracle_mask_aggregate_pub.restype = ctypes.c_int
_lib.oracle_aggregate_sigs.restype = ctypes.c_int
_lib.oracle_agg_verify.restype = ctypes.c_int
_lib.oracle_batch_agg_verify.restype = ctypes.c_int
_lib.oracle_nthreads.restype = ctypes.c_int


def _buf(n):
    return ctypes.create_string_buffer(n)


def pk_from_sk(sk32: bytes) -> bytes:
    out = _buf(48)
    if not _lib.oracle_pk_from_sk(sk32, out):
        raise ValueError("bad sk")
    return out.raw


def sign_hash(sk32: bytes, msg: bytes) -> bytes:
    out = _buf(96)
    if not _lib.oracle_sign_hash(sk32, msg, len(msg), out):
        raise ValueError("sign failed")
    return out.raw


def hash_to_g2(msg: bytes) -> bytes:
    out = _buf(96)
    if not _lib.oracle_hash_to_g2(msg, len(msg), out):
        raise ValueError("hash failed")
    return out.raw


def verify_hash(pk48: bytes, sig96: bytes, msg: bytes) -> bool:
    r = _lib.oracle_verify_hash(pk48, sig96, msg, len(msg))
    if r < 0:
        raise ValueError("deserialize error")
    return bool(r)


def g1_add(a: bytes, b: bytes, sub=False) -> bytes:
    out = _buf(48)
    if not _lib.oracle_g1_add_ser(a, b, out, int(sub)):
        raise ValueError("bad point")
    return out.raw


def g2_add(a: bytes, b: bytes, sub=False) -> bytes:
    out = _buf(96)
    if not _lib.oracle_g2_add_ser(a, b, out, int(sub)):
        raise ValueError("bad point")
    return out.raw


def g1_check(p48: bytes) -> bool:
    return bool(_lib.oracle_g1_deserialize_check(p48))


def g2_check(p96: bytes) -> bool:
    return bool(_lib.oracle_g2_deserialize_check(p96))


def mask_aggregate(pks: list, bitmap: bytes) -> bytes:
    cat = b"".join(pks)
    out = _buf(48)
    if not _lib.oracle_mask_aggregate_pub(cat, bitmap, len(pks), out):
        raise ValueError("bad key in table")
    return out.raw


def aggregate_sigs(sigs: list) -> bytes:
    cat = b"".join(sigs)
    out = _buf(96)
    if not _lib.oracle_aggregate_sigs(cat, len(sigs), out):
        raise ValueError("bad sig")
    return out.raw


def agg_verify(pks_cat: bytes, bitmap: bytes, n: int, sig96: bytes, msg: bytes) -> bool:
    r = _lib.oracle_agg_verify(pks_cat, bitmap, n, sig96, msg, len(msg))
    if r < 0:
        raise ValueError("deserialize error")
    return bool(r)


def batch_agg_verify(pks_cat: bytes, n: int, bitmaps: bytes, sigs: bytes,
                     msgs: bytes, mlen: int, batch: int):
    res = (ctypes.c_int32 * batch)()
    _lib.oracle_batch_agg_verify(pks_cat, n, bitmaps, sigs, msgs, mlen, batch, res)
    return list(res)


_lib.oracle_committee_build.restype = ctypes.c_void_p
_lib.oracle_committee_build.argtypes = [ctypes.c_char_p, ctypes.c_size_t]
_lib.oracle_agg_verify_tab.restype = ctypes.c_int
_lib.oracle_agg_verify_tab.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t]
_lib.oracle_batch_agg_verify_tab.restype = ctypes.c_int
_lib.oracle_batch_agg_verify_tab.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t, ctypes.c_size_t, ctypes.POINTER(ctypes.c_int32)]
_lib.oracle_mask_aggregate_tab.restype = ctypes.c_int
_lib.oracle_mask_aggregate_tab.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p]
_lib.oracle_committee_free.argtypes = [ctypes.c_void_p]


class Committee:
    """Pre-decompressed pubkey table (reference: cached PublicKeyWrapper.Object)."""

    def __init__(self, pks_cat: bytes, n: int):
        self.h = _lib.oracle_committee_build(pks_cat, n)
        if not self.h:
            raise ValueError("invalid pubkey in committee")
        self.n = n

    def agg_verify(self, bitmap: bytes, sig96: bytes, msg: bytes) -> bool:
        r = _lib.oracle_agg_verify_tab(self.h, bitmap, sig96, msg, len(msg))
        if r < 0:
            raise ValueError("deserialize error")
        return bool(r)

    def batch_agg_verify(self, bitmaps: bytes, sigs: bytes, msgs: bytes,
                         mlen: int, batch: int):
        res = (ctypes.c_int32 * batch)()
        _lib.oracle_batch_agg_verify_tab(self.h, bitmaps, sigs, msgs, mlen, batch, res)
        return list(res)

    def mask_aggregate(self, bitmap: bytes) -> bytes:
        out = ctypes.create_string_buffer(48)
        _lib.oracle_mask_aggregate_tab(self.h, bitmap, out)
        return out.raw

    def __del__(self):
        try:
            _lib.oracle_committee_free(self.h)
        except Exception:
            pass


def keccak256(data: bytes) -> bytes:
    out = _buf(32)
    _lib.oracle_keccak256(data, len(data), out)
    return out.raw


def set_g2_cofactor_mode(fast: bool):
    _lib.oracle_set_g2_cofactor_mode(int(fast))


def nthreads() -> int:
    return _lib.oracle_nthreads()


_lib.oracle_op_count.restype = ctypes.c_uint64


def reset_op_count():
    _lib.oracle_reset_op_count()


def op_count() -> int:
    """Fp-multiplications executed on THIS thread since the last reset —
    the algorithmic-work meter for roofline accounting."""
    return _lib.oracle_op_count()
