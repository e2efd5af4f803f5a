"""ctypes binding of libhbls.so — THE PRODUCT PATH (HIP/CDNA4 kernels behind
the C-ABI in include/hbls.h).

Fails loudly if the library is missing or no AMD GPU is present: there is NO
CPU fallback here.  The CPU oracle under oracle/ is test infrastructure only.
"""
import ctypes
import os

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.environ.get("HBLS_SO", os.path.join(_DIR, "libhbls.so"))

HBLS_OK = 1
HBLS_FALSE = 0
HBLS_ERR = -1
HBLS_ERR_NOGPU = -2
HBLS_ERR_BADINPUT = -3


class HblsError(RuntimeError):
    pass


class NoGpuError(HblsError):
    pass


def _load():
    if not os.path.exists(_SO):
        raise HblsError(
            f"libhbls.so not found at {_SO}: build it with "
            "`python -m harmony_amd.build` (hipcc --offload-arch=gfx950)")
    lib = ctypes.CDLL(_SO)
    lib.hbls_version.restype = ctypes.c_char_p
    lib.hbls_last_kernel_ns.restype = ctypes.c_uint64
    lib.hbls_last_stage_ns.restype = ctypes.c_uint64
    lib.hbls_mad_peak_ops.restype = ctypes.c_double
    lib.hbls_committee_build.restype = ctypes.c_void_p
    lib.hbls_committee_build.argtypes = [ctypes.c_char_p, ctypes.c_size_t]
    lib.hbls_committee_free.argtypes = [ctypes.c_void_p]
    lib.hbls_committee_size.argtypes = [ctypes.c_void_p]
    lib.hbls_committee_size.restype = ctypes.c_size_t
    lib.hbls_mask_aggregate_g1.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p]
    lib.hbls_agg_verify.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p,
                                    ctypes.c_char_p, ctypes.c_size_t]
    lib.hbls_batch_agg_verify.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p,
                                          ctypes.c_char_p, ctypes.c_size_t, ctypes.c_size_t,
                                          ctypes.POINTER(ctypes.c_int32)]
    lib.hbls_batch_verify_votes.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint32),
                                            ctypes.c_char_p, ctypes.c_char_p,
                                            ctypes.c_size_t, ctypes.c_size_t,
                                            ctypes.POINTER(ctypes.c_int32)]
    lib.hbls_construct_commit_payload.argtypes = [ctypes.c_uint64, ctypes.c_char_p,
                                                  ctypes.c_uint64, ctypes.c_int, ctypes.c_char_p]
    lib.hbls_batch_seal_verify.argtypes = [
        ctypes.c_void_p, ctypes.c_char_p, ctypes.c_size_t,
        ctypes.c_char_p, ctypes.c_size_t, ctypes.c_size_t,
        ctypes.POINTER(ctypes.c_int32)]
    lib.hbls_mask_partials.argtypes = [ctypes.c_void_p, ctypes.c_char_p,
                                       ctypes.c_size_t, ctypes.c_char_p]
    lib.hbls_msm_g1_committee.argtypes = [ctypes.c_void_p, ctypes.c_char_p,
                                          ctypes.c_char_p]
    lib.hbls_batch_agg_verify_partials.argtypes = [
        ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t,
        ctypes.c_char_p, ctypes.c_char_p, ctypes.c_size_t, ctypes.c_size_t,
        ctypes.POINTER(ctypes.c_int32)]
    lib.hbls_stream_create.restype = ctypes.c_void_p
    lib.hbls_stream_create.argtypes = [ctypes.c_void_p, ctypes.c_int]
    lib.hbls_stream_free.argtypes = [ctypes.c_void_p]
    lib.hbls_stream_set_rounds.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint32), ctypes.c_int,
        ctypes.c_char_p, ctypes.c_size_t]
    lib.hbls_stream_process.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint32),
        ctypes.POINTER(ctypes.c_uint32), ctypes.c_char_p,
        ctypes.POINTER(ctypes.c_uint32), ctypes.c_int,
        ctypes.c_size_t, ctypes.POINTER(ctypes.c_int32)]
    lib.hbls_stream_check.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint32), ctypes.c_int,
        ctypes.POINTER(ctypes.c_int32)]
    lib.hbls_stream_check_submit.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint32), ctypes.c_int]
    lib.hbls_stream_check_poll.argtypes = [
        ctypes.c_void_p, ctypes.POINTER(ctypes.c_int32)]
    lib.hbls_stream_get.argtypes = [ctypes.c_void_p, ctypes.c_uint32,
                                    ctypes.c_char_p, ctypes.c_char_p]
    lib.hbls_fpmul_bench_waves.restype = ctypes.c_double
    lib.hbls_fpmul_bench_waves.argtypes = [ctypes.c_int, ctypes.c_int]
    return lib


_lib = _load()


def _check(rc, what):
    if rc == HBLS_ERR_NOGPU:
        raise NoGpuError(f"{what}: no AMD GPU available (product path requires gfx950)")
    if rc == HBLS_ERR_BADINPUT:
        raise ValueError(f"{what}: bad input")
    if rc < 0:
        raise HblsError(f"{what}: error {rc}")
    return rc


def version() -> str:
    return _lib.hbls_version().decode()


def device_count() -> int:
    return _lib.hbls_device_count()


def init(device=-1):
    _check(_lib.hbls_init(device), "hbls_init")


def last_kernel_ns() -> int:
    return _lib.hbls_last_kernel_ns()


def set_g2_cofactor_mode(fast: bool):
    _lib.hbls_set_g2_cofactor_mode(int(fast))


def set_coop_threshold(n: int):
    """batches <= n use the wave-cooperative (latency) kernels; larger use
    thread-per-item (throughput).  -1 restores the env/default value."""
    _lib.hbls_set_coop_threshold(n)


def pk_from_sk(sk32: bytes) -> bytes:
    out = ctypes.create_string_buffer(48)
    _check(_lib.hbls_pk_from_sk(sk32, out), "pk_from_sk")
    return out.raw


def batch_pk_from_sk(sks: bytes, batch: int) -> bytes:
    out = ctypes.create_string_buffer(48 * batch)
    _check(_lib.hbls_batch_pk_from_sk(sks, batch, out), "batch_pk_from_sk")
    return out.raw


def sign_hash(sk32: bytes, msg: bytes) -> bytes:
    out = ctypes.create_string_buffer(96)
    _check(_lib.hbls_sign_hash(sk32, msg, len(msg), out), "sign_hash")
    return out.raw


def batch_sign(sks: bytes, msgs: bytes, mlen: int, batch: int) -> bytes:
    out = ctypes.create_string_buffer(96 * batch)
    _check(_lib.hbls_batch_sign(sks, msgs, mlen, batch, out), "batch_sign")
    return out.raw


def hash_to_g2(msg: bytes) -> bytes:
    out = ctypes.create_string_buffer(96)
    _check(_lib.hbls_hash_to_g2(msg, len(msg), out), "hash_to_g2")
    return out.raw


def batch_hash_to_g2(msgs: bytes, mlen: int, batch: int) -> bytes:
    out = ctypes.create_string_buffer(96 * batch)
    _check(_lib.hbls_batch_hash_to_g2(msgs, mlen, batch, out), "batch_hash_to_g2")
    return out.raw


def verify_hash(pk48: bytes, sig96: bytes, msg: bytes) -> bool:
    rc = _lib.hbls_verify_hash(pk48, sig96, msg, len(msg))
    return _check(rc, "verify_hash") == HBLS_OK


def g1_add(a: bytes, b: bytes) -> bytes:
    out = ctypes.create_string_buffer(48)
    _check(_lib.hbls_g1_add(a, b, out), "g1_add")
    return out.raw


def g1_sub(a: bytes, b: bytes) -> bytes:
    out = ctypes.create_string_buffer(48)
    _check(_lib.hbls_g1_sub(a, b, out), "g1_sub")
    return out.raw


def g2_add(a: bytes, b: bytes) -> bytes:
    out = ctypes.create_string_buffer(96)
    _check(_lib.hbls_g2_add(a, b, out), "g2_add")
    return out.raw


def g1_check(p48: bytes) -> bool:
    return _check(_lib.hbls_g1_check(p48), "g1_check") == HBLS_OK


def g2_check(p96: bytes) -> bool:
    return _check(_lib.hbls_g2_check(p96), "g2_check") == HBLS_OK


def msm_g1(points: bytes, scalars: bytes, n: int) -> bytes:
    out = ctypes.create_string_buffer(48)
    _check(_lib.hbls_msm_g1(points, scalars, n, out), "msm_g1")
    return out.raw


def g2_aggregate(sigs_cat: bytes, n: int) -> bytes:
    out = ctypes.create_string_buffer(96)
    _check(_lib.hbls_g2_aggregate(sigs_cat, n, out), "g2_aggregate")
    return out.raw


def batch_keccak256(msgs: bytes, mlen: int, batch: int) -> bytes:
    out = ctypes.create_string_buffer(32 * batch)
    _check(_lib.hbls_batch_keccak256(msgs, mlen, batch, out), "batch_keccak")
    return out.raw


def construct_commit_payload(block_num: int, hash32: bytes, view_id: int,
                             staking: bool = True) -> bytes:
    out = ctypes.create_string_buffer(48)
    n = _lib.hbls_construct_commit_payload(block_num, hash32, view_id, int(staking), out)
    return out.raw[:n]


def parse_commit_sig_bitmap(payload: bytes):
    sig = ctypes.create_string_buffer(96)
    bm = ctypes.create_string_buffer(max(len(payload), 96) - 96 + 1)
    n = _lib.hbls_parse_commit_sig_bitmap(payload, len(payload), sig, bm, len(payload))
    if n < 0:
        raise ValueError("payload too short")
    return sig.raw, bm.raw[:n]


class Committee:
    """Device-resident, validated pubkey table (UpdateParticipants equivalent)."""

    def __init__(self, pks_cat: bytes, n: int):
        self._h = _lib.hbls_committee_build(pks_cat, n)
        if not self._h:
            if device_count() == 0:
                raise NoGpuError("committee_build: no AMD GPU")
            raise ValueError("committee_build: invalid pubkey in table")
        self.n = n

    def _bitmap_len(self) -> int:
        return (self.n + 7) >> 3

    def _check_bitmaps(self, bitmaps: bytes, batch: int, what: str):
        """Mask.SetMask errors on a length mismatch (crypto/bls/mask.go:113-118);
        the FFI memcpys ceil(n/8) bytes per item, so a short buffer would read
        out of bounds.  Validate before crossing the ABI."""
        want = self._bitmap_len() * batch
        if len(bitmaps) != want:
            raise ValueError(
                f"{what}: mismatching bitmap lengths expected {want} got {len(bitmaps)}")

    def mask_aggregate(self, bitmap: bytes) -> bytes:
        self._check_bitmaps(bitmap, 1, "mask_aggregate")
        out = ctypes.create_string_buffer(48)
        _check(_lib.hbls_mask_aggregate_g1(self._h, bitmap, out), "mask_aggregate")
        return out.raw

    def agg_verify(self, bitmap: bytes, sig96: bytes, msg: bytes) -> bool:
        self._check_bitmaps(bitmap, 1, "agg_verify")
        rc = _lib.hbls_agg_verify(self._h, bitmap, sig96, msg, len(msg))
        return _check(rc, "agg_verify") == HBLS_OK

    def batch_agg_verify(self, bitmaps: bytes, sigs: bytes, msgs: bytes,
                         mlen: int, batch: int):
        self._check_bitmaps(bitmaps, batch, "batch_agg_verify")
        res = (ctypes.c_int32 * batch)()
        _check(_lib.hbls_batch_agg_verify(self._h, bitmaps, sigs, msgs, mlen,
                                          batch, res), "batch_agg_verify")
        return list(res)

    def mask_partials(self, bitmaps: bytes, batch: int) -> bytes:
        self._check_bitmaps(bitmaps, batch, "mask_partials")
        out = ctypes.create_string_buffer(48 * batch)
        _check(_lib.hbls_mask_partials(self._h, bitmaps, batch, out), "mask_partials")
        return out.raw

    def batch_agg_verify_partials(self, bitmaps: bytes, ext48s: bytes, n_ext: int,
                                  sigs: bytes, msgs: bytes, mlen: int, batch: int):
        self._check_bitmaps(bitmaps, batch, "batch_agg_verify_partials")
        res = (ctypes.c_int32 * batch)()
        _check(_lib.hbls_batch_agg_verify_partials(
            self._h, bitmaps, ext48s, n_ext, sigs, msgs, mlen, batch, res),
            "batch_agg_verify_partials")
        return list(res)

    def batch_seal_verify(self, sig_bitmaps: bytes, blob_len: int,
                          msgs: bytes, mlen: int, batch: int):
        res = (ctypes.c_int32 * batch)()
        _check(_lib.hbls_batch_seal_verify(self._h, sig_bitmaps, blob_len,
                                           msgs, mlen, batch, res),
               "batch_seal_verify")
        return list(res)

    def msm(self, scalars_cat: bytes) -> bytes:
        """Pippenger MSM over the resident validated table: sum s_i * pk_i"""
        if len(scalars_cat) != 32 * self.n:
            raise ValueError("msm: scalars must be n*32 bytes")
        out = ctypes.create_string_buffer(48)
        _check(_lib.hbls_msm_g1_committee(self._h, scalars_cat, out), "msm_committee")
        return out.raw

    def batch_verify_votes(self, key_idx, sigs: bytes, msgs: bytes, mlen: int):
        batch = len(key_idx)
        idx = (ctypes.c_uint32 * batch)(*key_idx)
        res = (ctypes.c_int32 * batch)()
        _check(_lib.hbls_batch_verify_votes(self._h, idx, sigs, msgs, mlen,
                                            batch, res), "batch_verify_votes")
        return list(res)

    def __del__(self):
        try:
            if getattr(self, "_h", None):
                _lib.hbls_committee_free(self._h)
        except Exception:
            pass


class Stream:
    """Device-resident FBFT vote stream (config 5): per-round hash points,
    bitmaps and running G2 aggregates live in HBM; one hbls_stream_process
    call per tick.  Mirrors the leader's per-message loop
    (consensus/leader.go:221-309) — see include/hbls.h."""

    def __init__(self, committee: "Committee", max_rounds: int):
        self.committee = committee        # keep alive: C ctx borrows its table
        self.n = committee.n
        self.max_rounds = max_rounds
        self._h = _lib.hbls_stream_create(committee._h, max_rounds)
        if not self._h:
            if device_count() == 0:
                raise NoGpuError("stream_create: no AMD GPU")
            raise HblsError("stream_create failed")

    def set_rounds(self, slots, payloads_cat: bytes, payload_len: int):
        k = len(slots)
        if payload_len * k != len(payloads_cat):
            raise ValueError("payloads must be k * payload_len bytes")
        arr = (ctypes.c_uint32 * k)(*slots)
        _check(_lib.hbls_stream_set_rounds(self._h, arr, k, payloads_cat,
                                           payload_len), "stream_set_rounds")

    def process(self, key_idx, round_idx, sigs_cat: bytes):
        batch = len(key_idx)
        if len(round_idx) != batch or len(sigs_cat) != 96 * batch:
            raise ValueError("key_idx/round_idx/sigs length mismatch")
        active = sorted(set(round_idx))
        ki = (ctypes.c_uint32 * batch)(*key_idx)
        ri = (ctypes.c_uint32 * batch)(*round_idx)
        act = (ctypes.c_uint32 * len(active))(*active)
        res = (ctypes.c_int32 * batch)()
        _check(_lib.hbls_stream_process(self._h, ki, ri, sigs_cat, act,
                                        len(active), batch, res), "stream_process")
        return list(res)

    def check(self, slots):
        k = len(slots)
        arr = (ctypes.c_uint32 * k)(*slots)
        ok = (ctypes.c_int32 * k)()
        _check(_lib.hbls_stream_check(self._h, arr, k, ok), "stream_check")
        return [bool(v == 1) for v in ok]

    def check_submit(self, slots):
        """async check: snapshots the rounds and runs the pairing chain on a
        side stream (overlaps later process() ticks); one in flight."""
        k = len(slots)
        arr = (ctypes.c_uint32 * k)(*slots)
        _check(_lib.hbls_stream_check_submit(self._h, arr, k), "stream_check_submit")
        self._pending_check = list(slots)

    def check_poll(self):
        """waits for the in-flight check; returns {slot: bool} (empty if none)"""
        slots = getattr(self, "_pending_check", None)
        if not slots:
            return {}
        ok = (ctypes.c_int32 * len(slots))()
        k = _check(_lib.hbls_stream_check_poll(self._h, ok), "stream_check_poll")
        self._pending_check = None
        return {s: bool(ok[i] == 1) for i, s in enumerate(slots[:k])}

    def get(self, slot: int):
        """returns (bitmap bytes, serialized aggregate sig 96B)"""
        bm = ctypes.create_string_buffer((self.n + 7) // 8)
        agg = ctypes.create_string_buffer(96)
        _check(_lib.hbls_stream_get(self._h, slot, bm, agg), "stream_get")
        return bm.raw, agg.raw

    def __del__(self):
        try:
            if getattr(self, "_h", None):
                _lib.hbls_stream_free(self._h)
        except Exception:
            pass
