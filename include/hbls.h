/* hbls.h — C-ABI of the MI355X-native BLS12-381 library (libhbls.so).
 *
 * This is the drop-in boundary for harmony-one/harmony's BLS FFI: the Go repo
 * reaches all curve/pairing arithmetic through the cgo package
 * `github.com/harmony-one/bls/ffi/go/bls` (imported as bls_core at e.g.
 * crypto/bls/bls.go:8, consensus/quorum/quorum.go:11, internal/chain/sig.go:6).
 * Each entry point below names the herumi bls C function (bls/bls.h /
 * src/bls_c_impl.hpp) the Go FFI binds and which this library replaces.
 * A maintainer wires this in with a cgo shim — see INTEGRATION.md.
 *
 * Conventions (match the herumi FFI the reference uses):
 *   - curve: BLS12-381 with BLS_SWAP_G=1 (reference Makefile:71-73):
 *     public keys in G1 (48 B compressed), signatures in G2 (96 B).
 *   - all points cross the ABI in herumi little-endian compressed form
 *     (or as opaque device-resident handles for committee tables).
 *   - secret keys: 32 B little-endian Fr, value < r.
 *   - return 0 = failure/reject, 1 = success/accept, negative = error,
 *     unless noted.  Thread-safe after hbls_init() (no mutable globals).
 *   - REQUIRES an AMD GPU (gfx950).  There is no CPU fallback: calls fail
 *     loudly (HBLS_ERR_NOGPU) when no device is present.
 */
#ifndef HBLS_H
#define HBLS_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

enum {
    HBLS_OK = 1,
    HBLS_FALSE = 0,
    HBLS_ERR = -1,
    HBLS_ERR_NOGPU = -2,
    HBLS_ERR_BADINPUT = -3,
};

/* blsInit(BLS12_381) (bls.h) — called once per process (crypto/bls/mask.go:18-20).
 * Initializes the HIP device context; device = HIP device index (normally
 * LOCAL_RANK; pass -1 for current device). */
int hbls_init(int device);

/* number of usable GPUs (0 -> every call returns HBLS_ERR_NOGPU) */
int hbls_device_count(void);

/* ---- scalar drop-ins (one call per object, like the cgo surface) ---- */

/* blsGetPublicKey: pk = sk * basePoint (SecretKey.GetPublicKey) */
int hbls_pk_from_sk(const uint8_t sk32[32], uint8_t pk48[48]);

/* blsSignHash: sig = sk * H2(msg)  (SecretKey.SignHash; construct.go:101,110) */
int hbls_sign_hash(const uint8_t sk32[32], const uint8_t *msg, size_t msg_len,
                   uint8_t sig96[96]);

/* blsVerifyHash: e(pub, H2(msg)) == e(base, sig)  (Sign.VerifyHash;
 * leader.go:173,287, validator.go:228, engine.go:638).
 * Returns HBLS_OK accept / HBLS_FALSE reject / HBLS_ERR_BADINPUT. */
int hbls_verify_hash(const uint8_t pk48[48], const uint8_t sig96[96],
                     const uint8_t *msg, size_t msg_len);

/* blsPublicKeyAdd / blsPublicKeySub (PublicKey.Add/Sub; mask.go:126-130).
 * Zero input (48 zero bytes) is the identity, matching zero-value structs. */
int hbls_g1_add(const uint8_t a48[48], const uint8_t b48[48], uint8_t out48[48]);
int hbls_g1_sub(const uint8_t a48[48], const uint8_t b48[48], uint8_t out48[48]);

/* blsSignatureAdd (Sign.Add; mask.go:57-64, construct.go:99-105) */
int hbls_g2_add(const uint8_t a96[96], const uint8_t b96[96], uint8_t out96[96]);

/* blsPublicKeyDeserialize / blsSignatureDeserialize validity check
 * (subgroup-checked, as herumi does on deserialize) */
int hbls_g1_check(const uint8_t p48[48]);
int hbls_g2_check(const uint8_t p96[96]);

/* blsHashToSignature equivalent: H2(msg) serialized (used by tests) */
int hbls_hash_to_g2(const uint8_t *msg, size_t msg_len, uint8_t out96[96]);

/* mcl G2 cofactor mode: 1 = Budroni-Pintore fast (default), 0 = full h2.
 * Mirrors mcl's useOriginalG2cofactor_ switch; see DESIGN.md parity notes. */
void hbls_set_g2_cofactor_mode(int fast);

/* ---- committee table (device-resident pubkey table) ----
 * Mirrors Decider.UpdateParticipants (consensus/quorum/quorum.go:326-334) +
 * the deserialized-pubkey cache (crypto/bls/bls.go:30-33, mask.go:13-15):
 * upload once per epoch, aggregate/verify against it many times. */
typedef struct hbls_committee hbls_committee_t;

/* build from n concatenated 48-B compressed keys; validates every key
 * (curve + subgroup) on the GPU; returns NULL on any invalid key */
hbls_committee_t *hbls_committee_build(const uint8_t *pks48, size_t n);
void hbls_committee_free(hbls_committee_t *c);
size_t hbls_committee_size(const hbls_committee_t *c);

/* Mask.SetMask / masked aggregate (crypto/bls/mask.go:113-134):
 * AggregatePublic = sum of pk_i with bit i set.  bitmap is ceil(n/8) bytes,
 * little-endian bit order (bit i of byte i/8 = cosigner i). */
int hbls_mask_aggregate_g1(const hbls_committee_t *c, const uint8_t *bitmap,
                           uint8_t out48[48]);

/* one aggregate verify: DecodeSigBitmap + IsQuorumAchievedByMask's cryptoleg +
 * VerifyHash (internal/chain/sig.go:37-49 -> engine.go:630-642) */
int hbls_agg_verify(const hbls_committee_t *c, const uint8_t *bitmap,
                    const uint8_t sig96[96], const uint8_t *msg, size_t msg_len);

/* ---- batch entry points (the GPU-native surface) ---- */

/* batch of independent aggregate-verifies against one committee:
 * bitmaps: batch * ceil(n/8) bytes; sigs: batch * 96; msgs: batch * msg_len.
 * results[j] = HBLS_OK / HBLS_FALSE / HBLS_ERR_BADINPUT. */
int hbls_batch_agg_verify(const hbls_committee_t *c, const uint8_t *bitmaps,
                          const uint8_t *sigs96, const uint8_t *msgs,
                          size_t msg_len, size_t batch, int32_t *results);

/* batch verify of per-signer votes (leader's onCommit loop, leader.go:221-301):
 * item j checks sig[j] by committee member key_idx[j] on msgs[j]. */
int hbls_batch_verify_votes(const hbls_committee_t *c, const uint32_t *key_idx,
                            const uint8_t *sigs96, const uint8_t *msgs,
                            size_t msg_len, size_t batch, int32_t *results);

/* batch hash-to-G2 (toG / mcl mapToG2 legacy) */
int hbls_batch_hash_to_g2(const uint8_t *msgs, size_t msg_len, size_t batch,
                          uint8_t *out96s);

/* batch sign: sigs[j] = sk[j] * H2(msgs[j]) */
int hbls_batch_sign(const uint8_t *sks32, const uint8_t *msgs, size_t msg_len,
                    size_t batch, uint8_t *sigs96);

/* batch sk->pk */
int hbls_batch_pk_from_sk(const uint8_t *sks32, size_t batch, uint8_t *pks48);

/* ---- config-4 (sharded 65536-key committee across GPUs) ----
 * Each rank holds one index-range slice of the committee as its table.
 * hbls_mask_partials computes this rank's per-item masked partial sums
 * (serialized, batch x 48 B) for the RCCL all-gather; the receiving rank
 * folds the other ranks' partials into its own slice's sums and finishes
 * the pairing check. */
int hbls_mask_partials(const hbls_committee_t *c, const uint8_t *bitmaps,
                       size_t batch, uint8_t *out48s);
int hbls_batch_agg_verify_partials(const hbls_committee_t *c, const uint8_t *bitmaps,
                                   const uint8_t *ext48s, size_t n_ext,
                                   const uint8_t *sigs96, const uint8_t *msgs,
                                   size_t msg_len, size_t batch, int32_t *results);

/* G1 MSM: out = sum_i scalar_i * P_i (scalars 32B LE each; Pippenger
 * bucket accumulation on device).  The serialized-points entry validates
 * every point (decompress + subgroup check); the committee entry runs the
 * MSM core against the resident, already-validated table — the production
 * shape (UpdateParticipants holds the table per epoch, quorum.go:326-334) */
int hbls_msm_g1(const uint8_t *points48, const uint8_t *scalars32, size_t n,
                uint8_t out48[48]);
int hbls_msm_g1_committee(const hbls_committee_t *c, const uint8_t *scalars32,
                          uint8_t out48[48]);

/* ConstructCommitPayload (consensus/signature/signature.go:12-24); returns 40/48 */
int hbls_construct_commit_payload(uint64_t block_num, const uint8_t hash32[32],
                                  uint64_t view_id, int staking, uint8_t out[48]);

/* ParseCommitSigAndBitmap (internal/chain/sig.go:22-35); returns bitmap length */
int hbls_parse_commit_sig_bitmap(const uint8_t *payload, size_t len,
                                 uint8_t sig96[96], uint8_t *bitmap, size_t bitmap_cap);

/* sync-path batch seal verification (stagedstreamsync/sig_verify.go:23-59):
 * items are raw commitSigAndBitmap blobs (96B sig || bitmap) + payloads */
int hbls_batch_seal_verify(const hbls_committee_t *c,
                           const uint8_t *sig_bitmaps, size_t blob_len,
                           const uint8_t *msgs, size_t msg_len,
                           size_t batch, int32_t *results);

/* AggregateSig batch form (crypto/bls/mask.go:57-64): out = sum of n sigs */
int hbls_g2_aggregate(const uint8_t *sigs96, size_t n, uint8_t out96[96]);

/* ---- device-resident vote stream (config 5: the FBFT leader's per-message
 * hot loop, consensus/leader.go:221-309, re-shaped for the GPU) ----
 * A stream context holds, in HBM: per-round hash-to-G2 points
 * (ConstructCommitPayload outputs), participation bitmaps
 * (commitBitmap/Mask, crypto/bls/mask.go:226-242) and running aggregate
 * signatures (AddNewVote/Sign.Add, quorum.go:354-394).  One tick uploads
 * (key_idx, round_idx, sig) per vote and runs decompress -> verify ->
 * dedup -> per-round accumulate entirely on device. */
typedef struct hbls_stream hbls_stream_t;
hbls_stream_t *hbls_stream_create(const hbls_committee_t *c, int max_rounds);
void hbls_stream_free(hbls_stream_t *s);
/* (re)open round slots with their commit payloads (hashed on device) */
int hbls_stream_set_rounds(hbls_stream_t *s, const uint32_t *slots, int k,
                           const uint8_t *payloads, size_t payload_len);
/* one tick; results[j]: 1 accepted, 2 valid duplicate (quorum.go:354-394
 * double-vote rejection), 0 invalid, <0 malformed.  active_slots lists the
 * distinct round slots present in the batch. */
int hbls_stream_process(hbls_stream_t *s, const uint32_t *key_idx,
                        const uint32_t *round_idx, const uint8_t *sigs96,
                        const uint32_t *active_slots, int n_active,
                        size_t batch, int32_t *results);
/* pairing-check rounds' resident aggregates vs their masks
 * (validator.go:224-228); ok[i] = 1/0 */
int hbls_stream_check(hbls_stream_t *s, const uint32_t *slots, int k, int32_t *ok);
/* async form: snapshot the rounds' state (ordered after prior ticks) and run
 * the latency-bound check chain on a dedicated HIP stream, overlapping
 * subsequent ticks; one check in flight at a time.  poll waits and returns
 * the submitted k (0 if none pending). */
int hbls_stream_check_submit(hbls_stream_t *s, const uint32_t *slots, int k);
int hbls_stream_check_poll(hbls_stream_t *s, int32_t *ok);
/* export a round's bitmap + serialized aggregate
 * (consensus_service.go:305-322 commitSigAndBitmap shape) */
int hbls_stream_get(hbls_stream_t *s, uint32_t slot, uint8_t *bitmap_out,
                    uint8_t agg96_out[96]);

/* batch Keccak-256 (consensus message digests, crypto/hash/hash.go:9-15) */
int hbls_batch_keccak256(const uint8_t *msgs, size_t msg_len, size_t batch,
                         uint8_t *out32s);

/* ---- introspection ---- */
const char *hbls_version(void);
/* measured device 64x64->128 integer-mad throughput (ops/s) — the VALU
 * roofline peak for this integer-bound path (no datasheet figure exists) */
double hbls_mad_peak_ops(void);
/* elapsed device-time of the last batch call on this thread's stream, in ns
 * (HIP events); 0 if unavailable.  For bench.py's roofline leg. */
uint64_t hbls_last_kernel_ns(void);

#ifdef __cplusplus
}
#endif
#endif /* HBLS_H */
