/* oracle/oracle.h — CPU oracle for the go-spacemesh POST hot path.
 *
 * TEST INFRASTRUCTURE ONLY. This library is the parity checker: a plain-C
 * restatement of the POST algorithm (scrypt labeling, AES proving scan,
 * verification) that the reference node reaches through the external
 * spacemeshos/post v0.12.9 (Go, go.mod:48) + post-rs v0.7.13
 * (Makefile-libs.Inc:49) libraries.  Those libraries are NOT vendored under
 * /root/reference; where this file restates their internals
 * the layout is marked RESTATED below and in DESIGN.md.
 *
 * Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
 * call this library.  The product path (go-spacemesh_amd/ + libpost_hip.so)
 * must never link or route through it, and fails loudly when the HIP engine
 * is missing.
 *
 * Pinning status:
 *  - scrypt / PBKDF2-HMAC-SHA256 / salsa20/8: PINNED — RFC 7914 known-answer
 *    vectors (tests/test_kats.py) + randomized cross-check vs Python
 *    hashlib.scrypt (OpenSSL) at the exact post parameters.
 *  - SHA-256 / HMAC: PINNED vs Python hashlib/hmac.
 *  - AES-128: PINNED vs FIPS-197 appendix C.1 vector; S-box derived
 *    algebraically (GF(2^8) inverse + affine), not transcribed.
 *  - BLAKE3: pinned by the two official test vectors known to the builders
 *    (empty input, 1-byte input) + an independent second implementation in
 *    the engine (go-spacemesh_amd/csrc/blake3_impl.h) that must agree on
 *    random inputs.  Multi-chunk inputs are never used on this path.
 *  - Label derivation input layout, AES key derivation, difficulty formulas,
 *    index bit-packing, K3 subset sampling: RESTATED from the published
 *    post-rs v0.7.13 algorithm; parity with the real post-rs binaries is
 *    UNPINNED in-container (no golden vectors vendored in go-spacemesh —
 *    verified in SURVEY.md §4).  Internal oracle<->HIP bit-exactness is fully
 *    checked.
 *  - k2pow: reference uses RandomX (post-rs pow module).  RandomX is
 *    UNPINNED and not reimplemented (SURVEY.md §7 hard part 1); this oracle
 *    and the engine implement POST_POW_MODE_BLAKE3 with the same
 *    (challenge, nonce_group, difficulty) -> u64 contract.
 *
 * Reference call sites this restates (file:line under /root/reference/):
 *   activation/post.go:295           init.Initialize -> label stream
 *   activation/post.go:355-361       initializer inputs (node id, atx, opts)
 *   activation/validation.go:182-222 verify inputs/options
 *   activation/validation.go:261-286 VRF nonce verify
 *   api/grpcserver/post_client.go:69-143  proof shape {nonce,indices,pow}
 *   activation/wire/wire_v1.go:41-45 Indices <=800 bytes, bit-packed
 */
#ifndef POST_ORACLE_H
#define POST_ORACLE_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---------- primitives (exported for KAT tests) ---------- */
void oracle_sha256(const uint8_t *msg, size_t len, uint8_t out[32]);
void oracle_hmac_sha256(const uint8_t *key, size_t keylen,
                        const uint8_t *msg, size_t msglen, uint8_t out[32]);
void oracle_pbkdf2_sha256(const uint8_t *pass, size_t passlen,
                          const uint8_t *salt, size_t saltlen,
                          uint32_t iters, uint8_t *out, size_t outlen);
void oracle_salsa20_8(uint8_t block[64]);
/* Returns 0 on success, -1 on bad params/alloc failure. */
int oracle_scrypt(const uint8_t *pass, size_t passlen,
                  const uint8_t *salt, size_t saltlen,
                  uint32_t n, uint32_t r, uint32_t p,
                  uint8_t *out, size_t outlen);
void oracle_blake3(const uint8_t *msg, size_t len, uint8_t out[32]);
/* XOF: arbitrary-length output of the BLAKE3 root (input <= 1024 bytes). */
void oracle_blake3_xof(const uint8_t *msg, size_t len,
                       uint8_t *out, size_t outlen);
void oracle_aes128_enc_block(const uint8_t key[16], const uint8_t in[16],
                             uint8_t out[16]);

/* ---------- POST semantics ---------- */

#define ORACLE_LABEL_SIZE 16        /* bytes stored per label (postdata_*.bin) */
#define ORACLE_FULL_LABEL_SIZE 32   /* bytes compared for the VRF nonce */
#define ORACLE_NONCES_PER_AES 2     /* RESTATED: 2 nonces per AES cipher (144 encs @288 nonces) */
#define ORACLE_NONCE_GROUP 16       /* RESTATED: k2pow granularity in nonces */
#define ORACLE_VRF_MARGIN 16        /* RESTATED: vrf threshold = 16*2^256/num_labels */

/* commitment = blake3(node_id[32] || commitment_atx_id[32])  (RESTATED) */
void oracle_commitment(const uint8_t node_id[32],
                       const uint8_t commitment_atx_id[32],
                       uint8_t out[32]);

/* label_index -> full 32-byte label:
 *   scrypt(P = commitment[32] || LE64(index), S = "", N=scrypt_n, r=1, p=1,
 *          dkLen=32)                                           (RESTATED) */
int oracle_label(const uint8_t commitment[32], uint64_t index,
                 uint32_t scrypt_n, uint8_t out[32]);

typedef struct {
  uint64_t index;
  uint8_t label[32];
  int found; /* 0 when no label beat the difficulty */
} OracleVrfNonce;

/* Compute labels for [start,end): writes 16-byte truncated labels to
 * out_labels (16*(end-start) bytes; may be NULL to skip output) and tracks
 * the minimum full label < *difficulty (lexicographic big-endian byte
 * compare), tightening as it goes — the reference initializer's nonce
 * search (activation/post.go:295, common/types/activation.go:311).
 * `best` is in-out and may carry state across ranges; difficulty may be NULL
 * to disable the search.  Multi-threaded over OpenMP when built with it. */
int oracle_init_range(const uint8_t commitment[32],
                      uint64_t start, uint64_t end, uint32_t scrypt_n,
                      uint8_t *out_labels,
                      const uint8_t difficulty[32],
                      OracleVrfNonce *best);

/* difficulty = floor(ORACLE_VRF_MARGIN * 2^256 / num_labels), 32 BE bytes (RESTATED) */
void oracle_vrf_difficulty(uint64_t num_labels, uint8_t out[32]);

/* proving difficulty = floor(k1 * 2^64 / num_labels)  (RESTATED) */
uint64_t oracle_proving_difficulty(uint32_t k1, uint64_t num_labels);

/* k2pow, POST_POW_MODE_BLAKE3: find the smallest pow such that
 * blake3("k2pow" || challenge[32] || LE32(nonce_group) || LE64(pow))
 * < pow_difficulty[32] (big-endian lexicographic).  The RandomX mode of the
 * reference (post_types.go:84-114) is unpinned and unsupported. */
uint64_t oracle_k2pow(const uint8_t challenge[32], uint32_t nonce_group,
                      const uint8_t pow_difficulty[32]);
int oracle_k2pow_verify(const uint8_t challenge[32], uint32_t nonce_group,
                        uint64_t pow, const uint8_t pow_difficulty[32]);

/* AES key for cipher c (covers nonces 2c and 2c+1):
 *   first 16 bytes of blake3(challenge[32] || LE32(c) || LE64(pow_of_group))
 * where pow_of_group = k2pow of group (2c)/ORACLE_NONCE_GROUP.  (RESTATED) */
void oracle_prove_cipher_key(const uint8_t challenge[32], uint32_t cipher,
                             uint64_t group_pow, uint8_t out[16]);

typedef struct {
  uint32_t nonce;
  uint64_t pow;
  uint16_t num_indices;     /* == k2 on success */
  uint8_t indices[800];     /* bit-packed, wire_v1.go:43 cap */
  uint32_t indices_len;     /* bytes used */
} OracleProof;

/* Scan `labels` (num_labels 16-byte labels, the full index space starting at
 * label index 0) for the given challenge.  Tries `nonces` nonces (multiple of
 * ORACLE_NONCE_GROUP); the winning nonce is the one whose K2-th passing index
 * is smallest (ties: lowest nonce) — the streaming-order winner.  Returns 0
 * and fills proof, or -1 when no nonce reached k2 passes. */
int oracle_prove(const uint8_t *labels, uint64_t num_labels,
                 const uint8_t challenge[32],
                 uint32_t k1, uint32_t k2, uint32_t nonces,
                 const uint8_t pow_difficulty[32],
                 OracleProof *proof);

/* Verification inputs mirror shared.ProofMetadata as assembled at
 * activation/validation.go:193-199. */
typedef struct {
  uint8_t node_id[32];
  uint8_t commitment_atx_id[32];
  uint8_t challenge[32];
  uint32_t num_units;
  uint64_t labels_per_unit;
} OracleProofMetadata;

#define ORACLE_VERIFY_OK 0
#define ORACLE_VERIFY_ERR_INVALID_INDEX 1 /* -> verifying.ErrInvalidIndex */
#define ORACLE_VERIFY_ERR_POW 2
#define ORACLE_VERIFY_ERR_MALFORMED 3

/* K3 subset sampling seed semantics: validation.go:206-209 verifying.Subset.
 * k3 >= k2 (or seed == NULL) verifies all indices (validation.go:176-178).
 * selected_index >= 0 verifies exactly that index position — the malfeasance
 * path (activation/malfeasance.go:161-169, verifying.SelectedIndex).
 * On ERR_INVALID_INDEX, *invalid_index holds the failing index position. */
int oracle_verify(const OracleProof *proof, const OracleProofMetadata *meta,
                  uint32_t scrypt_n, uint32_t k1, uint32_t k2, uint32_t k3,
                  const uint8_t *subset_seed, size_t subset_seed_len,
                  int32_t selected_index,
                  const uint8_t pow_difficulty[32],
                  uint32_t *invalid_index);

/* verifying.VerifyVRFNonce (validation.go:277): recompute label at `index`
 * and check full 32 bytes < oracle_vrf_difficulty(num_units*labels_per_unit). */
int oracle_verify_vrf_nonce(const OracleProofMetadata *meta, uint64_t index,
                            uint32_t scrypt_n);

/* index bit-packing helpers (activation/wire/wire_v1.go:41-45) */
uint32_t oracle_bits_per_index(uint64_t num_labels);
/* pack k indices; returns bytes written (<= 800) or 0 on overflow */
uint32_t oracle_pack_indices(const uint64_t *idx, uint32_t k,
                             uint32_t bits_per_index, uint8_t out[800]);
void oracle_unpack_indices(const uint8_t *packed, uint32_t k,
                           uint32_t bits_per_index, uint64_t *idx_out);

/* K3 subset sampling (RESTATED): partial Fisher-Yates over positions
 * [0,k2), randomness = blake3 XOF of seed, u64 little-endian draws,
 * j = i + draw % (k2 - i). Writes k3 selected positions. */
void oracle_subset(uint32_t k2, uint32_t k3, const uint8_t *seed,
                   size_t seed_len, uint32_t *positions_out);

#ifdef __cplusplus
}
#endif
#endif /* POST_ORACLE_H */
