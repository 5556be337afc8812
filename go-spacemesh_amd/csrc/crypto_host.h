/* crypto_host.h — the engine's own host-side crypto.
 *
 * Deliberately independent of oracle/ (the oracle is test infrastructure and
 * must never be linked into the product path): this is implementation #2 of
 * blake3/AES/scrypt, cross-checked against the oracle and the committed
 * OpenSSL/FIPS golden vectors in tests/test_engine_cpu.py.
 *
 * Used for: commitment derivation, AES key derivation + expansion + T-table
 * generation, blake3-mode k2pow, the per-batch reference-label self-check
 * (ErrReferenceLabelMismatch semantics, activation/post.go:299-312), and the
 * final AES predicate of verification.
 */
#ifndef POST_CRYPTO_HOST_H
#define POST_CRYPTO_HOST_H

#include <stddef.h>
#include <stdint.h>

namespace poste {

void blake3(const uint8_t *msg, size_t len, uint8_t *out, size_t outlen);

void sha256(const uint8_t *msg, size_t len, uint8_t out[32]);

/* scrypt, r=1 p=1 only (the fixed post parameters). Returns 0 on success. */
int scrypt_r1p1(const uint8_t *pass, size_t passlen, uint32_t n,
                uint8_t out[32]);

/* commitment = blake3(node_id || commitment_atx_id) */
void commitment(const uint8_t node_id[32], const uint8_t atx_id[32],
                uint8_t out[32]);

/* full 32-byte label at `index` (host reference path, self-check only) */
int host_label(const uint8_t commitment32[32], uint64_t index, uint32_t n,
               uint8_t out[32]);

/* AES-128 */
void aes128_expand(const uint8_t key[16], uint32_t rk_be[44]); /* BE words */
void aes128_enc_block(const uint32_t rk_be[44], const uint8_t in[16],
                      uint8_t out[16]);
/* Generate the 4 encryption T-tables (1024 u32) + sbox (256 bytes). */
void aes128_tables(uint32_t te[1024], uint8_t sbox[256]);

/* protocol derivations (see post_common.h layout notes) */
void prove_cipher_key(const uint8_t challenge[32], uint32_t cipher,
                      uint64_t group_pow, uint8_t out[16]);
int k2pow_verify_blake3(const uint8_t challenge[32], uint32_t nonce_group,
                        uint64_t pow, const uint8_t pow_difficulty[32]);
/* multithreaded minimal-pow search */
uint64_t k2pow_search_blake3(const uint8_t challenge[32], uint32_t nonce_group,
                             const uint8_t pow_difficulty[32],
                             uint32_t threads);

uint64_t proving_difficulty(uint32_t k1, uint64_t num_labels);
void vrf_difficulty(uint64_t num_labels, uint8_t out[32]);
uint32_t bits_per_index(uint64_t num_labels);
uint32_t pack_indices(const uint64_t *idx, uint32_t k, uint32_t bpi,
                      uint8_t *out, uint32_t cap);
void unpack_indices(const uint8_t *packed, uint32_t k, uint32_t bpi,
                    uint64_t *idx_out);
void subset_positions(uint32_t k2, uint32_t k3, const uint8_t *seed,
                      size_t seed_len, uint32_t *positions_out);

} // namespace poste

#endif
