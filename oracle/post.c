/* POST protocol semantics: labeling, nonce search, proving scan,
 * verification.  CPU restatement of the engine the reference reaches through
 * cgo (activation/post.go:295, activation/post_verifier.go:159) — see
 * oracle.h header for pinning status and reference citations.
 * Part of the CPU oracle (test infrastructure only). */
#include "oracle.h"
#include <stdlib.h>
#include <string.h>

#ifdef _OPENMP
#include <omp.h>
#endif

void oracle_commitment(const uint8_t node_id[32],
                       const uint8_t commitment_atx_id[32], uint8_t out[32]) {
  uint8_t buf[64];
  memcpy(buf, node_id, 32);
  memcpy(buf + 32, commitment_atx_id, 32);
  oracle_blake3(buf, 64, out);
}

int oracle_label(const uint8_t commitment[32], uint64_t index,
                 uint32_t scrypt_n, uint8_t out[32]) {
  uint8_t pass[40];
  memcpy(pass, commitment, 32);
  for (int i = 0; i < 8; i++) pass[32 + i] = (uint8_t)(index >> (8 * i));
  return oracle_scrypt(pass, 40, NULL, 0, scrypt_n, 1, 1, out, 32);
}

/* big-endian lexicographic compare of 32-byte values */
static int cmp32(const uint8_t *a, const uint8_t *b) {
  return memcmp(a, b, 32);
}

int oracle_init_range(const uint8_t commitment[32], uint64_t start,
                      uint64_t end, uint32_t scrypt_n, uint8_t *out_labels,
                      const uint8_t difficulty[32], OracleVrfNonce *best) {
  int err = 0;
#pragma omp parallel
  {
    OracleVrfNonce local = {0, {0}, 0};
#pragma omp for schedule(dynamic, 64)
    for (int64_t i = (int64_t)start; i < (int64_t)end; i++) {
      uint8_t full[32];
      if (oracle_label(commitment, (uint64_t)i, scrypt_n, full)) {
#pragma omp atomic write
        err = -1;
        continue;
      }
      if (out_labels)
        memcpy(out_labels + (size_t)((uint64_t)i - start) * ORACLE_LABEL_SIZE,
               full, ORACLE_LABEL_SIZE);
      if (difficulty && cmp32(full, difficulty) < 0) {
        if (!local.found || cmp32(full, local.label) < 0 ||
            (cmp32(full, local.label) == 0 && (uint64_t)i < local.index)) {
          local.found = 1;
          local.index = (uint64_t)i;
          memcpy(local.label, full, 32);
        }
      }
    }
    if (local.found && best) {
#pragma omp critical
      {
        if (!best->found || cmp32(local.label, best->label) < 0 ||
            (cmp32(local.label, best->label) == 0 &&
             local.index < best->index)) {
          *best = local;
        }
      }
    }
  }
  return err;
}

void oracle_vrf_difficulty(uint64_t num_labels, uint8_t out[32]) {
  /* floor(16 * 2^256 / num_labels), big-endian: the nonce threshold with a
   * x16 margin so a full init pass finds a qualifying label w.p.
   * 1 - e^-16 (RESTATED; margin constant ORACLE_VRF_MARGIN).  Long division
   * of 16*2^256 by num_labels over 64-bit limbs. */
  if (num_labels <= 16) {
    memset(out, 0xff, 32);
    return;
  }
  unsigned __int128 rem = 16; /* leading limb of 16 * 2^256 */
  for (int limb = 0; limb < 4; limb++) {
    rem <<= 64;
    uint64_t q = (uint64_t)(rem / num_labels);
    rem = rem % num_labels;
    for (int b = 0; b < 8; b++)
      out[limb * 8 + b] = (uint8_t)(q >> (56 - 8 * b));
  }
}

uint64_t oracle_proving_difficulty(uint32_t k1, uint64_t num_labels) {
  /* floor(k1 * 2^64 / num_labels) (RESTATED) */
  if (num_labels == 0) return 0; /* malformed metadata: nothing passes */
  unsigned __int128 x = (unsigned __int128)k1 << 64;
  unsigned __int128 d = x / num_labels;
  if (d > (unsigned __int128)UINT64_MAX) return UINT64_MAX;
  return (uint64_t)d;
}

static void k2pow_msg(const uint8_t challenge[32], uint32_t nonce_group,
                      uint64_t pow, uint8_t msg[49]) {
  memcpy(msg, "k2pow", 5);
  memcpy(msg + 5, challenge, 32);
  for (int i = 0; i < 4; i++) msg[37 + i] = (uint8_t)(nonce_group >> (8 * i));
  for (int i = 0; i < 8; i++) msg[41 + i] = (uint8_t)(pow >> (8 * i));
}

int oracle_k2pow_verify(const uint8_t challenge[32], uint32_t nonce_group,
                        uint64_t pow, const uint8_t pow_difficulty[32]) {
  uint8_t msg[49], h[32];
  k2pow_msg(challenge, nonce_group, pow, msg);
  oracle_blake3(msg, 49, h);
  return memcmp(h, pow_difficulty, 32) < 0 ? 0 : -1;
}

uint64_t oracle_k2pow(const uint8_t challenge[32], uint32_t nonce_group,
                      const uint8_t pow_difficulty[32]) {
  for (uint64_t pow = 0;; pow++) {
    if (oracle_k2pow_verify(challenge, nonce_group, pow, pow_difficulty) == 0)
      return pow;
  }
}

void oracle_prove_cipher_key(const uint8_t challenge[32], uint32_t cipher,
                             uint64_t group_pow, uint8_t out[16]) {
  uint8_t msg[44], h[32];
  memcpy(msg, challenge, 32);
  for (int i = 0; i < 4; i++) msg[32 + i] = (uint8_t)(cipher >> (8 * i));
  for (int i = 0; i < 8; i++) msg[36 + i] = (uint8_t)(group_pow >> (8 * i));
  oracle_blake3(msg, 44, h);
  memcpy(out, h, 16);
}

uint32_t oracle_bits_per_index(uint64_t num_labels) {
  uint32_t bits = 0;
  uint64_t v = num_labels - 1;
  while (v) {
    bits++;
    v >>= 1;
  }
  return bits ? bits : 1;
}

uint32_t oracle_pack_indices(const uint64_t *idx, uint32_t k,
                             uint32_t bits_per_index, uint8_t out[800]) {
  /* LSB-first bitstream: bit b of the stream lands in out[b/8] bit (b%8). */
  uint64_t total_bits = (uint64_t)k * bits_per_index;
  uint32_t bytes = (uint32_t)((total_bits + 7) / 8);
  if (bytes > 800) return 0;
  memset(out, 0, bytes);
  uint64_t bitpos = 0;
  for (uint32_t i = 0; i < k; i++) {
    for (uint32_t b = 0; b < bits_per_index; b++) {
      if ((idx[i] >> b) & 1) out[bitpos >> 3] |= (uint8_t)(1u << (bitpos & 7));
      bitpos++;
    }
  }
  return bytes;
}

void oracle_unpack_indices(const uint8_t *packed, uint32_t k,
                           uint32_t bits_per_index, uint64_t *idx_out) {
  uint64_t bitpos = 0;
  for (uint32_t i = 0; i < k; i++) {
    uint64_t v = 0;
    for (uint32_t b = 0; b < bits_per_index; b++) {
      if ((packed[bitpos >> 3] >> (bitpos & 7)) & 1) v |= (uint64_t)1 << b;
      bitpos++;
    }
    idx_out[i] = v;
  }
}

void oracle_subset(uint32_t k2, uint32_t k3, const uint8_t *seed,
                   size_t seed_len, uint32_t *positions_out) {
  uint32_t *perm = malloc(k2 * sizeof(uint32_t));
  for (uint32_t i = 0; i < k2; i++) perm[i] = i;
  /* draws: blake3 XOF of the seed, consumed as LE u64 (RESTATED) */
  size_t draws_bytes = (size_t)k3 * 8;
  uint8_t *stream = malloc(draws_bytes);
  oracle_blake3_xof(seed, seed_len, stream, draws_bytes);
  for (uint32_t i = 0; i < k3 && i < k2; i++) {
    uint64_t d = 0;
    for (int b = 0; b < 8; b++)
      d |= (uint64_t)stream[i * 8 + b] << (8 * b);
    uint32_t j = i + (uint32_t)(d % (k2 - i));
    uint32_t tmp = perm[i];
    perm[i] = perm[j];
    perm[j] = tmp;
    positions_out[i] = perm[i];
  }
  free(perm);
  free(stream);
}

/* Per-(cipher,label) pass predicate: out = AES_key(label16);
 * v_j = LE64(out[8j..8j+8)), nonce = 2*cipher + j, pass iff v_j < difficulty.
 * (RESTATED — post-rs Prover8_56 semantics, full-u64 equivalent form.) */
int oracle_prove(const uint8_t *labels, uint64_t num_labels,
                 const uint8_t challenge[32], uint32_t k1, uint32_t k2,
                 uint32_t nonces, const uint8_t pow_difficulty[32],
                 OracleProof *proof) {
  if (nonces == 0 || nonces % ORACLE_NONCE_GROUP != 0) return -1;
  uint32_t n_ciphers = nonces / ORACLE_NONCES_PER_AES;
  uint32_t n_groups = nonces / ORACLE_NONCE_GROUP;
  uint64_t difficulty = oracle_proving_difficulty(k1, num_labels);

  uint64_t *group_pow = malloc(n_groups * sizeof(uint64_t));
  for (uint32_t g = 0; g < n_groups; g++)
    group_pow[g] = oracle_k2pow(challenge, g, pow_difficulty);

  uint8_t(*keys)[16] = malloc((size_t)n_ciphers * 16);
  for (uint32_t c = 0; c < n_ciphers; c++) {
    uint32_t grp = (c * ORACLE_NONCES_PER_AES) / ORACLE_NONCE_GROUP;
    oracle_prove_cipher_key(challenge, c, group_pow[grp], keys[c]);
  }

  /* per-nonce collected indices (first k2 each, stream order = ascending) */
  uint64_t *hits = calloc((size_t)nonces * k2, sizeof(uint64_t));
  uint32_t *nhits = calloc(nonces, sizeof(uint32_t));

  for (uint64_t i = 0; i < num_labels; i++) {
    const uint8_t *lbl = labels + i * ORACLE_LABEL_SIZE;
    for (uint32_t c = 0; c < n_ciphers; c++) {
      uint8_t out[16];
      oracle_aes128_enc_block(keys[c], lbl, out);
      for (int j = 0; j < ORACLE_NONCES_PER_AES; j++) {
        uint64_t v = 0;
        for (int b = 0; b < 8; b++)
          v |= (uint64_t)out[8 * j + b] << (8 * b);
        if (v < difficulty) {
          uint32_t nn = c * ORACLE_NONCES_PER_AES + (uint32_t)j;
          if (nhits[nn] < k2) hits[(size_t)nn * k2 + nhits[nn]++] = i;
        }
      }
    }
  }

  /* winner: smallest k2-th passing index; tie -> lowest nonce */
  int64_t best_nonce = -1;
  uint64_t best_kth = UINT64_MAX;
  for (uint32_t nn = 0; nn < nonces; nn++) {
    if (nhits[nn] >= k2) {
      uint64_t kth = hits[(size_t)nn * k2 + k2 - 1];
      if (kth < best_kth) {
        best_kth = kth;
        best_nonce = nn;
      }
    }
  }
  int rc = -1;
  if (best_nonce >= 0) {
    proof->nonce = (uint32_t)best_nonce;
    proof->pow = group_pow[best_nonce / ORACLE_NONCE_GROUP];
    proof->num_indices = (uint16_t)k2;
    uint32_t bpi = oracle_bits_per_index(num_labels);
    proof->indices_len = oracle_pack_indices(
        &hits[(size_t)best_nonce * k2], k2, bpi, proof->indices);
    rc = proof->indices_len ? 0 : -1;
  }
  free(group_pow);
  free(keys);
  free(hits);
  free(nhits);
  return rc;
}

int oracle_verify(const OracleProof *proof, const OracleProofMetadata *meta,
                  uint32_t scrypt_n, uint32_t k1, uint32_t k2, uint32_t k3,
                  const uint8_t *subset_seed, size_t subset_seed_len,
                  int32_t selected_index, const uint8_t pow_difficulty[32],
                  uint32_t *invalid_index) {
  uint64_t num_labels = (uint64_t)meta->num_units * meta->labels_per_unit;
  if (num_labels == 0) return ORACLE_VERIFY_ERR_MALFORMED;
  if (proof->num_indices != k2) return ORACLE_VERIFY_ERR_MALFORMED;
  uint32_t bpi = oracle_bits_per_index(num_labels);
  if (proof->indices_len != ((uint64_t)k2 * bpi + 7) / 8)
    return ORACLE_VERIFY_ERR_MALFORMED;

  uint32_t group = proof->nonce / ORACLE_NONCE_GROUP;
  if (oracle_k2pow_verify(meta->challenge, group, proof->pow, pow_difficulty))
    return ORACLE_VERIFY_ERR_POW;

  uint64_t *idx = malloc((size_t)k2 * sizeof(uint64_t));
  oracle_unpack_indices(proof->indices, k2, bpi, idx);

  uint8_t commitment[32];
  oracle_commitment(meta->node_id, meta->commitment_atx_id, commitment);

  uint32_t cipher = proof->nonce / ORACLE_NONCES_PER_AES;
  uint32_t half = proof->nonce % ORACLE_NONCES_PER_AES;
  uint8_t key[16];
  oracle_prove_cipher_key(meta->challenge, cipher, proof->pow, key);
  uint64_t difficulty = oracle_proving_difficulty(k1, num_labels);

  /* which positions to check */
  uint32_t *positions = malloc((size_t)k2 * sizeof(uint32_t));
  uint32_t n_check;
  if (selected_index >= 0) {
    positions[0] = (uint32_t)selected_index;
    n_check = 1;
  } else if (subset_seed != NULL && k3 < k2) {
    oracle_subset(k2, k3, subset_seed, subset_seed_len, positions);
    n_check = k3;
  } else {
    for (uint32_t i = 0; i < k2; i++) positions[i] = i;
    n_check = k2;
  }

  int rc = ORACLE_VERIFY_OK;
  for (uint32_t p = 0; p < n_check; p++) {
    uint32_t pos = positions[p];
    if (pos >= k2) { rc = ORACLE_VERIFY_ERR_MALFORMED; break; }
    uint64_t label_idx = idx[pos];
    if (label_idx >= num_labels) {
      rc = ORACLE_VERIFY_ERR_INVALID_INDEX;
      if (invalid_index) *invalid_index = pos;
      break;
    }
    uint8_t full[32], out[16];
    if (oracle_label(commitment, label_idx, scrypt_n, full)) {
      rc = ORACLE_VERIFY_ERR_MALFORMED;
      break;
    }
    oracle_aes128_enc_block(key, full, out); /* first 16 bytes of label */
    uint64_t v = 0;
    for (int b = 0; b < 8; b++)
      v |= (uint64_t)out[8 * half + b] << (8 * b);
    if (v >= difficulty) {
      rc = ORACLE_VERIFY_ERR_INVALID_INDEX;
      if (invalid_index) *invalid_index = pos;
      break;
    }
  }
  free(idx);
  free(positions);
  return rc;
}

int oracle_verify_vrf_nonce(const OracleProofMetadata *meta, uint64_t index,
                            uint32_t scrypt_n) {
  uint64_t num_labels = (uint64_t)meta->num_units * meta->labels_per_unit;
  uint8_t commitment[32], full[32], difficulty[32];
  oracle_commitment(meta->node_id, meta->commitment_atx_id, commitment);
  if (oracle_label(commitment, index, scrypt_n, full)) return -1;
  oracle_vrf_difficulty(num_labels, difficulty);
  return memcmp(full, difficulty, 32) < 0 ? 0 : 1;
}
