/* scrypt (RFC 7914) with the salsa20/8 core — the label derivation of the
 * POST initializer (reached by the reference via activation/post.go:295 ->
 * post-rs CpuInitializer; restated per SURVEY.md §7 step 1).
 * Part of the CPU oracle (test infrastructure only — see oracle.h header).
 *
 * Pinned by RFC 7914 §12 known-answer vectors and randomized cross-checks
 * against Python hashlib.scrypt in tests/test_kats.py. */
#include "oracle.h"
#include <stdlib.h>
#include <string.h>

static inline uint32_t rotl32(uint32_t x, unsigned n) {
  return (x << n) | (x >> (32 - n));
}

/* Salsa20/8 core over a 64-byte block of 16 little-endian u32 words. */
static void salsa8_words(uint32_t b[16]) {
  uint32_t x[16];
  memcpy(x, b, 64);
  for (int round = 0; round < 8; round += 2) {
    /* column round */
    x[4] ^= rotl32(x[0] + x[12], 7);
    x[8] ^= rotl32(x[4] + x[0], 9);
    x[12] ^= rotl32(x[8] + x[4], 13);
    x[0] ^= rotl32(x[12] + x[8], 18);
    x[9] ^= rotl32(x[5] + x[1], 7);
    x[13] ^= rotl32(x[9] + x[5], 9);
    x[1] ^= rotl32(x[13] + x[9], 13);
    x[5] ^= rotl32(x[1] + x[13], 18);
    x[14] ^= rotl32(x[10] + x[6], 7);
    x[2] ^= rotl32(x[14] + x[10], 9);
    x[6] ^= rotl32(x[2] + x[14], 13);
    x[10] ^= rotl32(x[6] + x[2], 18);
    x[3] ^= rotl32(x[15] + x[11], 7);
    x[7] ^= rotl32(x[3] + x[15], 9);
    x[11] ^= rotl32(x[7] + x[3], 13);
    x[15] ^= rotl32(x[11] + x[7], 18);
    /* row round */
    x[1] ^= rotl32(x[0] + x[3], 7);
    x[2] ^= rotl32(x[1] + x[0], 9);
    x[3] ^= rotl32(x[2] + x[1], 13);
    x[0] ^= rotl32(x[3] + x[2], 18);
    x[6] ^= rotl32(x[5] + x[4], 7);
    x[7] ^= rotl32(x[6] + x[5], 9);
    x[4] ^= rotl32(x[7] + x[6], 13);
    x[5] ^= rotl32(x[4] + x[7], 18);
    x[11] ^= rotl32(x[10] + x[9], 7);
    x[8] ^= rotl32(x[11] + x[10], 9);
    x[9] ^= rotl32(x[8] + x[11], 13);
    x[10] ^= rotl32(x[9] + x[8], 18);
    x[12] ^= rotl32(x[15] + x[14], 7);
    x[13] ^= rotl32(x[12] + x[15], 9);
    x[14] ^= rotl32(x[13] + x[12], 13);
    x[15] ^= rotl32(x[14] + x[13], 18);
  }
  for (int i = 0; i < 16; i++) b[i] += x[i];
}

void oracle_salsa20_8(uint8_t block[64]) {
  uint32_t w[16];
  for (int i = 0; i < 16; i++)
    w[i] = (uint32_t)block[4 * i] | ((uint32_t)block[4 * i + 1] << 8) |
           ((uint32_t)block[4 * i + 2] << 16) |
           ((uint32_t)block[4 * i + 3] << 24);
  salsa8_words(w);
  for (int i = 0; i < 16; i++) {
    block[4 * i] = (uint8_t)w[i];
    block[4 * i + 1] = (uint8_t)(w[i] >> 8);
    block[4 * i + 2] = (uint8_t)(w[i] >> 16);
    block[4 * i + 3] = (uint8_t)(w[i] >> 24);
  }
}

/* scryptBlockMix (RFC 7914 §4) on 2r 16-word blocks, word-level. */
static void blockmix(uint32_t *B, uint32_t *Y, uint32_t r) {
  uint32_t X[16];
  memcpy(X, &B[(2 * r - 1) * 16], 64);
  for (uint32_t i = 0; i < 2 * r; i++) {
    for (int k = 0; k < 16; k++) X[k] ^= B[i * 16 + k];
    salsa8_words(X);
    memcpy(&Y[i * 16], X, 64);
  }
  /* B' = (Y0, Y2, ..., Y_{2r-2}, Y1, Y3, ..., Y_{2r-1}) */
  for (uint32_t i = 0; i < r; i++) memcpy(&B[i * 16], &Y[2 * i * 16], 64);
  for (uint32_t i = 0; i < r; i++)
    memcpy(&B[(r + i) * 16], &Y[(2 * i + 1) * 16], 64);
}

/* scryptROMix (RFC 7914 §5). B is 32r words, V is N*32r words scratch. */
static void romix(uint32_t *B, uint32_t n, uint32_t r, uint32_t *V,
                  uint32_t *Y) {
  uint32_t words = 32 * r;
  for (uint32_t i = 0; i < n; i++) {
    memcpy(&V[(size_t)i * words], B, (size_t)words * 4);
    blockmix(B, Y, r);
  }
  for (uint32_t i = 0; i < n; i++) {
    /* Integerify: LE u64 of the first 8 bytes of the last 64-byte block */
    uint64_t j = ((uint64_t)B[words - 16] | ((uint64_t)B[words - 15] << 32)) %
                 n;
    const uint32_t *Vj = &V[(size_t)j * words];
    for (uint32_t k = 0; k < words; k++) B[k] ^= Vj[k];
    blockmix(B, Y, r);
  }
}

int oracle_scrypt(const uint8_t *pass, size_t passlen, const uint8_t *salt,
                  size_t saltlen, uint32_t n, uint32_t r, uint32_t p,
                  uint8_t *out, size_t outlen) {
  if (n < 2 || (n & (n - 1)) || r == 0 || p == 0) return -1;
  size_t bbytes = (size_t)128 * r * p;
  uint8_t *B = malloc(bbytes);
  uint32_t *Bw = malloc((size_t)32 * r * 4);
  uint32_t *V = malloc((size_t)n * 32 * r * 4);
  uint32_t *Y = malloc((size_t)32 * r * 4);
  if (!B || !Bw || !V || !Y) {
    free(B); free(Bw); free(V); free(Y);
    return -1;
  }
  oracle_pbkdf2_sha256(pass, passlen, salt, saltlen, 1, B, bbytes);
  for (uint32_t blk = 0; blk < p; blk++) {
    uint8_t *Bi = B + (size_t)blk * 128 * r;
    for (uint32_t k = 0; k < 32 * r; k++)
      Bw[k] = (uint32_t)Bi[4 * k] | ((uint32_t)Bi[4 * k + 1] << 8) |
              ((uint32_t)Bi[4 * k + 2] << 16) | ((uint32_t)Bi[4 * k + 3] << 24);
    romix(Bw, n, r, V, Y);
    for (uint32_t k = 0; k < 32 * r; k++) {
      Bi[4 * k] = (uint8_t)Bw[k];
      Bi[4 * k + 1] = (uint8_t)(Bw[k] >> 8);
      Bi[4 * k + 2] = (uint8_t)(Bw[k] >> 16);
      Bi[4 * k + 3] = (uint8_t)(Bw[k] >> 24);
    }
  }
  oracle_pbkdf2_sha256(pass, passlen, B, bbytes, 1, out, outlen);
  free(B); free(Bw); free(V); free(Y);
  return 0;
}
