/* SHA-256 + HMAC + PBKDF2 (FIPS 180-4 / RFC 2104 / RFC 2898).
 * Part of the CPU oracle (test infrastructure only — see oracle.h header). */
#include "oracle.h"
#include <string.h>

typedef struct {
  uint32_t h[8];
  uint64_t len;
  uint8_t buf[64];
  size_t buflen;
} sha256_ctx;

static const uint32_t K256[64] = {
    0x428a2f98, 0x71374491, 0xb5c0fbcf, 0xe9b5dba5, 0x3956c25b, 0x59f111f1,
    0x923f82a4, 0xab1c5ed5, 0xd807aa98, 0x12835b01, 0x243185be, 0x550c7dc3,
    0x72be5d74, 0x80deb1fe, 0x9bdc06a7, 0xc19bf174, 0xe49b69c1, 0xefbe4786,
    0x0fc19dc6, 0x240ca1cc, 0x2de92c6f, 0x4a7484aa, 0x5cb0a9dc, 0x76f988da,
    0x983e5152, 0xa831c66d, 0xb00327c8, 0xbf597fc7, 0xc6e00bf3, 0xd5a79147,
    0x06ca6351, 0x14292967, 0x27b70a85, 0x2e1b2138, 0x4d2c6dfc, 0x53380d13,
    0x650a7354, 0x766a0abb, 0x81c2c92e, 0x92722c85, 0xa2bfe8a1, 0xa81a664b,
    0xc24b8b70, 0xc76c51a3, 0xd192e819, 0xd6990624, 0xf40e3585, 0x106aa070,
    0x19a4c116, 0x1e376c08, 0x2748774c, 0x34b0bcb5, 0x391c0cb3, 0x4ed8aa4a,
    0x5b9cca4f, 0x682e6ff3, 0x748f82ee, 0x78a5636f, 0x84c87814, 0x8cc70208,
    0x90befffa, 0xa4506ceb, 0xbef9a3f7, 0xc67178f2};

static inline uint32_t rotr32(uint32_t x, unsigned n) {
  return (x >> n) | (x << (32 - n));
}

static void sha256_compress(uint32_t h[8], const uint8_t block[64]) {
  uint32_t w[64];
  for (int i = 0; i < 16; i++)
    w[i] = ((uint32_t)block[4 * i] << 24) | ((uint32_t)block[4 * i + 1] << 16) |
           ((uint32_t)block[4 * i + 2] << 8) | block[4 * i + 3];
  for (int i = 16; i < 64; i++) {
    uint32_t s0 = rotr32(w[i - 15], 7) ^ rotr32(w[i - 15], 18) ^ (w[i - 15] >> 3);
    uint32_t s1 = rotr32(w[i - 2], 17) ^ rotr32(w[i - 2], 19) ^ (w[i - 2] >> 10);
    w[i] = w[i - 16] + s0 + w[i - 7] + s1;
  }
  uint32_t a = h[0], b = h[1], c = h[2], d = h[3];
  uint32_t e = h[4], f = h[5], g = h[6], hh = h[7];
  for (int i = 0; i < 64; i++) {
    uint32_t S1 = rotr32(e, 6) ^ rotr32(e, 11) ^ rotr32(e, 25);
    uint32_t ch = (e & f) ^ (~e & g);
    uint32_t t1 = hh + S1 + ch + K256[i] + w[i];
    uint32_t S0 = rotr32(a, 2) ^ rotr32(a, 13) ^ rotr32(a, 22);
    uint32_t maj = (a & b) ^ (a & c) ^ (b & c);
    uint32_t t2 = S0 + maj;
    hh = g; g = f; f = e; e = d + t1;
    d = c; c = b; b = a; a = t1 + t2;
  }
  h[0] += a; h[1] += b; h[2] += c; h[3] += d;
  h[4] += e; h[5] += f; h[6] += g; h[7] += hh;
}

static void sha256_init(sha256_ctx *c) {
  static const uint32_t H0[8] = {0x6a09e667, 0xbb67ae85, 0x3c6ef372,
                                 0xa54ff53a, 0x510e527f, 0x9b05688c,
                                 0x1f83d9ab, 0x5be0cd19};
  memcpy(c->h, H0, sizeof(H0));
  c->len = 0;
  c->buflen = 0;
}

static void sha256_update(sha256_ctx *c, const uint8_t *msg, size_t len) {
  c->len += len;
  if (c->buflen) {
    size_t take = 64 - c->buflen;
    if (take > len) take = len;
    memcpy(c->buf + c->buflen, msg, take);
    c->buflen += take;
    msg += take;
    len -= take;
    if (c->buflen == 64) {
      sha256_compress(c->h, c->buf);
      c->buflen = 0;
    }
  }
  while (len >= 64) {
    sha256_compress(c->h, msg);
    msg += 64;
    len -= 64;
  }
  if (len) {
    memcpy(c->buf, msg, len);
    c->buflen = len;
  }
}

static void sha256_final(sha256_ctx *c, uint8_t out[32]) {
  uint64_t bitlen = c->len * 8;
  uint8_t pad = 0x80;
  sha256_update(c, &pad, 1);
  uint8_t zero = 0;
  while (c->buflen != 56) sha256_update(c, &zero, 1);
  uint8_t lenb[8];
  for (int i = 0; i < 8; i++) lenb[i] = (uint8_t)(bitlen >> (56 - 8 * i));
  sha256_update(c, lenb, 8);
  for (int i = 0; i < 8; i++) {
    out[4 * i] = (uint8_t)(c->h[i] >> 24);
    out[4 * i + 1] = (uint8_t)(c->h[i] >> 16);
    out[4 * i + 2] = (uint8_t)(c->h[i] >> 8);
    out[4 * i + 3] = (uint8_t)c->h[i];
  }
}

void oracle_sha256(const uint8_t *msg, size_t len, uint8_t out[32]) {
  sha256_ctx c;
  sha256_init(&c);
  sha256_update(&c, msg, len);
  sha256_final(&c, out);
}

void oracle_hmac_sha256(const uint8_t *key, size_t keylen, const uint8_t *msg,
                        size_t msglen, uint8_t out[32]) {
  uint8_t k[64] = {0};
  if (keylen > 64)
    oracle_sha256(key, keylen, k); /* hashed key, rest zero */
  else
    memcpy(k, key, keylen);
  uint8_t ipad[64], opad[64];
  for (int i = 0; i < 64; i++) {
    ipad[i] = k[i] ^ 0x36;
    opad[i] = k[i] ^ 0x5c;
  }
  sha256_ctx c;
  uint8_t inner[32];
  sha256_init(&c);
  sha256_update(&c, ipad, 64);
  sha256_update(&c, msg, msglen);
  sha256_final(&c, inner);
  sha256_init(&c);
  sha256_update(&c, opad, 64);
  sha256_update(&c, inner, 32);
  sha256_final(&c, out);
}

void oracle_pbkdf2_sha256(const uint8_t *pass, size_t passlen,
                          const uint8_t *salt, size_t saltlen, uint32_t iters,
                          uint8_t *out, size_t outlen) {
  uint32_t blocks = (uint32_t)((outlen + 31) / 32);
  uint8_t saltint[4];
  uint8_t u[32], t[32];
  for (uint32_t b = 1; b <= blocks; b++) {
    saltint[0] = (uint8_t)(b >> 24);
    saltint[1] = (uint8_t)(b >> 16);
    saltint[2] = (uint8_t)(b >> 8);
    saltint[3] = (uint8_t)b;
    /* U1 = HMAC(P, S || INT(b)) */
    uint8_t k[64] = {0};
    if (passlen > 64)
      oracle_sha256(pass, passlen, k);
    else
      memcpy(k, pass, passlen);
    uint8_t ipad[64], opad[64];
    for (int i = 0; i < 64; i++) {
      ipad[i] = k[i] ^ 0x36;
      opad[i] = k[i] ^ 0x5c;
    }
    sha256_ctx c;
    uint8_t inner[32];
    sha256_init(&c);
    sha256_update(&c, ipad, 64);
    sha256_update(&c, salt, saltlen);
    sha256_update(&c, saltint, 4);
    sha256_final(&c, inner);
    sha256_init(&c);
    sha256_update(&c, opad, 64);
    sha256_update(&c, inner, 32);
    sha256_final(&c, u);
    memcpy(t, u, 32);
    for (uint32_t it = 1; it < iters; it++) {
      oracle_hmac_sha256(pass, passlen, u, 32, u);
      for (int i = 0; i < 32; i++) t[i] ^= u[i];
    }
    size_t off = (size_t)(b - 1) * 32;
    size_t take = outlen - off < 32 ? outlen - off : 32;
    memcpy(out + off, t, take);
  }
}
