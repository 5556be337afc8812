/* BLAKE3 (single-chunk inputs, <= 1024 bytes) — used on this path only for
 * the commitment hash (node_id || commitment_atx_id, 64 bytes), AES key
 * derivation and the blake3-mode k2pow; all inputs are <= 1024 bytes so the
 * chunk tree is never needed (asserted).
 *
 * Part of the CPU oracle (test infrastructure only — see oracle.h header).
 * Pinning: official vectors for empty and 1-byte inputs embedded in
 * tests/test_kats.py, plus agreement on random inputs with the INDEPENDENT
 * second implementation in go-spacemesh_amd/csrc/blake3_impl.h. */
#include "oracle.h"
#include <assert.h>
#include <string.h>

#define B3_CHUNK_START (1u << 0)
#define B3_CHUNK_END (1u << 1)
#define B3_ROOT (1u << 3)

static const uint32_t B3_IV[8] = {0x6a09e667, 0xbb67ae85, 0x3c6ef372,
                                  0xa54ff53a, 0x510e527f, 0x9b05688c,
                                  0x1f83d9ab, 0x5be0cd19};

static const uint8_t B3_PERM[16] = {2, 6,  3,  10, 7, 0,  4,  13,
                                    1, 11, 12, 5,  9, 14, 15, 8};

static inline uint32_t rotr32(uint32_t x, unsigned n) {
  return (x >> n) | (x << (32 - n));
}

static inline void g(uint32_t *v, int a, int b, int c, int d, uint32_t mx,
                     uint32_t my) {
  v[a] = v[a] + v[b] + mx;
  v[d] = rotr32(v[d] ^ v[a], 16);
  v[c] = v[c] + v[d];
  v[b] = rotr32(v[b] ^ v[c], 12);
  v[a] = v[a] + v[b] + my;
  v[d] = rotr32(v[d] ^ v[a], 8);
  v[c] = v[c] + v[d];
  v[b] = rotr32(v[b] ^ v[c], 7);
}

/* One compression; writes the full 16-word extended output to out16. */
static void b3_compress(const uint32_t h[8], const uint32_t m_in[16],
                        uint64_t t, uint32_t block_len, uint32_t flags,
                        uint32_t out16[16]) {
  uint32_t v[16], m[16], mp[16];
  memcpy(m, m_in, 64);
  for (int i = 0; i < 8; i++) v[i] = h[i];
  v[8] = B3_IV[0];
  v[9] = B3_IV[1];
  v[10] = B3_IV[2];
  v[11] = B3_IV[3];
  v[12] = (uint32_t)t;
  v[13] = (uint32_t)(t >> 32);
  v[14] = block_len;
  v[15] = flags;
  for (int round = 0; round < 7; round++) {
    g(v, 0, 4, 8, 12, m[0], m[1]);
    g(v, 1, 5, 9, 13, m[2], m[3]);
    g(v, 2, 6, 10, 14, m[4], m[5]);
    g(v, 3, 7, 11, 15, m[6], m[7]);
    g(v, 0, 5, 10, 15, m[8], m[9]);
    g(v, 1, 6, 11, 12, m[10], m[11]);
    g(v, 2, 7, 8, 13, m[12], m[13]);
    g(v, 3, 4, 9, 14, m[14], m[15]);
    if (round < 6) {
      for (int i = 0; i < 16; i++) mp[i] = m[B3_PERM[i]];
      memcpy(m, mp, 64);
    }
  }
  for (int i = 0; i < 8; i++) {
    out16[i] = v[i] ^ v[i + 8];
    out16[i + 8] = v[i + 8] ^ h[i];
  }
}

static void load_block(const uint8_t *p, size_t len, uint32_t m[16]) {
  uint8_t buf[64] = {0};
  memcpy(buf, p, len);
  for (int i = 0; i < 16; i++)
    m[i] = (uint32_t)buf[4 * i] | ((uint32_t)buf[4 * i + 1] << 8) |
           ((uint32_t)buf[4 * i + 2] << 16) | ((uint32_t)buf[4 * i + 3] << 24);
}

/* Chain through the chunk's blocks; returns the h state before the final
 * block, plus the final block's words/len/flags so the caller can run the
 * root (output) compressions with varying t. */
static void b3_chunk_prepare(const uint8_t *msg, size_t len, uint32_t h[8],
                             uint32_t last_m[16], uint32_t *last_len,
                             uint32_t *last_flags) {
  assert(len <= 1024 && "single-chunk oracle blake3");
  memcpy(h, B3_IV, 32);
  size_t nblocks = len == 0 ? 1 : (len + 63) / 64;
  uint32_t out16[16];
  for (size_t b = 0; b + 1 < nblocks; b++) {
    uint32_t m[16];
    load_block(msg + b * 64, 64, m);
    uint32_t flags = (b == 0 ? B3_CHUNK_START : 0);
    b3_compress(h, m, 0, 64, flags, out16);
    memcpy(h, out16, 32);
  }
  size_t last_off = (nblocks - 1) * 64;
  uint32_t llen = (uint32_t)(len - last_off);
  load_block(msg + last_off, llen, last_m);
  *last_len = llen;
  *last_flags = (nblocks == 1 ? B3_CHUNK_START : 0) | B3_CHUNK_END | B3_ROOT;
}

void oracle_blake3_xof(const uint8_t *msg, size_t len, uint8_t *out,
                       size_t outlen) {
  uint32_t h[8], m[16], llen, flags;
  b3_chunk_prepare(msg, len, h, m, &llen, &flags);
  uint64_t t = 0;
  size_t off = 0;
  uint32_t out16[16];
  while (off < outlen) {
    b3_compress(h, m, t, llen, flags, out16);
    uint8_t block[64];
    for (int i = 0; i < 16; i++) {
      block[4 * i] = (uint8_t)out16[i];
      block[4 * i + 1] = (uint8_t)(out16[i] >> 8);
      block[4 * i + 2] = (uint8_t)(out16[i] >> 16);
      block[4 * i + 3] = (uint8_t)(out16[i] >> 24);
    }
    size_t take = outlen - off < 64 ? outlen - off : 64;
    memcpy(out + off, block, take);
    off += take;
    t++;
  }
}

void oracle_blake3(const uint8_t *msg, size_t len, uint8_t out[32]) {
  oracle_blake3_xof(msg, len, out, 32);
}
