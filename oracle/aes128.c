/* AES-128 encryption (FIPS-197) — the proving scan's per-nonce PRF
 * (post-rs prover; SURVEY.md §8(a) proving row).  S-box is derived
 * algebraically (GF(2^8) inverse + affine transform), not transcribed, and
 * pinned by the FIPS-197 C.1 vector in tests/test_kats.py.
 * Part of the CPU oracle (test infrastructure only — see oracle.h header). */
#include "oracle.h"
#include <string.h>

static uint8_t SBOX[256];
static int sbox_ready = 0;

static uint8_t gf_mul(uint8_t a, uint8_t b) {
  uint8_t r = 0;
  while (b) {
    if (b & 1) r ^= a;
    uint8_t hi = a & 0x80;
    a <<= 1;
    if (hi) a ^= 0x1b;
    b >>= 1;
  }
  return r;
}

static void sbox_init(void) {
  if (sbox_ready) return;
  for (int i = 0; i < 256; i++) {
    uint8_t inv = 0;
    if (i) { /* brute-force inverse in GF(2^8) */
      for (int j = 1; j < 256; j++)
        if (gf_mul((uint8_t)i, (uint8_t)j) == 1) { inv = (uint8_t)j; break; }
    }
    uint8_t b = inv;
    uint8_t s = b;
    for (int k = 1; k <= 4; k++)
      s ^= (uint8_t)((b << k) | (b >> (8 - k)));
    SBOX[i] = s ^ 0x63;
  }
  sbox_ready = 1;
}

static void key_expand(const uint8_t key[16], uint8_t rk[176]) {
  sbox_init();
  memcpy(rk, key, 16);
  uint8_t rcon = 1;
  for (int i = 16; i < 176; i += 4) {
    uint8_t t[4];
    memcpy(t, rk + i - 4, 4);
    if (i % 16 == 0) {
      uint8_t tmp = t[0];
      t[0] = SBOX[t[1]] ^ rcon;
      t[1] = SBOX[t[2]];
      t[2] = SBOX[t[3]];
      t[3] = SBOX[tmp];
      rcon = gf_mul(rcon, 2);
    }
    for (int k = 0; k < 4; k++) rk[i + k] = rk[i - 16 + k] ^ t[k];
  }
}

void oracle_aes128_enc_block(const uint8_t key[16], const uint8_t in[16],
                             uint8_t out[16]) {
  uint8_t rk[176];
  key_expand(key, rk);
  uint8_t s[16];
  for (int i = 0; i < 16; i++) s[i] = in[i] ^ rk[i];
  for (int round = 1; round <= 10; round++) {
    uint8_t t[16];
    /* SubBytes + ShiftRows (column-major state: s[r + 4c]) */
    for (int c = 0; c < 4; c++)
      for (int r = 0; r < 4; r++)
        t[r + 4 * c] = SBOX[s[r + 4 * ((c + r) & 3)]];
    if (round < 10) { /* MixColumns */
      for (int c = 0; c < 4; c++) {
        uint8_t a0 = t[4 * c], a1 = t[4 * c + 1], a2 = t[4 * c + 2],
                a3 = t[4 * c + 3];
        s[4 * c] = gf_mul(a0, 2) ^ gf_mul(a1, 3) ^ a2 ^ a3;
        s[4 * c + 1] = a0 ^ gf_mul(a1, 2) ^ gf_mul(a2, 3) ^ a3;
        s[4 * c + 2] = a0 ^ a1 ^ gf_mul(a2, 2) ^ gf_mul(a3, 3);
        s[4 * c + 3] = gf_mul(a0, 3) ^ a1 ^ a2 ^ gf_mul(a3, 2);
      }
    } else {
      memcpy(s, t, 16);
    }
    for (int i = 0; i < 16; i++) s[i] ^= rk[16 * round + i];
  }
  memcpy(out, s, 16);
}
