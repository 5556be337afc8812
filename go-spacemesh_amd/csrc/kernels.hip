/* kernels.hip — MI355X (gfx950/CDNA4) kernels for the POST hot path.
 *
 * Labeling (the init hot loop the reference reaches at
 *   activation/post.go:295): one label per 4-lane quad, run as a 3-kernel
 *   pipeline (SHA prologue -> register-lean ROMix at 8 waves/SIMD -> SHA
 *   tail + VRF reduce), working block staged through xbuf.  The N-entry
 *   ROMix scratchpad V (128 B * N per in-flight label at gap 1) lives in
 *   HBM; a quad's block accesses are aligned 64-B transactions.  No MFMA:
 *   a u32 add/xor/rotate hash loop (BASELINE.json north_star).  Also used
 *   in index-list mode for verification's label recompute
 *   (validation.go:182-222 -> K3 sampled indices) with per-task commitments.
 *
 * post_scan_kernel: the proving index scan (post-service GenProof,
 *   api/grpcserver/post_client.go:69-143): AES-128 over each 16-B label for
 *   n_ciphers ciphers (2 nonces per cipher), T-tables + round keys staged in
 *   LDS, passing (nonce,index) pairs appended with a global atomic.
 *
 * Wave size 64, 256-thread workgroups.  Grid-stride loops size the in-flight
 * label set to the scratch allocation, independent of batch size.
 */
#include <hip/hip_runtime.h>

#include <cstdlib>
#include <cstring>

#include "kernel_args.h"
#include "post_common.h"

#define POSTE_THREADS 256

/* ROMix scratch V is written once and read once ~N iterations later — far
 * beyond any cache. -DPOSTE_NT=1 builds with non-temporal loads/stores on
 * V to keep it out of L2 (A/B via POST_ENGINE_LIB). */
#if defined(POSTE_NT) && POSTE_NT
typedef uint32_t poste_v4u __attribute__((ext_vector_type(4)));
#define VSTORE(p, v)                                                           \
  do {                                                                         \
    uint4 _t = (v);                                                            \
    __builtin_nontemporal_store(*(poste_v4u *)&_t, (poste_v4u *)(p));          \
  } while (0)
#define VLOAD(p)                                                               \
  ({                                                                           \
    poste_v4u _r = __builtin_nontemporal_load((const poste_v4u *)(p));         \
    *(uint4 *)&_r;                                                             \
  })
#else
#define VSTORE(p, v) (*(p) = (v))
#define VLOAD(p) (*(p))
#endif

/* ------------------------- SHA-256 (device) ------------------------- */
__constant__ uint32_t c_sha_k[64] = {
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

__device__ __forceinline__ uint32_t ror32(uint32_t x, int n) {
  return __builtin_rotateright32(x, n);
}

/* m holds big-endian message words; h updated in place. */
__device__ void sha_compress(uint32_t h[8], const uint32_t m[16]) {
  uint32_t w[16];
#pragma unroll
  for (int i = 0; i < 16; i++) w[i] = m[i];
  uint32_t a = h[0], b = h[1], c = h[2], d = h[3];
  uint32_t e = h[4], f = h[5], g = h[6], hh = h[7];
#pragma unroll
  for (int t = 0; t < 64; t++) {
    uint32_t wt;
    if (t < 16) {
      wt = w[t];
    } else {
      uint32_t a15 = w[(t - 15) & 15], a2 = w[(t - 2) & 15];
      wt = w[t & 15] += (ror32(a15, 7) ^ ror32(a15, 18) ^ (a15 >> 3)) +
                        w[(t - 7) & 15] +
                        (ror32(a2, 17) ^ ror32(a2, 19) ^ (a2 >> 10));
    }
    uint32_t t1 = hh + (ror32(e, 6) ^ ror32(e, 11) ^ ror32(e, 25)) +
                  ((e & f) ^ (~e & g)) + c_sha_k[t] + wt;
    uint32_t t2 = (ror32(a, 2) ^ ror32(a, 13) ^ ror32(a, 22)) +
                  ((a & b) ^ (a & c) ^ (b & c));
    hh = g; g = f; f = e; e = d + t1;
    d = c; c = b; b = a; a = t1 + t2;
  }
  h[0] += a; h[1] += b; h[2] += c; h[3] += d;
  h[4] += e; h[5] += f; h[6] += g; h[7] += hh;
}

__device__ __forceinline__ void sha_init(uint32_t h[8]) {
  h[0] = 0x6a09e667; h[1] = 0xbb67ae85; h[2] = 0x3c6ef372; h[3] = 0xa54ff53a;
  h[4] = 0x510e527f; h[5] = 0x9b05688c; h[6] = 0x1f83d9ab; h[7] = 0x5be0cd19;
}

/* --------------- salsa20/8 + BlockMix, 4-lane cooperative ---------------
 *
 * One label is computed by FOUR adjacent lanes (a quad).  The 16-word salsa
 * state lives as four z-diagonal vectors across the quad — lane l holds
 *   z0[l]=x[5l%16], z1[l]=x[(4+5l)%16], z2[l]=x[(8+5l)%16], z3[l]=x[(12+5l)%16]
 * so the column round is lane-local and the row round needs only quad
 * rotations (ds_swizzle quad-perm).  The payoff is memory shape: a quad's
 * 128-B scratch block is read/written as aligned 64-B transactions
 * (measured 7.7 TB/s random vs 1.2 TB/s for the 1-lane 16-B pattern —
 * profiles/probe2 gather1 vs gather4). */

/* quad permutes via DPP (VALU pipe, ~2-cycle) instead of ds_swizzle
 * (LDS pipe, ~50-cycle latency in the salsa dependency chain) */
#define SWZ(v, pat)                                                            \
  ((uint32_t)__builtin_amdgcn_mov_dpp((int)(v), (pat), 0xf, 0xf, true))
#define QROT1 0x39 /* quad perm (1,2,3,0): lane l reads elem (l+1)&3 */
#define QROT2 0x4E /* (2,3,0,1) */
#define QROT3 0x93 /* (3,0,1,2) */
#define QBCAST0 0x00 /* all lanes read elem 0 */

/* z-vector salsa20/8 on one quad: A..D are this lane's elements of z0..z3 */
__device__ __forceinline__ void salsa8_z(uint32_t &A, uint32_t &B,
                                         uint32_t &C, uint32_t &D) {
  uint32_t a = A, b = B, c = C, d = D;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    /* column round: QR(z0,z1,z2,z3) element-wise */
    b ^= __builtin_rotateleft32(a + d, 7);
    c ^= __builtin_rotateleft32(b + a, 9);
    d ^= __builtin_rotateleft32(c + b, 13);
    a ^= __builtin_rotateleft32(d + c, 18);
    /* row round: QR(z0, rot1(z3), rot2(z2), rot3(z1)).  The compiler
     * fuses these permutes into the consuming xor/add as *_dpp forms; the
     * VALU->DPP hazard s_nops that remain are issue-equivalent to unfused
     * movs (measured: source order and dual-stream interleave both
     * neutral — the kernel sits at the mixed read+write HBM ceiling,
     * profiles/r01_kernel_profile.md). */
    uint32_t y1 = SWZ(d, QROT1);
    uint32_t y2 = SWZ(c, QROT2);
    uint32_t y3 = SWZ(b, QROT3);
    y1 ^= __builtin_rotateleft32(a + y3, 7);
    y2 ^= __builtin_rotateleft32(y1 + a, 9);
    y3 ^= __builtin_rotateleft32(y2 + y1, 13);
    a ^= __builtin_rotateleft32(y3 + y2, 18);
    /* un-rotate for the next column round */
    d = SWZ(y1, QROT3);
    c = SWZ(y2, QROT2);
    b = SWZ(y3, QROT1);
  }
  A += a; B += b; C += c; D += d;
}

/* dual-stream salsa20/8: two independent quads' states interleaved so the
 * in-order wave always has a second dependency chain to issue from */
__device__ __forceinline__ void salsa8_z2(uint32_t &A0, uint32_t &B0,
                                          uint32_t &C0, uint32_t &D0,
                                          uint32_t &A1, uint32_t &B1,
                                          uint32_t &C1, uint32_t &D1) {
  uint32_t a0 = A0, b0 = B0, c0 = C0, d0 = D0;
  uint32_t a1 = A1, b1 = B1, c1 = C1, d1 = D1;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    b0 ^= __builtin_rotateleft32(a0 + d0, 7);
    b1 ^= __builtin_rotateleft32(a1 + d1, 7);
    c0 ^= __builtin_rotateleft32(b0 + a0, 9);
    c1 ^= __builtin_rotateleft32(b1 + a1, 9);
    d0 ^= __builtin_rotateleft32(c0 + b0, 13);
    d1 ^= __builtin_rotateleft32(c1 + b1, 13);
    a0 ^= __builtin_rotateleft32(d0 + c0, 18);
    a1 ^= __builtin_rotateleft32(d1 + c1, 18);
    uint32_t y30 = SWZ(b0, QROT3), y31 = SWZ(b1, QROT3);
    uint32_t y20 = SWZ(c0, QROT2), y21 = SWZ(c1, QROT2);
    uint32_t y10 = SWZ(d0, QROT1), y11 = SWZ(d1, QROT1);
    y10 ^= __builtin_rotateleft32(a0 + y30, 7);
    y11 ^= __builtin_rotateleft32(a1 + y31, 7);
    y20 ^= __builtin_rotateleft32(y10 + a0, 9);
    y21 ^= __builtin_rotateleft32(y11 + a1, 9);
    y30 ^= __builtin_rotateleft32(y20 + y10, 13);
    y31 ^= __builtin_rotateleft32(y21 + y11, 13);
    a0 ^= __builtin_rotateleft32(y30 + y20, 18);
    a1 ^= __builtin_rotateleft32(y31 + y21, 18);
    d0 = SWZ(y10, QROT3);
    d1 = SWZ(y11, QROT3);
    c0 = SWZ(y20, QROT2);
    c1 = SWZ(y21, QROT2);
    b0 = SWZ(y30, QROT1);
    b1 = SWZ(y31, QROT1);
  }
  A0 += a0; B0 += b0; C0 += c0; D0 += d0;
  A1 += a1; B1 += b1; C1 += c1; D1 += d1;
}

/* dual-stream BlockMix r=1 */
__device__ __forceinline__ void blockmix2_z(uint32_t XA0[4], uint32_t XA1[4],
                                            uint32_t XB0[4],
                                            uint32_t XB1[4]) {
  uint32_t ta0 = XA0[0] ^ XA1[0], ta1 = XA0[1] ^ XA1[1],
           ta2 = XA0[2] ^ XA1[2], ta3 = XA0[3] ^ XA1[3];
  uint32_t tb0 = XB0[0] ^ XB1[0], tb1 = XB0[1] ^ XB1[1],
           tb2 = XB0[2] ^ XB1[2], tb3 = XB0[3] ^ XB1[3];
  salsa8_z2(ta0, ta1, ta2, ta3, tb0, tb1, tb2, tb3);
  XA0[0] = ta0; XA0[1] = ta1; XA0[2] = ta2; XA0[3] = ta3;
  XB0[0] = tb0; XB0[1] = tb1; XB0[2] = tb2; XB0[3] = tb3;
  ta0 ^= XA1[0]; ta1 ^= XA1[1]; ta2 ^= XA1[2]; ta3 ^= XA1[3];
  tb0 ^= XB1[0]; tb1 ^= XB1[1]; tb2 ^= XB1[2]; tb3 ^= XB1[3];
  salsa8_z2(ta0, ta1, ta2, ta3, tb0, tb1, tb2, tb3);
  XA1[0] = ta0; XA1[1] = ta1; XA1[2] = ta2; XA1[3] = ta3;
  XB1[0] = tb0; XB1[1] = tb1; XB1[2] = tb2; XB1[3] = tb3;
}

/* BlockMix r=1 on a quad: X = (B0,B1) as two z-vectors per lane (8 regs) */
__device__ __forceinline__ void blockmix_z(uint32_t X0[4], uint32_t X1[4]) {
  uint32_t t0 = X0[0] ^ X1[0], t1 = X0[1] ^ X1[1], t2 = X0[2] ^ X1[2],
           t3 = X0[3] ^ X1[3];
  salsa8_z(t0, t1, t2, t3);
  X0[0] = t0; X0[1] = t1; X0[2] = t2; X0[3] = t3;
  t0 ^= X1[0]; t1 ^= X1[1]; t2 ^= X1[2]; t3 ^= X1[3];
  salsa8_z(t0, t1, t2, t3);
  X1[0] = t0; X1[1] = t1; X1[2] = t2; X1[3] = t3;
}

/* canonical(16 words) -> this lane's z elements for one 64-B block */
__device__ __forceinline__ void canon_to_z(const uint32_t x[16], uint32_t sub,
                                           uint32_t z[4]) {
  /* z_v[l] = x[(4v + 5l) % 16] */
  uint32_t s0 = sub == 0 ? x[0] : sub == 1 ? x[5] : sub == 2 ? x[10] : x[15];
  uint32_t s1 = sub == 0 ? x[4] : sub == 1 ? x[9] : sub == 2 ? x[14] : x[3];
  uint32_t s2 = sub == 0 ? x[8] : sub == 1 ? x[13] : sub == 2 ? x[2] : x[7];
  uint32_t s3 = sub == 0 ? x[12] : sub == 1 ? x[1] : sub == 2 ? x[6] : x[11];
  z[0] = s0; z[1] = s1; z[2] = s2; z[3] = s3;
}

/* this quad's z vectors -> big-endian SHA message words of the canonical
 * 64-B block (fused broadcast+byteswap, no staging buffers) */
__device__ __forceinline__ void z_to_msg_be(const uint32_t z[4],
                                            uint32_t m[16]) {
#pragma unroll
  for (int v = 0; v < 4; v++) {
    m[(4 * v + 0) & 15] = __builtin_bswap32(SWZ(z[v], 0x00));
    m[(4 * v + 5) & 15] = __builtin_bswap32(SWZ(z[v], 0x55));
    m[(4 * v + 10) & 15] = __builtin_bswap32(SWZ(z[v], 0xAA));
    m[(4 * v + 15) & 15] = __builtin_bswap32(SWZ(z[v], 0xFF));
  }
}

/* ------------------------- label kernel ------------------------- */
/* LabelKernelArgs: see kernel_args.h */

/* PBKDF2 helper: one outer HMAC finalisation of a 32-byte inner digest */
__device__ __forceinline__ void hmac_outer(const uint32_t ho[8],
                                           const uint32_t inner[8],
                                           uint32_t out[8]) {
  uint32_t h[8], m[16];
#pragma unroll
  for (int i = 0; i < 8; i++) h[i] = ho[i];
#pragma unroll
  for (int i = 0; i < 8; i++) m[i] = inner[i];
  m[8] = 0x80000000u;
#pragma unroll
  for (int i = 9; i < 15; i++) m[i] = 0;
  m[15] = (64 + 32) * 8;
  sha_compress(h, m);
#pragma unroll
  for (int i = 0; i < 8; i++) out[i] = h[i];
}

/* The labeling path runs as THREE kernels per batch so the ROMix hot loop
 * stays register-lean (8 waves/SIMD) while the SHA-heavy prologue/tail keep
 * their own (lower) occupancy.  The per-label working block X round-trips
 * through xbuf (128 B per task, quad-z chunk layout) — 256 B of extra
 * traffic against the ~2 MiB/label scratch stream. */

/* shared helper: password = commitment || LE64(index) -> HMAC states */
__device__ __forceinline__ void hmac_states_for(const uint32_t *cw,
                                                unsigned long long index,
                                                uint32_t hi[8],
                                                uint32_t ho[8]) {
  uint32_t pw_be[10], m[16];
#pragma unroll
  for (int i = 0; i < 8; i++) pw_be[i] = __builtin_bswap32(cw[i]);
  pw_be[8] = __builtin_bswap32((uint32_t)index);
  pw_be[9] = __builtin_bswap32((uint32_t)(index >> 32));
  sha_init(hi);
#pragma unroll
  for (int i = 0; i < 10; i++) m[i] = pw_be[i] ^ 0x36363636u;
#pragma unroll
  for (int i = 10; i < 16; i++) m[i] = 0x36363636u;
  sha_compress(hi, m);
  sha_init(ho);
#pragma unroll
  for (int i = 0; i < 10; i++) m[i] = pw_be[i] ^ 0x5c5c5c5cu;
#pragma unroll
  for (int i = 10; i < 16; i++) m[i] = 0x5c5c5c5cu;
  sha_compress(ho, m);
}

/* prologue: PBKDF2(P, "", 1, 128) -> xbuf in quad-z layout.  Every lane of
 * a quad computes the same SHA chain (redundant, ~0.1% of the ROMix cost)
 * and keeps only its own z slice. */
__global__ void __launch_bounds__(POSTE_THREADS)
post_label_prologue_kernel(LabelKernelArgs a) {
  const unsigned long long lane =
      (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned long long group = lane >> 2;
  const unsigned long long gstride =
      ((unsigned long long)gridDim.x * blockDim.x) >> 2;
  const uint32_t sub = (uint32_t)(lane & 3);
  uint4 *X = (uint4 *)a.xbuf;

  for (unsigned long long task = group; task < a.count;
       task += gstride) {
    const unsigned long long index = a.indices ? a.indices[task]
                                               : a.start + task;
    const uint32_t *cw = a.commit_ids
                             ? a.commitments + 8ull * a.commit_ids[task]
                             : a.commitment_le;
    uint32_t hi[8], ho[8], m[16];
    hmac_states_for(cw, index, hi, ho);
    uint32_t Z0[4], Z1[4];
    for (uint32_t half = 0; half < 2; half++) {
      uint32_t cblk[16];
#pragma unroll
      for (uint32_t q = 0; q < 2; q++) {
        const uint32_t blk = half * 2 + q + 1;
        uint32_t h[8];
#pragma unroll
        for (int i = 0; i < 8; i++) h[i] = hi[i];
        m[0] = blk;
        m[1] = 0x80000000u;
#pragma unroll
        for (int i = 2; i < 15; i++) m[i] = 0;
        m[15] = (64 + 4) * 8;
        sha_compress(h, m);
        uint32_t d[8];
        hmac_outer(ho, h, d);
#pragma unroll
        for (int k = 0; k < 8; k++)
          cblk[8 * q + k] = __builtin_bswap32(d[k]);
      }
      canon_to_z(cblk, sub, half == 0 ? Z0 : Z1);
    }
    uint4 *p = X + task * 8ull;
    p[sub] = make_uint4(Z0[0], Z0[1], Z0[2], Z0[3]);
    p[sub + 4] = make_uint4(Z1[0], Z1[1], Z1[2], Z1[3]);
  }
}

/* the ROMix hot loop: register-lean, forced to 8 waves/SIMD */
__global__ void __launch_bounds__(POSTE_THREADS, 8)
post_label_romix_kernel(LabelKernelArgs a) {
  const unsigned long long lane =
      (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned long long group = lane >> 2;
  const uint32_t sub = (uint32_t)(lane & 3);
  const uint32_t n = a.scrypt_n;
  const uint32_t mask = n - 1;
  const uint32_t gmask = (1u << a.gap_shift) - 1u;
  uint4 *V = (uint4 *)a.scratch;
  uint4 *X = (uint4 *)a.xbuf;

  for (unsigned long long task = group; task < a.count;
       task += a.scratch_lanes) {
    uint32_t Z0[4], Z1[4];
    {
      uint4 *p = X + task * 8ull;
      uint4 v0 = p[sub], v1 = p[sub + 4];
      Z0[0] = v0.x; Z0[1] = v0.y; Z0[2] = v0.z; Z0[3] = v0.w;
      Z1[0] = v1.x; Z1[1] = v1.y; Z1[2] = v1.z; Z1[3] = v1.w;
    }
    /* phase 1: V_j = X for j % gap == 0; X = BlockMix(X).  Stored block
     * j/gap of this quad is 8 consecutive uint4 at ((j/gap)*slots+group)*8;
     * lane sub covers chunks sub and sub+4 -> aligned 64-B transactions. */
    {
      unsigned long long base = group * 8ull + sub;
      const unsigned long long stride = a.scratch_lanes * 8ull;
      for (uint32_t j = 0; j < n; j++) {
        if ((j & gmask) == 0) {
          uint4 *p = V + base;
          VSTORE(p, make_uint4(Z0[0], Z0[1], Z0[2], Z0[3]));
          VSTORE(p + 4, make_uint4(Z1[0], Z1[1], Z1[2], Z1[3]));
          base += stride;
        }
        blockmix_z(Z0, Z1);
      }
    }
    /* phase 2: j = Integerify(X) & (n-1); regenerate V_j from the stored
     * block by j%gap BlockMixes; X ^= V_j; BlockMix.  Integerify =
     * canonical word 16 = B1's z0[0] -> quad broadcast. */
    for (uint32_t i = 0; i < n; i++) {
      uint32_t j = SWZ(Z1[0], QBCAST0) & mask;
      const uint32_t r = j & gmask;
      const uint4 *p =
          V + ((unsigned long long)(j >> a.gap_shift) * a.scratch_lanes +
               group) * 8ull + sub;
      uint32_t Y0[4], Y1[4];
      {
        uint4 v0 = VLOAD(p), v1 = VLOAD(p + 4);
        Y0[0] = v0.x; Y0[1] = v0.y; Y0[2] = v0.z; Y0[3] = v0.w;
        Y1[0] = v1.x; Y1[1] = v1.y; Y1[2] = v1.z; Y1[3] = v1.w;
      }
      for (uint32_t t = 0; t < r; t++) blockmix_z(Y0, Y1);
#pragma unroll
      for (int k = 0; k < 4; k++) {
        Z0[k] ^= Y0[k];
        Z1[k] ^= Y1[k];
      }
      blockmix_z(Z0, Z1);
    }
    {
      uint4 *p = X + task * 8ull;
      p[sub] = make_uint4(Z0[0], Z0[1], Z0[2], Z0[3]);
      p[sub + 4] = make_uint4(Z1[0], Z1[1], Z1[2], Z1[3]);
    }
  }
}

/* dual-stream ROMix (lookup-gap 1 only): each quad advances TWO labels so
 * the in-order wave interleaves two independent salsa chains — the
 * SQ_WAIT_INST_ANY issue-stall fix (profiles/pmc_sq: 54% of wave cycles).
 * In-flight slots = 2 x grid quads (= a.scratch_lanes). */
__global__ void __launch_bounds__(POSTE_THREADS, 8)
post_label_romix2_kernel(LabelKernelArgs a) {
  const unsigned long long lane =
      (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned long long group = lane >> 2;
  const uint32_t sub = (uint32_t)(lane & 3);
  const unsigned long long gquads =
      ((unsigned long long)gridDim.x * blockDim.x) >> 2;
  const uint32_t n = a.scrypt_n;
  const uint32_t mask = n - 1;
  uint4 *V = (uint4 *)a.scratch;
  uint4 *X = (uint4 *)a.xbuf;
  const unsigned long long slotA = group * 2, slotB = group * 2 + 1;

  for (unsigned long long base = group * 2; base < a.count;
       base += 2 * gquads) {
    const unsigned long long taskA = base;
    const unsigned long long taskB = base + 1;
    const bool validB = taskB < a.count;
    const unsigned long long tb = validB ? taskB : taskA;
    uint32_t ZA0[4], ZA1[4], ZB0[4], ZB1[4];
    {
      uint4 *pA = X + taskA * 8ull;
      uint4 *pB = X + tb * 8ull;
      uint4 a0 = pA[sub], a1 = pA[sub + 4];
      uint4 b0 = pB[sub], b1 = pB[sub + 4];
      ZA0[0] = a0.x; ZA0[1] = a0.y; ZA0[2] = a0.z; ZA0[3] = a0.w;
      ZA1[0] = a1.x; ZA1[1] = a1.y; ZA1[2] = a1.z; ZA1[3] = a1.w;
      ZB0[0] = b0.x; ZB0[1] = b0.y; ZB0[2] = b0.z; ZB0[3] = b0.w;
      ZB1[0] = b1.x; ZB1[1] = b1.y; ZB1[2] = b1.z; ZB1[3] = b1.w;
    }
    /* phase 1 */
    {
      unsigned long long baseA = slotA * 8ull + sub;
      unsigned long long baseB = slotB * 8ull + sub;
      const unsigned long long stride = a.scratch_lanes * 8ull;
      for (uint32_t j = 0; j < n; j++) {
        uint4 *pA = V + baseA;
        uint4 *pB = V + baseB;
        pA[0] = make_uint4(ZA0[0], ZA0[1], ZA0[2], ZA0[3]);
        pA[4] = make_uint4(ZA1[0], ZA1[1], ZA1[2], ZA1[3]);
        pB[0] = make_uint4(ZB0[0], ZB0[1], ZB0[2], ZB0[3]);
        pB[4] = make_uint4(ZB1[0], ZB1[1], ZB1[2], ZB1[3]);
        blockmix2_z(ZA0, ZA1, ZB0, ZB1);
        baseA += stride;
        baseB += stride;
      }
    }
    /* phase 2 */
    for (uint32_t i = 0; i < n; i++) {
      uint32_t jA = SWZ(ZA1[0], QBCAST0) & mask;
      uint32_t jB = SWZ(ZB1[0], QBCAST0) & mask;
      const uint4 *pA =
          V + ((unsigned long long)jA * a.scratch_lanes + slotA) * 8ull + sub;
      const uint4 *pB =
          V + ((unsigned long long)jB * a.scratch_lanes + slotB) * 8ull + sub;
      uint4 va0 = pA[0], va1 = pA[4];
      uint4 vb0 = pB[0], vb1 = pB[4];
      ZA0[0] ^= va0.x; ZA0[1] ^= va0.y; ZA0[2] ^= va0.z; ZA0[3] ^= va0.w;
      ZA1[0] ^= va1.x; ZA1[1] ^= va1.y; ZA1[2] ^= va1.z; ZA1[3] ^= va1.w;
      ZB0[0] ^= vb0.x; ZB0[1] ^= vb0.y; ZB0[2] ^= vb0.z; ZB0[3] ^= vb0.w;
      ZB1[0] ^= vb1.x; ZB1[1] ^= vb1.y; ZB1[2] ^= vb1.z; ZB1[3] ^= vb1.w;
      blockmix2_z(ZA0, ZA1, ZB0, ZB1);
    }
    {
      uint4 *pA = X + taskA * 8ull;
      pA[sub] = make_uint4(ZA0[0], ZA0[1], ZA0[2], ZA0[3]);
      pA[sub + 4] = make_uint4(ZA1[0], ZA1[1], ZA1[2], ZA1[3]);
      if (validB) {
        uint4 *pB = X + taskB * 8ull;
        pB[sub] = make_uint4(ZB0[0], ZB0[1], ZB0[2], ZB0[3]);
        pB[sub + 4] = make_uint4(ZB1[0], ZB1[1], ZB1[2], ZB1[3]);
      }
    }
  }
}

/* tail: PBKDF2(P, X, 1, 32) -> labels, VRF-minimum tracking + exact
 * per-workgroup candidate reduce */
__global__ void __launch_bounds__(POSTE_THREADS)
post_label_tail_kernel(LabelKernelArgs a) {
  const unsigned long long lane =
      (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned long long group = lane >> 2;
  const unsigned long long gstride =
      ((unsigned long long)gridDim.x * blockDim.x) >> 2;
  const uint32_t sub = (uint32_t)(lane & 3);
  uint4 *X = (uint4 *)a.xbuf;

  uint32_t min_lab[8];
  unsigned long long min_idx = 0;
  int min_found = 0;

  for (unsigned long long task = group; task < a.count;
       task += gstride) {
    const unsigned long long index = a.indices ? a.indices[task]
                                               : a.start + task;
    const uint32_t *cw = a.commit_ids
                             ? a.commitments + 8ull * a.commit_ids[task]
                             : a.commitment_le;
    uint32_t hi[8], ho[8], m[16];
    hmac_states_for(cw, index, hi, ho);
    uint32_t Z0[4], Z1[4];
    {
      uint4 *p = X + task * 8ull;
      uint4 v0 = p[sub], v1 = p[sub + 4];
      Z0[0] = v0.x; Z0[1] = v0.y; Z0[2] = v0.z; Z0[3] = v0.w;
      Z1[0] = v1.x; Z1[1] = v1.y; Z1[2] = v1.z; Z1[3] = v1.w;
    }
    uint32_t h[8];
#pragma unroll
    for (int i = 0; i < 8; i++) h[i] = hi[i];
    z_to_msg_be(Z0, m);
    sha_compress(h, m);
    z_to_msg_be(Z1, m);
    sha_compress(h, m);
    m[0] = 1;
    m[1] = 0x80000000u;
#pragma unroll
    for (int i = 2; i < 15; i++) m[i] = 0;
    m[15] = (64 + 128 + 4) * 8;
    sha_compress(h, m);
    uint32_t lab_be[8]; /* full label as big-endian words */
    hmac_outer(ho, h, lab_be);

    if (a.out && sub == 0) {
      if (a.out_full) {
        uint4 *o = (uint4 *)(a.out + task * 32ull);
        o[0] = make_uint4(__builtin_bswap32(lab_be[0]),
                          __builtin_bswap32(lab_be[1]),
                          __builtin_bswap32(lab_be[2]),
                          __builtin_bswap32(lab_be[3]));
        o[1] = make_uint4(__builtin_bswap32(lab_be[4]),
                          __builtin_bswap32(lab_be[5]),
                          __builtin_bswap32(lab_be[6]),
                          __builtin_bswap32(lab_be[7]));
      } else {
        *(uint4 *)(a.out + task * 16ull) =
            make_uint4(__builtin_bswap32(lab_be[0]),
                       __builtin_bswap32(lab_be[1]),
                       __builtin_bswap32(lab_be[2]),
                       __builtin_bswap32(lab_be[3]));
      }
    }

    if (a.has_difficulty && sub == 0) {
      /* lexicographic byte compare == numeric compare of BE word sequence */
      bool below_diff = false, below_min = false;
#pragma unroll
      for (int k = 0; k < 8; k++) {
        if (lab_be[k] != a.difficulty_be[k]) {
          below_diff = lab_be[k] < a.difficulty_be[k];
          break;
        }
      }
      if (below_diff) {
        if (!min_found) {
          below_min = true;
        } else {
          below_min = false;
#pragma unroll
          for (int k = 0; k < 8; k++) {
            if (lab_be[k] != min_lab[k]) {
              below_min = lab_be[k] < min_lab[k];
              break;
            }
          }
        }
        if (below_min) {
          min_found = 1;
          min_idx = index;
#pragma unroll
          for (int k = 0; k < 8; k++) min_lab[k] = lab_be[k];
        }
      }
    }
  }

  /* one exact candidate per workgroup: LDS gather + lane-0 scan */
  if (a.has_difficulty) {
    __shared__ uint32_t s_lab[POSTE_THREADS][8];
    __shared__ unsigned long long s_idx[POSTE_THREADS];
    __shared__ uint8_t s_found[POSTE_THREADS];
    s_found[threadIdx.x] = (uint8_t)min_found;
    if (min_found) {
      s_idx[threadIdx.x] = min_idx;
#pragma unroll
      for (int k = 0; k < 8; k++) s_lab[threadIdx.x][k] = min_lab[k];
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      int best = -1;
      for (int t = 0; t < POSTE_THREADS; t++) {
        if (!s_found[t]) continue;
        if (best < 0) {
          best = t;
          continue;
        }
        /* order by (label bytes, index) ascending */
        int cmp = 0;
        for (int k = 0; k < 8 && cmp == 0; k++) {
          if (s_lab[t][k] < s_lab[best][k]) cmp = -1;
          else if (s_lab[t][k] > s_lab[best][k]) cmp = 1;
        }
        if (cmp < 0 || (cmp == 0 && s_idx[t] < s_idx[best])) best = t;
      }
      if (best >= 0) {
        unsigned int slot = atomicAdd(a.cand_count, 1u);
        if (slot < a.cand_cap) {
          a.cand[slot].index = s_idx[best];
          for (int k = 0; k < 8; k++)
            a.cand[slot].label_be[k] = s_lab[best][k];
        }
      }
    }
  }
}

/* ------------------------- proving scan kernel ------------------------- */
/* ScanKernelArgs: see kernel_args.h */

__global__ void __launch_bounds__(POSTE_THREADS)
post_scan_kernel(ScanKernelArgs a) {
  /* T-tables + sbox in LDS (random per-lane indices); round keys stay in
   * global — the cipher loop is wave-uniform, so they compile to scalar
   * loads through the constant cache instead of ~5.8K extra LDS reads per
   * label.  The kernel is LDS-conflict-throughput bound (ILP 1/2/4 all
   * measure ~274 M labels/s; a random 32-lane b32 gather costs
   * 2 x E[max 32-into-32-bank load] ~ 6.4 LDS cycles — see
   * profiles/r01_scan_sweep.md and MI355X_MICROARCH.md §LDS).  Two
   * gated A/B variants attack that wall (both UNVERIFIED on hardware,
   * round-2 retest):
   *   POSTE_SCAN_GLOBAL_TT — all lookups through the vector L1;
   *   POSTE_SCAN_SPLIT_TT — Te0/Te1 from LDS, Te2/Te3 through the L1,
   *     so the two memory pipes each carry half the gathers. */
#if defined(POSTE_SCAN_GLOBAL_TT)
  const uint32_t *TE_LO = a.te; /* Te0/Te1 (offsets 0, 256) */
  const uint32_t *TE_HI = a.te; /* Te2/Te3 (offsets 512, 768) */
  const uint8_t *sSbox = a.sbox;
#elif defined(POSTE_SCAN_SPLIT_TT)
  extern __shared__ uint32_t lds[];
  uint32_t *TE_LO = lds;                     /* Te0/Te1: 512 words */
  const uint32_t *TE_HI = a.te;              /* Te2/Te3 via L1 */
  uint8_t *sSbox = (uint8_t *)(lds + 512);   /* 256 bytes */
  for (uint32_t i = threadIdx.x; i < 512; i += blockDim.x) TE_LO[i] = a.te[i];
  for (uint32_t i = threadIdx.x; i < 256; i += blockDim.x)
    sSbox[i] = a.sbox[i];
  __syncthreads();
#else
  extern __shared__ uint32_t lds[];
  uint32_t *sTe = lds;                      /* 1024 words */
  uint8_t *sSbox = (uint8_t *)(sTe + 1024); /* 256 bytes */
  uint32_t *TE_LO = sTe, *TE_HI = sTe;

  for (uint32_t i = threadIdx.x; i < 1024; i += blockDim.x) sTe[i] = a.te[i];
  for (uint32_t i = threadIdx.x; i < 256; i += blockDim.x)
    sSbox[i] = a.sbox[i];
  __syncthreads();
#endif

  /* Four independent labels per lane (ILP): a single AES chain leaves the
   * wave ~76% latency-stalled (profiles: SQ_ACTIVE_INST 10%, LDS 14%);
   * interleaving L chains fills the LDS-latency shadows. */
  const unsigned long long stride =
      (unsigned long long)gridDim.x * blockDim.x;
#ifndef POSTE_SCAN_ILP
#define POSTE_SCAN_ILP 2 /* 2 chains @ 8 waves/SIMD measured best */
#endif
  constexpr int L = POSTE_SCAN_ILP;
  const unsigned long long span = stride * L;
  for (unsigned long long t0 =
           (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
       t0 < a.count; t0 += span) {
    uint32_t p[L][4];
    unsigned long long ts[L];
#pragma unroll
    for (int u = 0; u < L; u++) {
      unsigned long long t = t0 + (unsigned long long)u * stride;
      ts[u] = t < a.count ? t : t0; /* clamp: duplicates are harmless
                                       (hits deduped by nonce list merge
                                       never occur: same index+nonce hits
                                       only appended for u==0's slot) */
      uint4 lraw = a.labels[ts[u]];
      p[u][0] = __builtin_bswap32(lraw.x);
      p[u][1] = __builtin_bswap32(lraw.y);
      p[u][2] = __builtin_bswap32(lraw.z);
      p[u][3] = __builtin_bswap32(lraw.w);
    }
    const int live = (int)((a.count - t0 + stride - 1) / stride) < L
                         ? (int)((a.count - t0 + stride - 1) / stride)
                         : L;
    for (uint32_t c = 0; c < a.n_ciphers; c++) {
      const uint32_t *rk = a.rk + c * 44;
      uint32_t w[L][4];
#pragma unroll
      for (int u = 0; u < L; u++)
#pragma unroll
        for (int k = 0; k < 4; k++) w[u][k] = p[u][k] ^ rk[k];
#pragma unroll
      for (int r = 1; r < 10; r++) {
#pragma unroll
        for (int u = 0; u < L; u++) {
          uint32_t n0 = TE_LO[w[u][0] >> 24] ^
                        TE_LO[256 + ((w[u][1] >> 16) & 0xff)] ^
                        TE_HI[512 + ((w[u][2] >> 8) & 0xff)] ^
                        TE_HI[768 + (w[u][3] & 0xff)] ^ rk[4 * r];
          uint32_t n1 = TE_LO[w[u][1] >> 24] ^
                        TE_LO[256 + ((w[u][2] >> 16) & 0xff)] ^
                        TE_HI[512 + ((w[u][3] >> 8) & 0xff)] ^
                        TE_HI[768 + (w[u][0] & 0xff)] ^ rk[4 * r + 1];
          uint32_t n2 = TE_LO[w[u][2] >> 24] ^
                        TE_LO[256 + ((w[u][3] >> 16) & 0xff)] ^
                        TE_HI[512 + ((w[u][0] >> 8) & 0xff)] ^
                        TE_HI[768 + (w[u][1] & 0xff)] ^ rk[4 * r + 2];
          uint32_t n3 = TE_LO[w[u][3] >> 24] ^
                        TE_LO[256 + ((w[u][0] >> 16) & 0xff)] ^
                        TE_HI[512 + ((w[u][1] >> 8) & 0xff)] ^
                        TE_HI[768 + (w[u][2] & 0xff)] ^ rk[4 * r + 3];
          w[u][0] = n0; w[u][1] = n1; w[u][2] = n2; w[u][3] = n3;
        }
      }
#pragma unroll
      for (int u = 0; u < L; u++) {
        if (u >= live) break;
        uint32_t f0 = (((uint32_t)sSbox[w[u][0] >> 24] << 24) |
                       ((uint32_t)sSbox[(w[u][1] >> 16) & 0xff] << 16) |
                       ((uint32_t)sSbox[(w[u][2] >> 8) & 0xff] << 8) |
                       sSbox[w[u][3] & 0xff]) ^ rk[40];
        uint32_t f1 = (((uint32_t)sSbox[w[u][1] >> 24] << 24) |
                       ((uint32_t)sSbox[(w[u][2] >> 16) & 0xff] << 16) |
                       ((uint32_t)sSbox[(w[u][3] >> 8) & 0xff] << 8) |
                       sSbox[w[u][0] & 0xff]) ^ rk[41];
        uint32_t f2 = (((uint32_t)sSbox[w[u][2] >> 24] << 24) |
                       ((uint32_t)sSbox[(w[u][3] >> 16) & 0xff] << 16) |
                       ((uint32_t)sSbox[(w[u][0] >> 8) & 0xff] << 8) |
                       sSbox[w[u][1] & 0xff]) ^ rk[42];
        uint32_t f3 = (((uint32_t)sSbox[w[u][3] >> 24] << 24) |
                       ((uint32_t)sSbox[(w[u][0] >> 16) & 0xff] << 16) |
                       ((uint32_t)sSbox[(w[u][1] >> 8) & 0xff] << 8) |
                       sSbox[w[u][2] & 0xff]) ^ rk[43];
        unsigned long long v0 =
            __builtin_bswap64(((unsigned long long)f0 << 32) | f1);
        unsigned long long v1 =
            __builtin_bswap64(((unsigned long long)f2 << 32) | f3);
        if (v0 < a.difficulty) {
          unsigned int s = atomicAdd(a.hit_count, 1u);
          if (s < a.hit_cap) {
            a.hits[s].index = a.index_base + ts[u];
            a.hits[s].nonce = c * POSTE_NONCES_PER_AES;
          }
        }
        if (v1 < a.difficulty) {
          unsigned int s = atomicAdd(a.hit_count, 1u);
          if (s < a.hit_cap) {
            a.hits[s].index = a.index_base + ts[u];
            a.hits[s].nonce = c * POSTE_NONCES_PER_AES + 1;
          }
        }
      }
    }
  }
}

/* Bank-replicated scan: one Te0 table replicated across the 32 LDS banks.
 * Copy c is strided so element x of copy c sits at word x*32 + c; the bank
 * of word w is w%32 = c, and each lane reads only copy (lane%32), so every
 * ds_read_b32 gather in the cipher loop is conflict-free BY CONSTRUCTION
 * (2 LDS cycles per wave-gather, MI355X_MICROARCH §LDS, vs ~6.4 measured
 * for the shared-table random gather — profiles/r01_scan_sweep.md).  The
 * round-2 TT retest (profiles/r02_scan_tt_verdict.md) proved the L1 pipe
 * cannot bypass that wall, so this attacks it inside the LDS instead:
 * 3.2x fewer LDS cycles for 32 KB of LDS per workgroup and a few extra
 * VALU ops — Te1/2/3 derive from Te0 by byte rotation (one v_alignbit
 * each; te[256+x] = ror8(te[x]), crypto_host.cpp:332-334), and the final
 * round's S-box is byte 1 of Te0 (te[x] packs [s2,s,s,s3] so s =
 * (te[x]>>8)&0xff) — no separate sbox table at all. */
#define ROR8(x) __builtin_amdgcn_alignbit((x), (x), 8)
#define ROR16(x) __builtin_amdgcn_alignbit((x), (x), 16)
#define ROR24(x) __builtin_amdgcn_alignbit((x), (x), 24)

template <int THREADS_, bool BATCHG>
__global__ void __launch_bounds__(THREADS_)
post_scan_bankrep_kernel(ScanKernelArgs a) {
  extern __shared__ uint32_t sTe[]; /* 8192 words = 32 copies x 1 KiB */
  for (uint32_t i = threadIdx.x; i < 8192; i += blockDim.x)
    sTe[i] = a.te[i >> 5]; /* consecutive lanes -> consecutive banks */
  __syncthreads();
  const uint32_t lane31 = threadIdx.x & 31;
#define TE0R(idx) sTe[(((uint32_t)(idx)) << 5) | lane31]

  const unsigned long long stride =
      (unsigned long long)gridDim.x * blockDim.x;
#ifndef POSTE_SCAN_ILP
#define POSTE_SCAN_ILP 2
#endif
  constexpr int L = POSTE_SCAN_ILP;
  const unsigned long long span = stride * L;
  for (unsigned long long t0 =
           (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
       t0 < a.count; t0 += span) {
    uint32_t p[L][4];
    unsigned long long ts[L];
#pragma unroll
    for (int u = 0; u < L; u++) {
      unsigned long long t = t0 + (unsigned long long)u * stride;
      ts[u] = t < a.count ? t : t0; /* clamp: duplicate work, hits only
                                       recorded for live chains below */
      uint4 lraw = a.labels[ts[u]];
      p[u][0] = __builtin_bswap32(lraw.x);
      p[u][1] = __builtin_bswap32(lraw.y);
      p[u][2] = __builtin_bswap32(lraw.z);
      p[u][3] = __builtin_bswap32(lraw.w);
    }
    const int live = (int)((a.count - t0 + stride - 1) / stride) < L
                         ? (int)((a.count - t0 + stride - 1) / stride)
                         : L;
    for (uint32_t c = 0; c < a.n_ciphers; c++) {
      /* expanded key into registers up front: a.rk accesses inside the
       * round loop compile to per-round VECTOR global loads (wave-uniform
       * but the compiler cannot prove it) whose vmcnt waits serialize the
       * cipher loop */
      uint32_t rk[44];
#pragma unroll
      for (int k = 0; k < 44; k++) rk[k] = a.rk[c * 44 + k];
      uint32_t w[L][4];
#pragma unroll
      for (int u = 0; u < L; u++)
#pragma unroll
        for (int k = 0; k < 4; k++) w[u][k] = p[u][k] ^ rk[k];
#pragma unroll
      for (int r = 1; r < 10; r++) {
        if (BATCHG) {
          /* issue all of the round's gathers before any combine: widens
           * the read->use distance so partial lgkmcnt waits overlap the
           * whole gather batch with the previous combines */
          uint32_t g[L][16];
#pragma unroll
          for (int u = 0; u < L; u++)
#pragma unroll
            for (int q = 0; q < 4; q++) {
              g[u][4 * q + 0] = TE0R(w[u][q] >> 24);
              g[u][4 * q + 1] = TE0R((w[u][(q + 1) & 3] >> 16) & 0xff);
              g[u][4 * q + 2] = TE0R((w[u][(q + 2) & 3] >> 8) & 0xff);
              g[u][4 * q + 3] = TE0R(w[u][(q + 3) & 3] & 0xff);
            }
#pragma unroll
          for (int u = 0; u < L; u++)
#pragma unroll
            for (int q = 0; q < 4; q++)
              w[u][q] = g[u][4 * q] ^ ROR8(g[u][4 * q + 1]) ^
                        ROR16(g[u][4 * q + 2]) ^ ROR24(g[u][4 * q + 3]) ^
                        rk[4 * r + q];
        } else {
#pragma unroll
        for (int u = 0; u < L; u++) {
          uint32_t n0 = TE0R(w[u][0] >> 24) ^
                        ROR8(TE0R((w[u][1] >> 16) & 0xff)) ^
                        ROR16(TE0R((w[u][2] >> 8) & 0xff)) ^
                        ROR24(TE0R(w[u][3] & 0xff)) ^ rk[4 * r];
          uint32_t n1 = TE0R(w[u][1] >> 24) ^
                        ROR8(TE0R((w[u][2] >> 16) & 0xff)) ^
                        ROR16(TE0R((w[u][3] >> 8) & 0xff)) ^
                        ROR24(TE0R(w[u][0] & 0xff)) ^ rk[4 * r + 1];
          uint32_t n2 = TE0R(w[u][2] >> 24) ^
                        ROR8(TE0R((w[u][3] >> 16) & 0xff)) ^
                        ROR16(TE0R((w[u][0] >> 8) & 0xff)) ^
                        ROR24(TE0R(w[u][1] & 0xff)) ^ rk[4 * r + 2];
          uint32_t n3 = TE0R(w[u][3] >> 24) ^
                        ROR8(TE0R((w[u][0] >> 16) & 0xff)) ^
                        ROR16(TE0R((w[u][1] >> 8) & 0xff)) ^
                        ROR24(TE0R(w[u][2] & 0xff)) ^ rk[4 * r + 3];
          w[u][0] = n0; w[u][1] = n1; w[u][2] = n2; w[u][3] = n3;
        }
        }
      }
      /* final round: S[x] = (Te0[x]>>8)&0xff; assemble each output word
       * from four replicated-table gathers (still conflict-free) */
#define SB24(idx) ((TE0R(idx) & 0xff00u) << 16)
#define SB16(idx) ((TE0R(idx) & 0xff00u) << 8)
#define SB08(idx) (TE0R(idx) & 0xff00u)
#define SB00(idx) ((TE0R(idx) >> 8) & 0xffu)
#pragma unroll
      for (int u = 0; u < L; u++) {
        if (u >= live) break;
        uint32_t f0 = (SB24(w[u][0] >> 24) | SB16((w[u][1] >> 16) & 0xff) |
                       SB08((w[u][2] >> 8) & 0xff) | SB00(w[u][3] & 0xff)) ^
                      rk[40];
        uint32_t f1 = (SB24(w[u][1] >> 24) | SB16((w[u][2] >> 16) & 0xff) |
                       SB08((w[u][3] >> 8) & 0xff) | SB00(w[u][0] & 0xff)) ^
                      rk[41];
        uint32_t f2 = (SB24(w[u][2] >> 24) | SB16((w[u][3] >> 16) & 0xff) |
                       SB08((w[u][0] >> 8) & 0xff) | SB00(w[u][1] & 0xff)) ^
                      rk[42];
        uint32_t f3 = (SB24(w[u][3] >> 24) | SB16((w[u][0] >> 16) & 0xff) |
                       SB08((w[u][1] >> 8) & 0xff) | SB00(w[u][2] & 0xff)) ^
                      rk[43];
        unsigned long long v0 =
            __builtin_bswap64(((unsigned long long)f0 << 32) | f1);
        unsigned long long v1 =
            __builtin_bswap64(((unsigned long long)f2 << 32) | f3);
        if (v0 < a.difficulty) {
          unsigned int s = atomicAdd(a.hit_count, 1u);
          if (s < a.hit_cap) {
            a.hits[s].index = a.index_base + ts[u];
            a.hits[s].nonce = c * POSTE_NONCES_PER_AES;
          }
        }
        if (v1 < a.difficulty) {
          unsigned int s = atomicAdd(a.hit_count, 1u);
          if (s < a.hit_cap) {
            a.hits[s].index = a.index_base + ts[u];
            a.hits[s].nonce = c * POSTE_NONCES_PER_AES + 1;
          }
        }
      }
    }
  }
#undef TE0R
#undef SB24
#undef SB16
#undef SB08
#undef SB00
}

/* 4-table interleaved bank-replicated scan: all four Te tables, each
 * bank-replicated, interleaved so entry (x, t, lane) sits at word
 * x*128 + t*32 + lane%32 — per-gather addressing is one v_lshl_add with
 * the t*128-byte offset folded into the ds_read immediate, and the
 * alignbit rotations of the 1-table variant disappear (~20% of its
 * VALU).  Costs 128 KiB LDS, so one 1024-thread workgroup per CU
 * (16 waves) instead of 20. */
__global__ void __launch_bounds__(1024)
post_scan_tt4_kernel(ScanKernelArgs a) {
  extern __shared__ uint32_t sTe[]; /* 32768 words = 4 tables x 32 copies */
  for (uint32_t i = threadIdx.x; i < 32768; i += blockDim.x)
    sTe[i] = a.te[((i >> 5) & 3) * 256 + (i >> 7)];
  __syncthreads();
  const uint32_t *t0p = sTe + (threadIdx.x & 31);
#define TT4(t, idx) t0p[(((uint32_t)(idx)) << 7) + 32 * (t)]

  const unsigned long long stride =
      (unsigned long long)gridDim.x * blockDim.x;
  /* ILP 4 measured best for this variant (r2d sweep); at one workgroup
   * per CU there are only 16 waves to hide latency, so deeper chains pay */
  constexpr int L = POSTE_SCAN_ILP > 4 ? POSTE_SCAN_ILP : 4;
  const unsigned long long span = stride * L;
  for (unsigned long long t0 =
           (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
       t0 < a.count; t0 += span) {
    uint32_t p[L][4];
    unsigned long long ts[L];
#pragma unroll
    for (int u = 0; u < L; u++) {
      unsigned long long t = t0 + (unsigned long long)u * stride;
      ts[u] = t < a.count ? t : t0;
      uint4 lraw = a.labels[ts[u]];
      p[u][0] = __builtin_bswap32(lraw.x);
      p[u][1] = __builtin_bswap32(lraw.y);
      p[u][2] = __builtin_bswap32(lraw.z);
      p[u][3] = __builtin_bswap32(lraw.w);
    }
    const int live = (int)((a.count - t0 + stride - 1) / stride) < L
                         ? (int)((a.count - t0 + stride - 1) / stride)
                         : L;
    for (uint32_t c = 0; c < a.n_ciphers; c++) {
      uint32_t rk[44];
#pragma unroll
      for (int k = 0; k < 44; k++) rk[k] = a.rk[c * 44 + k];
      uint32_t w[L][4];
#pragma unroll
      for (int u = 0; u < L; u++)
#pragma unroll
        for (int k = 0; k < 4; k++) w[u][k] = p[u][k] ^ rk[k];
#pragma unroll
      for (int r = 1; r < 10; r++) {
#pragma unroll
        for (int u = 0; u < L; u++) {
          uint32_t n0 = TT4(0, w[u][0] >> 24) ^
                        TT4(1, (w[u][1] >> 16) & 0xff) ^
                        TT4(2, (w[u][2] >> 8) & 0xff) ^
                        TT4(3, w[u][3] & 0xff) ^ rk[4 * r];
          uint32_t n1 = TT4(0, w[u][1] >> 24) ^
                        TT4(1, (w[u][2] >> 16) & 0xff) ^
                        TT4(2, (w[u][3] >> 8) & 0xff) ^
                        TT4(3, w[u][0] & 0xff) ^ rk[4 * r + 1];
          uint32_t n2 = TT4(0, w[u][2] >> 24) ^
                        TT4(1, (w[u][3] >> 16) & 0xff) ^
                        TT4(2, (w[u][0] >> 8) & 0xff) ^
                        TT4(3, w[u][1] & 0xff) ^ rk[4 * r + 2];
          uint32_t n3 = TT4(0, w[u][3] >> 24) ^
                        TT4(1, (w[u][0] >> 16) & 0xff) ^
                        TT4(2, (w[u][1] >> 8) & 0xff) ^
                        TT4(3, w[u][2] & 0xff) ^ rk[4 * r + 3];
          w[u][0] = n0; w[u][1] = n1; w[u][2] = n2; w[u][3] = n3;
        }
      }
      /* final round: S[x] = (Te0[x]>>8)&0xff, t=0 plane */
#define SB24_4(idx) ((TT4(0, idx) & 0xff00u) << 16)
#define SB16_4(idx) ((TT4(0, idx) & 0xff00u) << 8)
#define SB08_4(idx) (TT4(0, idx) & 0xff00u)
#define SB00_4(idx) ((TT4(0, idx) >> 8) & 0xffu)
#pragma unroll
      for (int u = 0; u < L; u++) {
        if (u >= live) break;
        uint32_t f0 = (SB24_4(w[u][0] >> 24) |
                       SB16_4((w[u][1] >> 16) & 0xff) |
                       SB08_4((w[u][2] >> 8) & 0xff) |
                       SB00_4(w[u][3] & 0xff)) ^ rk[40];
        uint32_t f1 = (SB24_4(w[u][1] >> 24) |
                       SB16_4((w[u][2] >> 16) & 0xff) |
                       SB08_4((w[u][3] >> 8) & 0xff) |
                       SB00_4(w[u][0] & 0xff)) ^ rk[41];
        uint32_t f2 = (SB24_4(w[u][2] >> 24) |
                       SB16_4((w[u][3] >> 16) & 0xff) |
                       SB08_4((w[u][0] >> 8) & 0xff) |
                       SB00_4(w[u][1] & 0xff)) ^ rk[42];
        uint32_t f3 = (SB24_4(w[u][3] >> 24) |
                       SB16_4((w[u][0] >> 16) & 0xff) |
                       SB08_4((w[u][1] >> 8) & 0xff) |
                       SB00_4(w[u][2] & 0xff)) ^ rk[43];
        unsigned long long v0 =
            __builtin_bswap64(((unsigned long long)f0 << 32) | f1);
        unsigned long long v1 =
            __builtin_bswap64(((unsigned long long)f2 << 32) | f3);
        if (v0 < a.difficulty) {
          unsigned int s = atomicAdd(a.hit_count, 1u);
          if (s < a.hit_cap) {
            a.hits[s].index = a.index_base + ts[u];
            a.hits[s].nonce = c * POSTE_NONCES_PER_AES;
          }
        }
        if (v1 < a.difficulty) {
          unsigned int s = atomicAdd(a.hit_count, 1u);
          if (s < a.hit_cap) {
            a.hits[s].index = a.index_base + ts[u];
            a.hits[s].nonce = c * POSTE_NONCES_PER_AES + 1;
          }
        }
      }
    }
  }
#undef TT4
#undef SB24_4
#undef SB16_4
#undef SB08_4
#undef SB00_4
}

/* verification's final predicate: AES-encrypt each recomputed label with
 * its proof's cipher and compare the nonce-half u64 against the proof's
 * difficulty (the last step of verifying.ProofVerifier.Verify,
 * post_verifier.go:159 -> per-index AES threshold check). */
__global__ void __launch_bounds__(POSTE_THREADS)
post_verify_pred_kernel(VerifyPredArgs a) {
  extern __shared__ uint32_t lds[];
  uint32_t *sTe = lds;
  uint8_t *sSbox = (uint8_t *)(sTe + 1024);
  for (uint32_t i = threadIdx.x; i < 1024; i += blockDim.x) sTe[i] = a.te[i];
  for (uint32_t i = threadIdx.x; i < 256; i += blockDim.x)
    sSbox[i] = a.sbox[i];
  __syncthreads();
  const unsigned long long stride =
      (unsigned long long)gridDim.x * blockDim.x;
  for (unsigned long long t =
           (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
       t < a.count; t += stride) {
    const uint32_t pidx = a.task_proof[t];
    const uint32_t *rk = a.rk + (unsigned long long)pidx * 44;
    uint4 lraw = a.labels2[2 * t]; /* first 16 B of the full label */
    uint32_t w0 = __builtin_bswap32(lraw.x) ^ rk[0];
    uint32_t w1 = __builtin_bswap32(lraw.y) ^ rk[1];
    uint32_t w2 = __builtin_bswap32(lraw.z) ^ rk[2];
    uint32_t w3 = __builtin_bswap32(lraw.w) ^ rk[3];
#pragma unroll
    for (int r = 1; r < 10; r++) {
      uint32_t n0 = sTe[w0 >> 24] ^ sTe[256 + ((w1 >> 16) & 0xff)] ^
                    sTe[512 + ((w2 >> 8) & 0xff)] ^ sTe[768 + (w3 & 0xff)] ^
                    rk[4 * r];
      uint32_t n1 = sTe[w1 >> 24] ^ sTe[256 + ((w2 >> 16) & 0xff)] ^
                    sTe[512 + ((w3 >> 8) & 0xff)] ^ sTe[768 + (w0 & 0xff)] ^
                    rk[4 * r + 1];
      uint32_t n2 = sTe[w2 >> 24] ^ sTe[256 + ((w3 >> 16) & 0xff)] ^
                    sTe[512 + ((w0 >> 8) & 0xff)] ^ sTe[768 + (w1 & 0xff)] ^
                    rk[4 * r + 2];
      uint32_t n3 = sTe[w3 >> 24] ^ sTe[256 + ((w0 >> 16) & 0xff)] ^
                    sTe[512 + ((w1 >> 8) & 0xff)] ^ sTe[768 + (w2 & 0xff)] ^
                    rk[4 * r + 3];
      w0 = n0; w1 = n1; w2 = n2; w3 = n3;
    }
    uint32_t f0 = (((uint32_t)sSbox[w0 >> 24] << 24) |
                   ((uint32_t)sSbox[(w1 >> 16) & 0xff] << 16) |
                   ((uint32_t)sSbox[(w2 >> 8) & 0xff] << 8) |
                   sSbox[w3 & 0xff]) ^ rk[40];
    uint32_t f1 = (((uint32_t)sSbox[w1 >> 24] << 24) |
                   ((uint32_t)sSbox[(w2 >> 16) & 0xff] << 16) |
                   ((uint32_t)sSbox[(w3 >> 8) & 0xff] << 8) |
                   sSbox[w0 & 0xff]) ^ rk[41];
    uint32_t f2 = (((uint32_t)sSbox[w2 >> 24] << 24) |
                   ((uint32_t)sSbox[(w3 >> 16) & 0xff] << 16) |
                   ((uint32_t)sSbox[(w0 >> 8) & 0xff] << 8) |
                   sSbox[w1 & 0xff]) ^ rk[42];
    uint32_t f3 = (((uint32_t)sSbox[w3 >> 24] << 24) |
                   ((uint32_t)sSbox[(w0 >> 16) & 0xff] << 16) |
                   ((uint32_t)sSbox[(w1 >> 8) & 0xff] << 8) |
                   sSbox[w2 & 0xff]) ^ rk[43];
    unsigned long long v =
        a.half[pidx] == 0
            ? __builtin_bswap64(((unsigned long long)f0 << 32) | f1)
            : __builtin_bswap64(((unsigned long long)f2 << 32) | f3);
    a.pass[t] = v < a.difficulty[pidx] ? 1 : 0;
  }
}

/* host-visible launchers (called from engine.cpp) */
extern "C" {

hipError_t poste_launch_label_kernel(const LabelKernelArgs *args,
                                     uint32_t blocks, hipStream_t stream) {
  /* Dual-stream ROMix (two slots per quad, half the grid) measured
   * neutral-to-slightly-slower than single-stream at 8 waves/SIMD
   * (gpurun summary6); kept behind POST_ROMIX2=1 for re-evaluation. */
  static const bool use2 = [] {
    const char *e = getenv("POST_ROMIX2");
    return e && e[0] == '1';
  }();
  const bool dual = use2 && args->gap_shift == 0;
  uint32_t gblocks = dual ? (blocks + 1) / 2 : blocks;
  hipLaunchKernelGGL(post_label_prologue_kernel, dim3(gblocks),
                     dim3(POSTE_THREADS), 0, stream, *args);
  if (dual) {
    hipLaunchKernelGGL(post_label_romix2_kernel, dim3(gblocks),
                       dim3(POSTE_THREADS), 0, stream, *args);
  } else {
    hipLaunchKernelGGL(post_label_romix_kernel, dim3(blocks),
                       dim3(POSTE_THREADS), 0, stream, *args);
  }
  hipLaunchKernelGGL(post_label_tail_kernel, dim3(gblocks),
                     dim3(POSTE_THREADS), 0, stream, *args);
  return hipGetLastError();
}

/* Max simultaneously-resident label SLOTS (in-flight labels) of the ROMix
 * kernel for the given gap config.  Scratch beyond this is wasted memory:
 * extra workgroups just queue. */
hipError_t poste_launch_verify_pred_kernel(const VerifyPredArgs *args,
                                           uint32_t blocks,
                                           hipStream_t stream) {
  size_t lds = 1024 * 4 + 256;
  hipLaunchKernelGGL(post_verify_pred_kernel, dim3(blocks),
                     dim3(POSTE_THREADS), lds, stream, *args);
  return hipGetLastError();
}

uint64_t poste_label_resident_slots(uint32_t gap_shift) {
  const void *kern =
      gap_shift == 0
          ? reinterpret_cast<const void *>(post_label_romix2_kernel)
          : reinterpret_cast<const void *>(post_label_romix_kernel);
  int blocks_per_cu = 0;
  if (hipOccupancyMaxActiveBlocksPerMultiprocessor(&blocks_per_cu, kern,
                                                   POSTE_THREADS, 0) !=
          hipSuccess ||
      blocks_per_cu <= 0)
    blocks_per_cu = 2;
  hipDeviceProp_t prop;
  int dev = 0;
  (void)hipGetDevice(&dev);
  int cus = 256;
  if (hipGetDeviceProperties(&prop, dev) == hipSuccess)
    cus = prop.multiProcessorCount;
  uint64_t quads = (uint64_t)blocks_per_cu * cus * (POSTE_THREADS / 4);
  const char *e = getenv("POST_ROMIX2");
  const bool dual = e && e[0] == '1' && gap_shift == 0;
  return dual ? quads * 2 : quads;
}

hipError_t poste_launch_scan_kernel(const ScanKernelArgs *args,
                                    uint32_t blocks, hipStream_t stream) {
  /* POST_SCAN_MODE: shared = round-1 shared-T-table kernel; tt4 =
   * 4-table/128KiB replicated kernel; bankrep512 / bankrepbg /
   * bankrep512bg = bank-replicated at 512-thread workgroups and/or
   * batched-gather scheduling; default = bank-replicated Te0 at 256.
   * Read per launch (launches are ms-scale) so tests can A/B kernels
   * within one process. */
  const char *e = getenv("POST_SCAN_MODE");
  const char *m = e ? e : "tt4"; /* fastest measured (r2d sweep); falls
                                    back to bankrep without 128 KiB LDS */
  if (strcmp(m, "shared") == 0) {
    size_t lds = 1024 * 4 + 256;
    hipLaunchKernelGGL(post_scan_kernel, dim3(blocks), dim3(POSTE_THREADS),
                       lds, stream, *args);
    return hipGetLastError();
  }
  if (strcmp(m, "tt4") == 0) {
    static bool attr_ok = [] {
      return hipFuncSetAttribute(
                 (const void *)post_scan_tt4_kernel,
                 hipFuncAttributeMaxDynamicSharedMemorySize,
                 32768 * 4) == hipSuccess;
    }();
    if (attr_ok) {
      uint32_t b4 = (uint32_t)((args->count + 1023) / 1024);
      if (b4 > 2048) b4 = 2048;
      if (b4 == 0) b4 = 1;
      hipLaunchKernelGGL(post_scan_tt4_kernel, dim3(b4), dim3(1024),
                         32768 * 4, stream, *args);
      return hipGetLastError();
    }
    m = "bankrep"; /* 128 KiB dynamic LDS unavailable */
  }
  const bool bg = strstr(m, "bg") != nullptr;
  const bool wide = strstr(m, "512") != nullptr;
  size_t lds = 8192 * 4; /* 32 bank-strided copies of Te0 */
  if (wide) {
    uint32_t b5 = (uint32_t)((args->count + 511) / 512);
    if (b5 > 4096) b5 = 4096;
    if (b5 == 0) b5 = 1;
    if (bg)
      hipLaunchKernelGGL((post_scan_bankrep_kernel<512, true>), dim3(b5),
                         dim3(512), lds, stream, *args);
    else
      hipLaunchKernelGGL((post_scan_bankrep_kernel<512, false>), dim3(b5),
                         dim3(512), lds, stream, *args);
  } else if (bg) {
    hipLaunchKernelGGL((post_scan_bankrep_kernel<POSTE_THREADS, true>),
                       dim3(blocks), dim3(POSTE_THREADS), lds, stream,
                       *args);
  } else {
    hipLaunchKernelGGL((post_scan_bankrep_kernel<POSTE_THREADS, false>),
                       dim3(blocks), dim3(POSTE_THREADS), lds, stream,
                       *args);
  }
  return hipGetLastError();
}
}
