/* probe_romix.hip — standalone microbenchmarks isolating the labeling
 * kernel's cost components on gfx950 (not part of the product library).
 * Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 probe_romix.hip -o probe
 * Variants:
 *   salsa  — register-only salsa20/8 chain (VALU issue/latency ceiling)
 *   write  — phase-1 shaped: blockmix + streaming 128-B stores
 *   read   — phase-2 shaped: random 128-B gathers + blockmix
 *   full   — both phases (the real ROMix shape)
 */
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>

#define THREADS 256

__device__ __forceinline__ void salsa8(uint32_t b[16]) {
  uint32_t x0 = b[0], x1 = b[1], x2 = b[2], x3 = b[3], x4 = b[4], x5 = b[5],
           x6 = b[6], x7 = b[7], x8 = b[8], x9 = b[9], x10 = b[10],
           x11 = b[11], x12 = b[12], x13 = b[13], x14 = b[14], x15 = b[15];
#define QR(a, bq, c, d)                                                        \
  x##bq ^= __builtin_rotateleft32(x##a + x##d, 7);                             \
  x##c ^= __builtin_rotateleft32(x##bq + x##a, 9);                             \
  x##d ^= __builtin_rotateleft32(x##c + x##bq, 13);                            \
  x##a ^= __builtin_rotateleft32(x##d + x##c, 18);
#pragma unroll
  for (int i = 0; i < 4; i++) {
    QR(0, 4, 8, 12) QR(5, 9, 13, 1) QR(10, 14, 2, 6) QR(15, 3, 7, 11)
    QR(0, 1, 2, 3) QR(5, 6, 7, 4) QR(10, 11, 8, 9) QR(15, 12, 13, 14)
  }
  b[0] += x0; b[1] += x1; b[2] += x2; b[3] += x3;
  b[4] += x4; b[5] += x5; b[6] += x6; b[7] += x7;
  b[8] += x8; b[9] += x9; b[10] += x10; b[11] += x11;
  b[12] += x12; b[13] += x13; b[14] += x14; b[15] += x15;
}

__device__ __forceinline__ void blockmix_r1(uint32_t X[32]) {
  uint32_t T[16];
#pragma unroll
  for (int k = 0; k < 16; k++) T[k] = X[k] ^ X[16 + k];
  salsa8(T);
#pragma unroll
  for (int k = 0; k < 16; k++) {
    uint32_t y0 = T[k];
    T[k] = y0 ^ X[16 + k];
    X[k] = y0;
  }
  salsa8(T);
#pragma unroll
  for (int k = 0; k < 16; k++) X[16 + k] = T[k];
}

__global__ void __launch_bounds__(THREADS) k_salsa(uint32_t n, uint32_t seed,
                                                   uint32_t *sink) {
  uint32_t X[32];
#pragma unroll
  for (int k = 0; k < 32; k++)
    X[k] = seed + k + (blockIdx.x * blockDim.x + threadIdx.x);
  for (uint32_t i = 0; i < 2 * n; i++) blockmix_r1(X);
  if (X[0] == 0xdeadbeef) sink[0] = X[1];
}

__global__ void __launch_bounds__(THREADS) k_write(uint32_t n, uint32_t seed,
                                                   uint4 *V, uint32_t *sink) {
  const unsigned long long lane =
      (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned long long lanes =
      (unsigned long long)gridDim.x * blockDim.x;
  uint32_t X[32];
#pragma unroll
  for (int k = 0; k < 32; k++) X[k] = seed + k + (uint32_t)lane;
  unsigned long long base = lane * 8ull;
  const unsigned long long stride = lanes * 8ull;
  for (uint32_t j = 0; j < n; j++) {
    uint4 *p = V + base;
#pragma unroll
    for (int c = 0; c < 8; c++)
      p[c] = make_uint4(X[4 * c], X[4 * c + 1], X[4 * c + 2], X[4 * c + 3]);
    blockmix_r1(X);
    base += stride;
  }
  if (X[0] == 0xdeadbeef) sink[0] = X[1];
}

__global__ void __launch_bounds__(THREADS) k_read(uint32_t n, uint32_t seed,
                                                  const uint4 *V,
                                                  uint32_t *sink) {
  const unsigned long long lane =
      (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned long long lanes =
      (unsigned long long)gridDim.x * blockDim.x;
  uint32_t X[32];
#pragma unroll
  for (int k = 0; k < 32; k++) X[k] = seed + k + (uint32_t)lane;
  const uint32_t mask = n - 1;
  for (uint32_t i = 0; i < n; i++) {
    uint32_t j = X[16] & mask;
    const uint4 *p = V + ((unsigned long long)j * lanes + lane) * 8ull;
#pragma unroll
    for (int c = 0; c < 8; c++) {
      uint4 v = p[c];
      X[4 * c] ^= v.x;
      X[4 * c + 1] ^= v.y;
      X[4 * c + 2] ^= v.z;
      X[4 * c + 3] ^= v.w;
    }
    blockmix_r1(X);
  }
  if (X[0] == 0xdeadbeef) sink[0] = X[1];
}

__global__ void __launch_bounds__(THREADS) k_full(uint32_t n, uint32_t seed,
                                                  uint4 *V, uint32_t *sink) {
  const unsigned long long lane =
      (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned long long lanes =
      (unsigned long long)gridDim.x * blockDim.x;
  uint32_t X[32];
#pragma unroll
  for (int k = 0; k < 32; k++) X[k] = seed + k + (uint32_t)lane;
  {
    unsigned long long base = lane * 8ull;
    const unsigned long long stride = lanes * 8ull;
    for (uint32_t j = 0; j < n; j++) {
      uint4 *p = V + base;
#pragma unroll
      for (int c = 0; c < 8; c++)
        p[c] = make_uint4(X[4 * c], X[4 * c + 1], X[4 * c + 2], X[4 * c + 3]);
      blockmix_r1(X);
      base += stride;
    }
  }
  const uint32_t mask = n - 1;
  for (uint32_t i = 0; i < n; i++) {
    uint32_t j = X[16] & mask;
    const uint4 *p = V + ((unsigned long long)j * lanes + lane) * 8ull;
#pragma unroll
    for (int c = 0; c < 8; c++) {
      uint4 v = p[c];
      X[4 * c] ^= v.x;
      X[4 * c + 1] ^= v.y;
      X[4 * c + 2] ^= v.z;
      X[4 * c + 3] ^= v.w;
    }
    blockmix_r1(X);
  }
  if (X[0] == 0xdeadbeef) sink[0] = X[1];
}

/* gather-pattern probes: dependent random 128-B block reads, 1 lane per
 * block (8 x 16-B scattered requests per wave instr) vs 4 cooperating lanes
 * (aligned 64-B transactions).  Hypothesis test for the 4-lane kernel. */
__global__ void __launch_bounds__(THREADS) k_gather1(uint32_t n,
                                                     uint32_t iters,
                                                     const uint4 *V,
                                                     uint32_t *sink) {
  const unsigned long long lane =
      (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned long long lanes =
      (unsigned long long)gridDim.x * blockDim.x;
  const uint32_t mask = n - 1;
  uint32_t acc = (uint32_t)lane * 2654435761u;
  for (uint32_t i = 0; i < iters; i++) {
    uint32_t j = acc & mask;
    const uint4 *p = V + ((unsigned long long)j * lanes + lane) * 8ull;
    uint4 a = p[0], b = p[1], c = p[2], d = p[3];
    uint4 e = p[4], f = p[5], g = p[6], h = p[7];
    acc ^= a.x ^ a.y ^ a.z ^ a.w ^ b.x ^ b.y ^ b.z ^ b.w;
    acc ^= c.x ^ c.y ^ c.z ^ c.w ^ d.x ^ d.y ^ d.z ^ d.w;
    acc ^= e.x ^ e.y ^ e.z ^ e.w ^ f.x ^ f.y ^ f.z ^ f.w;
    acc ^= g.x ^ g.y ^ g.z ^ g.w ^ h.x ^ h.y ^ h.z ^ h.w;
    acc = acc * 1664525u + 1013904223u;
  }
  if (acc == 0xdeadbeef) sink[0] = acc;
}

__global__ void __launch_bounds__(THREADS) k_gather4(uint32_t n,
                                                     uint32_t iters,
                                                     const uint4 *V,
                                                     uint32_t *sink) {
  /* 4 lanes share one 128-B block: group g reads block j_g; lane l of the
   * group reads chunks l and l+4 -> each wave instruction is 16 aligned
   * 64-B transactions instead of 64 scattered 16-B ones. */
  const unsigned long long lane =
      (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned long long lanes =
      (unsigned long long)gridDim.x * blockDim.x;
  const unsigned long long group = lane >> 2;
  const uint32_t sub = (uint32_t)(lane & 3);
  const unsigned long long groups = lanes >> 2;
  const uint32_t mask = n - 1;
  uint32_t acc = (uint32_t)group * 2654435761u;
  for (uint32_t i = 0; i < iters; i++) {
    uint32_t j = acc & mask;
    /* block of group g stored as 8 consecutive uint4 at (j*groups + g) */
    const uint4 *p = V + ((unsigned long long)j * groups + group) * 8ull;
    uint4 a = p[sub], b = p[sub + 4];
    uint32_t x = a.x ^ a.y ^ a.z ^ a.w ^ b.x ^ b.y ^ b.z ^ b.w;
    /* xor-reduce across the 4 lanes (quad swizzle) so acc stays uniform */
    x ^= __builtin_amdgcn_ds_swizzle(x, 0x80B1); /* quad perm 1,0,3,2 */
    x ^= __builtin_amdgcn_ds_swizzle(x, 0x804E); /* quad perm 2,3,0,1 */
    acc ^= x;
    acc = acc * 1664525u + 1013904223u;
  }
  if (acc == 0xdeadbeef) sink[0] = acc;
}

__global__ void __launch_bounds__(THREADS) k_gather8(uint32_t n,
                                                     uint32_t iters,
                                                     const uint4 *V,
                                                     uint32_t *sink) {
  /* 8 lanes share one 128-B block, one 16-B uint4 per lane at consecutive
   * addresses -> each wave instruction is 8 aligned 128-B transactions.
   * Round-2 hypothesis test: do 128-B requests beat the 64-B pattern
   * (7.7 TB/s measured) enough to justify an oct-cooperative labeling
   * kernel? */
  const unsigned long long lane =
      (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned long long lanes =
      (unsigned long long)gridDim.x * blockDim.x;
  const unsigned long long group = lane >> 3;
  const uint32_t sub = (uint32_t)(lane & 7);
  const unsigned long long groups = lanes >> 3;
  const uint32_t mask = n - 1;
  uint32_t acc = (uint32_t)group * 2654435761u;
  for (uint32_t i = 0; i < iters; i++) {
    uint32_t j = acc & mask;
    /* block of group g stored as 8 consecutive uint4 at (j*groups + g) */
    const uint4 *p = V + ((unsigned long long)j * groups + group) * 8ull;
    uint4 a = p[sub];
    uint32_t x = a.x ^ a.y ^ a.z ^ a.w;
    /* xor-reduce across the 8 lanes so acc stays group-uniform */
    x ^= __shfl_xor(x, 1);
    x ^= __shfl_xor(x, 2);
    x ^= __shfl_xor(x, 4);
    acc ^= x;
    acc = acc * 1664525u + 1013904223u;
  }
  if (acc == 0xdeadbeef) sink[0] = acc;
}

int main(int argc, char **argv) {
  uint32_t n = argc > 1 ? (uint32_t)atoi(argv[1]) : 8192;
  uint32_t blocks = argc > 2 ? (uint32_t)atoi(argv[2]) : 768;
  uint64_t lanes = (uint64_t)blocks * THREADS;
  uint4 *V = nullptr;
  uint32_t *sink = nullptr;
  size_t vbytes = (size_t)lanes * n * 128;
  if (hipMalloc(&V, vbytes) != hipSuccess) {
    printf("alloc failed (%zu bytes)\n", vbytes);
    return 1;
  }
  (void)hipMalloc(&sink, 4);
  (void)hipMemset(V, 0x5a, vbytes);

  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);

  int occ_salsa = 0, occ_full = 0;
  (void)hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &occ_salsa, reinterpret_cast<const void *>(k_salsa), THREADS, 0);
  (void)hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &occ_full, reinterpret_cast<const void *>(k_full), THREADS, 0);
  printf("n=%u blocks=%u lanes=%llu scratch=%.1f GiB occ(salsa)=%d "
         "occ(full)=%d blk/CU\n",
         n, blocks, (unsigned long long)lanes, vbytes / 1073741824.0,
         occ_salsa, occ_full);

  struct Case { const char *name; int which; double bytes_per_iter; };
  Case cases[] = {{"salsa(2n blockmix, no mem)", 0, 0.0},
                  {"write(phase1)", 1, 128.0},
                  {"read(phase2 random)", 2, 128.0},
                  {"full(romix)", 3, 256.0},
                  {"gather1(16B reqs)", 4, 128.0},
                  {"gather4(64B reqs)", 5, 32.0},
                  {"gather8(128B reqs)", 6, 16.0}};
  for (auto &c : cases) {
    /* warmup + timed */
    for (int rep = 0; rep < 2; rep++) {
      (void)hipEventRecord(e0, nullptr);
      switch (c.which) {
      case 0:
        hipLaunchKernelGGL(k_salsa, dim3(blocks), dim3(THREADS), 0, 0, n,
                           rep + 1, sink);
        break;
      case 1:
        hipLaunchKernelGGL(k_write, dim3(blocks), dim3(THREADS), 0, 0, n,
                           rep + 1, V, sink);
        break;
      case 2:
        hipLaunchKernelGGL(k_read, dim3(blocks), dim3(THREADS), 0, 0, n,
                           rep + 1, V, sink);
        break;
      case 3:
        hipLaunchKernelGGL(k_full, dim3(blocks), dim3(THREADS), 0, 0, n,
                           rep + 1, V, sink);
        break;
      case 4:
        hipLaunchKernelGGL(k_gather1, dim3(blocks), dim3(THREADS), 0, 0, n,
                           n, V, sink);
        break;
      case 5:
        hipLaunchKernelGGL(k_gather4, dim3(blocks), dim3(THREADS), 0, 0, n,
                           4 * n, V, sink);
        break;
      case 6:
        hipLaunchKernelGGL(k_gather8, dim3(blocks), dim3(THREADS), 0, 0, n,
                           8 * n, V, sink);
        break;
      }
      (void)hipEventRecord(e1, nullptr);
      if (hipEventSynchronize(e1) != hipSuccess) {
        printf("%s: kernel failed\n", c.name);
        return 1;
      }
      if (rep == 0) continue;
      float ms = 0;
      (void)hipEventElapsedTime(&ms, e0, e1);
      double iters = (double)lanes * n *
                     (c.which == 0 || c.which == 3
                          ? 2.0
                          : (c.which == 5 ? 4.0
                                          : (c.which == 6 ? 8.0 : 1.0)));
      double gops = iters * 2.0 * 416 / (ms / 1e3) / 1e9;
      double gbs = iters * c.bytes_per_iter / (ms / 1e3) / 1e9;
      double cyc_per_iter = (ms / 1e3) * 2.4e9 / ((double)n *
                            (c.which == 0 ? 2.0 : (c.which == 3 ? 2.0 : 1.0)));
      printf("%-28s %8.1f ms  %8.1f Gop/s(u32)  %8.1f GB/s  "
             "%8.0f cyc/lane-iter\n",
             c.name, ms, gops, gbs, cyc_per_iter);
    }
  }
  (void)hipFree(V);
  (void)hipFree(sink);
  return 0;
}

