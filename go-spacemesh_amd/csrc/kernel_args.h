/* kernel_args.h — argument blocks shared between engine.cpp (host) and
 * kernels.hip (device).  Kept POD; pointers are device pointers. */
#ifndef POST_KERNEL_ARGS_H
#define POST_KERNEL_ARGS_H

#include <hip/hip_runtime.h>

#include "post_common.h"

struct LabelKernelArgs {
  uint32_t commitment_le[8];
  uint64_t start;
  uint64_t count;
  uint32_t scrypt_n;
  uint32_t gap_shift; /* lookup-gap = 1<<gap_shift: store every gap-th ROMix
                         block, recompute the rest on read (SURVEY §7 step 3
                         HBM-traffic/VALU trade; scratch = N>>gap_shift
                         blocks per lane) */
  uint32_t out_full;
  uint32_t *scratch;
  uint64_t scratch_lanes;
  uint8_t *out;
  const uint64_t *indices;
  const uint32_t *commit_ids;
  const uint32_t *commitments;
  uint32_t *xbuf; /* staging for the per-label working block X between the
                     prologue/romix/tail kernels: 8 uint4 per task, quad-z
                     chunk layout (task*8 + sub, task*8 + sub + 4) */
  uint32_t has_difficulty;
  uint32_t difficulty_be[8];
  PostVrfCandidate *cand;
  unsigned int *cand_count;
  uint32_t cand_cap;
};

struct ScanKernelArgs {
  const uint4 *labels;
  uint64_t count;
  uint64_t index_base;
  const uint32_t *te;
  const uint8_t *sbox;
  const uint32_t *rk;
  uint32_t n_ciphers;
  uint64_t difficulty;
  PostScanHit *hits;
  unsigned int *hit_count;
  uint32_t hit_cap;
};

struct VerifyPredArgs {
  const uint4 *labels2;      /* 2 x uint4 per task (full 32-B labels) */
  uint64_t count;
  const uint32_t *te;        /* 1024 words */
  const uint8_t *sbox;       /* 256 bytes */
  const uint32_t *rk;        /* 44 BE words per PROOF */
  const uint32_t *task_proof;/* proof index per task */
  const uint8_t *half;       /* per-proof nonce half (0/1) */
  const uint64_t *difficulty;/* per-proof u64 threshold */
  uint8_t *pass;             /* per-task verdict out */
};

extern "C" {
hipError_t poste_launch_label_kernel(const LabelKernelArgs *args,
                                     uint32_t blocks, hipStream_t stream);
hipError_t poste_launch_verify_pred_kernel(const VerifyPredArgs *args,
                                           uint32_t blocks,
                                           hipStream_t stream);
hipError_t poste_launch_scan_kernel(const ScanKernelArgs *args,
                                    uint32_t blocks, hipStream_t stream);
uint64_t poste_label_resident_slots(uint32_t gap_shift);
}

#endif
