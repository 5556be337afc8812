/* post_common.h — internal constants and derivation layouts of the engine.
 *
 * The derivation layouts here are the protocol surface shared with the CPU
 * oracle (oracle/oracle.h documents the same constants with pinning notes);
 * they restate post-rs v0.7.13 semantics.  Any change here
 * must be mirrored in oracle/ and breaks parity fixtures.
 */
#ifndef POST_COMMON_H
#define POST_COMMON_H

#include <stdint.h>

#define POSTE_LABEL_SIZE 16
#define POSTE_FULL_LABEL_SIZE 32
#define POSTE_NONCES_PER_AES 2 /* 2 nonces per AES cipher */
#define POSTE_NONCE_GROUP 16   /* k2pow granularity in nonces */
#define POSTE_K2POW_PREFIX "k2pow" /* 5 bytes, blake3-mode k2pow domain tag */
#define POSTE_VRF_MARGIN 16        /* vrf threshold = 16*2^256/num_labels */

/* scrypt fixed params on this path (activation/post.go:155): r=1, p=1. */
#define POSTE_SCRYPT_R 1
#define POSTE_SCRYPT_P 1

/* device-side structures */
typedef struct {
  unsigned long long index;
  uint32_t label_be[8]; /* full label as 8 big-endian-ordered words */
} PostVrfCandidate;

typedef struct {
  unsigned long long index;
  uint32_t nonce;
  uint32_t pad;
} PostScanHit;

#endif /* POST_COMMON_H */
