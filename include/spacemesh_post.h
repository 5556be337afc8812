/* spacemesh_post.h — C-ABI of the MI355X-native POST engine (libpost_hip.so).
 *
 * This is the drop-in boundary of the rebuild: it plays the
 * role of post-rs's post.h (fetched prebuilt by the reference,
 * Makefile-libs.Inc:49,60-68, consumed via cgo by spacemeshos/post v0.12.9),
 * exporting exactly the entry points the reference's Go side needs:
 *
 *   - provider enumeration/benchmark  -> initialization.OpenCLProviders /
 *     CPUProviderID / Benchmark (activation/post_supervisor.go:105-127,
 *     api/grpcserver/smesher_service.go:248-275)
 *   - incremental init with cancel, progress and resume ->
 *     initialization.NewInitializer(...).Initialize(ctx) /
 *     NumLabelsWritten (activation/post.go:261,267-271,295,355-361)
 *   - prove(challenge) -> {nonce u32, indices bytes, pow u64} ->
 *     the post-service GenProof protocol (api/grpcserver/post_client.go:69-143)
 *   - verify(proof, metadata, opts) -> verifying.ProofVerifier.Verify
 *     (activation/post_verifier.go:150-160, validation.go:182-222), incl.
 *     K3 subset (validation.go:206-209) and selected-index
 *     (activation/malfeasance.go:161-169)
 *   - verify_vrf_nonce -> verifying.VerifyVRFNonce (validation.go:277)
 *
 * The Go binding a maintainer would write against this header (cgo shim
 * satisfying PostSetupProvider / PostVerifier / PostClient) is shown in
 * INTEGRATION.md.  No GPU/torch types cross this boundary: plain pointers,
 * sizes and status codes only.
 *
 * Threading contract: post_verify / post_verify_vrf_nonce are safe to call
 * concurrently from N workers (the reference's pool, post_verifier.go:227).
 * An init session is single-threaded per session object (post.go:276-281);
 * progress/cancel accessors may be called from other threads.
 */
#ifndef SPACEMESH_POST_H
#define SPACEMESH_POST_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---------- status codes ---------- */
#define POST_OK 0
#define POST_ERR 1              /* generic failure (see post_last_error) */
#define POST_ERR_INVALID_ARGS 2
#define POST_ERR_NO_GPU 3       /* no HIP device / kernels unavailable */
#define POST_ERR_OOM 4
#define POST_ERR_IO 5
#define POST_ERR_CANCELLED 6    /* maps to context.Canceled (post.go:297) */
#define POST_ERR_POW 7          /* invalid proof-of-work */
#define POST_ERR_INVALID_INDEX 8 /* maps to verifying.ErrInvalidIndex */
#define POST_ERR_UNSUPPORTED 9  /* e.g. RandomX pow mode (unpinned) */
#define POST_ERR_NO_NONCE 10    /* prove: no nonce reached K2 */

/* thread-local description of the last error in this thread */
const char *post_last_error(void);

/* ---------- constants ---------- */
#define POST_LABEL_SIZE 16       /* bytes per label in postdata_*.bin */
#define POST_FULL_LABEL_SIZE 32  /* bytes compared for the VRF nonce */
#define POST_MAX_INDICES_BYTES 800 /* wire cap, activation/wire/wire_v1.go:43 */

/* k2pow modes (PostRandomXMode / PowFlags, activation/post_types.go:84-144).
 * RandomX is the reference's mode; it is parity-unpinned in this build
 * (SURVEY.md §7 hard part 1) and returns POST_ERR_UNSUPPORTED. */
#define POST_POW_MODE_RANDOMX 0
#define POST_POW_MODE_BLAKE3 1

/* ---------- provider enumeration ---------- */
typedef struct {
  uint32_t id;
  char model[256];       /* device name, surfaced verbatim over gRPC
                            (api/grpcserver/smesher_service.go:270-275) */
  uint32_t device_type;  /* 0 = GPU (HIP); no CPU provider in this engine */
  uint64_t memory_bytes; /* HBM capacity */
  uint64_t performance;  /* labels/sec estimate; filled by post_benchmark */
} PostProvider;

/* Fills providers[0..cap) and sets *count. POST_OK even when count==0. */
int post_providers(PostProvider *providers, uint32_t cap, uint32_t *count);

/* Measures labels/sec for a provider at the given scrypt N
 * (initialization.Benchmark, activation/post_supervisor.go:120-127). */
int post_benchmark(uint32_t provider_id, uint32_t scrypt_n,
                   uint64_t *labels_per_sec);

/* ---------- configuration ---------- */
typedef struct {
  uint8_t node_id[32];
  uint8_t commitment_atx_id[32];
  uint32_t num_units;
  uint64_t labels_per_unit;   /* mainnet 2^32, config/mainnet.go:186 */
  uint64_t max_file_size;     /* postdata file split, activation/post.go:56 */
  uint32_t scrypt_n;          /* r=1, p=1 fixed (dep default, post.go:155) */
  uint32_t provider_id;
  /* Index-range shard of this session: [index_start, index_end) of the
   * global label space; 0,0 means the whole space.  This is the reference's
   * file/range parallelism (activation/post.go:56-60,166-182) and the
   * 8-GPU sharding axis. */
  uint64_t index_start;
  uint64_t index_end;
  /* Directory for postdata_*.bin + postdata_metadata.json; NULL for
   * in-memory sink mode (benchmark, BASELINE config 2). */
  const char *data_dir;
  /* scratch budget for the ROMix tables, in bytes; 0 = auto (most of free) */
  uint64_t scratch_bytes;
} PostInitConfig;

typedef struct PostInitSession PostInitSession;

/* Create a session. Scans data_dir for existing postdata_*.bin and resumes
 * after the last complete label; the persisted metadata Nonce/NonceValue
 * (re-written whenever the running VRF minimum improves) seeds the
 * session's minimum, so sharded sessions keep their shard-local minimum
 * across kill+resume (StartSession resume semantics,
 * activation/post.go:267-271). */
int post_init_new(const PostInitConfig *cfg, PostInitSession **out);

/* Run initialization to completion (or cancel). Blocking; call from the
 * session's own thread exactly like init.Initialize(ctx) (post.go:295).
 * Returns POST_OK, POST_ERR_CANCELLED, or an error. */
int post_init_run(PostInitSession *s);

/* Process up to max_labels further labels of the session's range (one or
 * more kernel launches), blocking until they are complete.  *done receives
 * the number processed (0 at end of range).  post_init_run is a loop over
 * this.  Exposed for benchmarking (a bench "step"). */
int post_init_step(PostInitSession *s, uint64_t max_labels, uint64_t *done);

/* Kernel-only time of the last post_init_step in milliseconds, measured
 * with HIP events on the session stream (for roofline accounting). */
double post_init_last_kernel_ms(const PostInitSession *s);

/* Progress counter: labels written so far within this session's range
 * (init.NumLabelsWritten, activation/post.go:261). Thread-safe. */
uint64_t post_init_num_labels_written(const PostInitSession *s);

/* Request cancellation; post_init_run returns POST_ERR_CANCELLED.
 * Thread-safe. */
void post_init_cancel(PostInitSession *s);

/* Best VRF nonce found so far: smallest full 32-byte label (big-endian
 * lexicographic) and its index (common/types/activation.go:311-313).
 * Returns POST_OK when one exists, POST_ERR otherwise. */
int post_init_nonce(const PostInitSession *s, uint64_t *index,
                    uint8_t label[32]);

/* In sink mode: borrow the device-resident labels of the last batch /
 * the whole range when kept (for the prove path and parity tests).
 * out_host must hold 16*(count) bytes; copies labels [first,first+count)
 * of the session range to host. */
int post_init_copy_labels(const PostInitSession *s, uint64_t first,
                          uint64_t count, uint8_t *out_host);

void post_init_free(PostInitSession *s);

/* ---------- proving ---------- */
typedef struct {
  uint32_t nonce;
  uint64_t pow;
  uint8_t indices[POST_MAX_INDICES_BYTES];
  uint32_t indices_len;
  uint16_t num_indices;
} PostProof;

typedef struct {
  uint8_t challenge[32];
  uint32_t k1;
  uint32_t k2;
  uint32_t nonces;           /* mainnet 288, config/mainnet.go:61 */
  uint8_t pow_difficulty[32];/* config/mainnet.go:41 */
  uint32_t pow_mode;         /* POST_POW_MODE_* */
  uint32_t pow_threads;      /* host threads for k2pow; 0 = all */
  uint32_t provider_id;
} PostProveConfig;

/* Generate a proof over labels in data_dir (postdata_*.bin +
 * postdata_metadata.json), the post-service GenProof role
 * (api/grpcserver/post_client.go:69-143). */
int post_prove(const char *data_dir, const PostProveConfig *cfg,
               PostProof *out);

/* Same, over a host buffer of 16-byte labels for index range
 * [0, num_labels) (bench/tests). */
int post_prove_buffer(const uint8_t *labels, uint64_t num_labels,
                      const uint8_t node_id[32],
                      const uint8_t commitment_atx_id[32],
                      const PostProveConfig *cfg, PostProof *out);

/* Multi-GPU proving shard: scan labels [index_base,
 * index_base + count) of a space of total_labels labels and return the
 * raw passing (index, nonce) pairs; a coordinator merges shards and packs
 * the winning nonce's indices (go-spacemesh_amd/proving.py).  group_pows
 * must hold nonces/16 k2pow values for the challenge (computed once,
 * shared by all shards). */
typedef struct {
  uint64_t index;
  uint32_t nonce;
  uint32_t pad;
} PostScanHitOut;
int post_prove_scan(const uint8_t *labels, uint64_t count,
                    uint64_t index_base, uint64_t total_labels,
                    const PostProveConfig *cfg, const uint64_t *group_pows,
                    PostScanHitOut *hits, uint32_t cap, uint32_t *n_hits);

/* k2pow helpers so a coordinator can compute/verify group pows once */
int post_k2pow_search(const uint8_t challenge[32], uint32_t nonce_group,
                      const uint8_t pow_difficulty[32], uint32_t pow_mode,
                      uint32_t threads, uint64_t *out);

/* ---------- verification ---------- */
typedef struct {
  /* shared.ProofMetadata as assembled at activation/validation.go:193-199 */
  uint8_t node_id[32];
  uint8_t commitment_atx_id[32];
  uint8_t challenge[32];
  uint32_t num_units;
  uint64_t labels_per_unit;
} PostProofMetadata;

typedef struct {
  uint32_t k1;
  uint32_t k2;
  uint32_t k3;              /* k3 >= k2 -> full verify (validation.go:176) */
  const uint8_t *subset_seed; /* NULL -> full; verifying.Subset seed */
  size_t subset_seed_len;
  int32_t selected_index;   /* >=0 -> verifying.SelectedIndex
                               (malfeasance.go:165); else -1 */
  uint8_t pow_difficulty[32];
  uint32_t pow_mode;
  uint32_t scrypt_n;
  uint32_t provider_id;
} PostVerifyConfig;

/* Verify one proof. POST_OK; POST_ERR_INVALID_INDEX (+ *invalid_index =
 * failing position, for the InvalidPostIndex malfeasance proof,
 * handler_v1.go:228-247); POST_ERR_POW; POST_ERR_INVALID_ARGS.
 * Safe for concurrent callers. */
int post_verify(const PostProof *proof, const PostProofMetadata *meta,
                const PostVerifyConfig *cfg, uint32_t *invalid_index);

/* Batched verification (the verifier-pool steady state, BASELINE config 5):
 * n proofs with per-proof metadata; statuses[i] and invalid_indices[i]
 * receive per-proof results.  One kernel launch recomputes all sampled
 * labels and one applies the AES threshold predicate. */
int post_verify_batch(const PostProof *proofs, const PostProofMetadata *metas,
                      uint32_t n, const PostVerifyConfig *cfg,
                      int *statuses, uint32_t *invalid_indices);

/* Same with PER-PROOF subset seeds (each gossip verification samples with
 * its own peer-derived seed, validation.go:206-209): subset_seeds holds n
 * fixed-length seeds back to back (NULL -> cfg->subset_seed for all). */
int post_verify_batch_seeded(const PostProof *proofs,
                             const PostProofMetadata *metas, uint32_t n,
                             const PostVerifyConfig *cfg,
                             const uint8_t *subset_seeds, size_t seed_len,
                             int *statuses, uint32_t *invalid_indices);

/* verifying.VerifyVRFNonce (validation.go:261-286). */
int post_verify_vrf_nonce(const PostProofMetadata *meta, uint64_t index,
                          uint32_t scrypt_n, uint32_t provider_id);

/* ---------- introspection / self-test ---------- */
/* Engine's independent blake3 (for cross-checking vs the oracle's in
 * tests; not a product entry point). */
void post_selftest_blake3(const uint8_t *msg, size_t len, uint8_t out[32]);
/* Engine's host AES-128 single block (same purpose). */
void post_selftest_aes128(const uint8_t key[16], const uint8_t in[16],
                          uint8_t out[16]);
/* Engine's host reference label (used only by tests to cross-check the
 * device path; computed with the engine's own host scrypt). */
int post_selftest_label(const uint8_t node_id[32],
                        const uint8_t commitment_atx_id[32], uint64_t index,
                        uint32_t scrypt_n, uint8_t out[32]);

/* version / build info string */
const char *post_engine_version(void);

#ifdef __cplusplus
}
#endif
#endif /* SPACEMESH_POST_H */
