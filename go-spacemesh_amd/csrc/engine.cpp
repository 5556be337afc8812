/* engine.cpp — C-ABI implementation of the MI355X POST engine
 * (include/spacemesh_post.h).  Host orchestration around the HIP kernels:
 * sessions, streams, file IO, resume, nonce tracking, proving pipeline,
 * batched verification.  No CPU compute fallback: every compute entry
 * requires a HIP device and fails loudly without one (POST_ERR_NO_GPU).
 *
 * Reference lifecycle being mirrored (file:line under /root/reference/):
 *   activation/post.go:261,267-331   init progress/resume/cancel/state
 *   activation/post.go:299-312       reference-label self-check
 *   api/grpcserver/post_client.go:69-143  prove protocol results
 *   activation/post_verifier.go:150-160 + validation.go:182-222  verify
 */
#include "../../include/spacemesh_post.h"

#include <hip/hip_runtime.h>

#include <sys/stat.h>
#include <sys/types.h>

#include <algorithm>
#include <array>
#include <atomic>
#include <thread>
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <functional>
#include <map>
#include <mutex>
#include <string>
#include <vector>

#include "crypto_host.h"
#include "kernel_args.h"
#include "post_common.h"

namespace {

thread_local std::string g_last_error;

void set_error(const std::string &e) { g_last_error = e; }

int hip_fail(const char *what, hipError_t err) {
  set_error(std::string(what) + ": " + hipGetErrorString(err));
  return err == hipErrorOutOfMemory ? POST_ERR_OOM : POST_ERR;
}

#define HIP_TRY(expr)                                                          \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) return hip_fail(#expr, _e);                          \
  } while (0)

int require_gpu(uint32_t provider_id) {
  int count = 0;
  hipError_t e = hipGetDeviceCount(&count);
  if (e != hipSuccess || count == 0) {
    set_error("no HIP device available (the HIP engine has no CPU fallback)");
    return POST_ERR_NO_GPU;
  }
  if (provider_id >= (uint32_t)count) {
    set_error("provider id out of range");
    return POST_ERR_INVALID_ARGS;
  }
  e = hipSetDevice((int)provider_id);
  if (e != hipSuccess) return hip_fail("hipSetDevice", e);
  return POST_OK;
}

void load_commitment_words(const uint8_t c[32], uint32_t w[8]) {
  std::memcpy(w, c, 32); /* little-endian host/device byte stream words */
}

void difficulty_to_be_words(const uint8_t d[32], uint32_t w[8]) {
  for (int i = 0; i < 8; i++)
    w[i] = ((uint32_t)d[4 * i] << 24) | ((uint32_t)d[4 * i + 1] << 16) |
           ((uint32_t)d[4 * i + 2] << 8) | d[4 * i + 3];
}

constexpr uint32_t THREADS = 256;
constexpr uint32_t CAND_CAP = 1 << 16;

/* lookup-gap (kernel_args.h).  Measured on MI355X (gpurun summary3-5
 * logs): with the quad-cooperative kernels the scratch fits at gap 1 and
 * extra recompute only costs, so the default is gap 1 (shift 0).
 * POST_GAP_SHIFT overrides for scratch-constrained configs. */
uint32_t pick_gap_shift(uint32_t scrypt_n) {
  const char *env = getenv("POST_GAP_SHIFT");
  if (env) {
    uint32_t g = (uint32_t)atoi(env);
    while ((scrypt_n >> g) == 0) g--;
    return g;
  }
  return 0;
}

/* Cached per-device verification workspace: the worker-pool call pattern
 * (one proof per post_verify call, post_verifier.go:150-160) would
 * otherwise pay scratch/staging allocation on every call.  Calls remain
 * safe for concurrent workers; they serialize on the device section. */
struct VerifyWorkspace {
  std::mutex mu;
  uint32_t *d_scratch = nullptr;
  size_t scratch_cap = 0; /* bytes */
  uint64_t *d_idx = nullptr;
  uint32_t *d_cid = nullptr;
  uint8_t *d_out = nullptr;
  uint32_t *d_xbuf = nullptr;
  size_t tasks_cap = 0;
  uint32_t *d_cm = nullptr;
  size_t cm_cap = 0; /* words */
  uint32_t *d_vrk = nullptr; /* 44 words per proof */
  uint8_t *d_half = nullptr;
  uint64_t *d_vdiff = nullptr;
  size_t proofs_cap = 0;
  uint8_t *d_pass = nullptr; /* per task, with tasks_cap */
};
std::mutex g_vws_mu;
std::map<int, VerifyWorkspace *> g_vws;

VerifyWorkspace *get_verify_ws(int dev) {
  std::lock_guard<std::mutex> lk(g_vws_mu);
  auto it = g_vws.find(dev);
  if (it != g_vws.end()) return it->second;
  auto *w = new VerifyWorkspace();
  g_vws[dev] = w;
  return w;
}

int ws_reserve_proofs(VerifyWorkspace *w, size_t proofs) {
  if (proofs > w->proofs_cap) {
    size_t cap = std::max<size_t>(proofs, 1024);
    if (w->d_vrk) (void)hipFree(w->d_vrk);
    if (w->d_half) (void)hipFree(w->d_half);
    if (w->d_vdiff) (void)hipFree(w->d_vdiff);
    w->d_vrk = nullptr; w->d_half = nullptr; w->d_vdiff = nullptr;
    w->proofs_cap = 0;
    HIP_TRY(hipMalloc(&w->d_vrk, cap * 44 * 4));
    HIP_TRY(hipMalloc(&w->d_half, cap));
    HIP_TRY(hipMalloc(&w->d_vdiff, cap * 8));
    w->proofs_cap = cap;
  }
  return POST_OK;
}

int ws_reserve(VerifyWorkspace *w, size_t scratch_bytes, size_t tasks,
               size_t cm_words) {
  if (scratch_bytes > w->scratch_cap) {
    if (w->d_scratch) (void)hipFree(w->d_scratch);
    w->d_scratch = nullptr;
    w->scratch_cap = 0;
    HIP_TRY(hipMalloc(&w->d_scratch, scratch_bytes));
    w->scratch_cap = scratch_bytes;
  }
  if (tasks > w->tasks_cap) {
    size_t cap = std::max<size_t>(tasks, 4096);
    if (w->d_idx) (void)hipFree(w->d_idx);
    if (w->d_cid) (void)hipFree(w->d_cid);
    if (w->d_out) (void)hipFree(w->d_out);
    if (w->d_xbuf) (void)hipFree(w->d_xbuf);
    if (w->d_pass) (void)hipFree(w->d_pass);
    w->d_idx = nullptr; w->d_cid = nullptr; w->d_out = nullptr;
    w->d_xbuf = nullptr; w->d_pass = nullptr;
    w->tasks_cap = 0;
    HIP_TRY(hipMalloc(&w->d_idx, cap * 8));
    HIP_TRY(hipMalloc(&w->d_cid, cap * 4));
    HIP_TRY(hipMalloc(&w->d_out, cap * 32));
    HIP_TRY(hipMalloc(&w->d_xbuf, cap * 128));
    HIP_TRY(hipMalloc(&w->d_pass, cap));
    w->tasks_cap = cap;
  }
  if (cm_words > w->cm_cap) {
    size_t cap = std::max<size_t>(cm_words, 1024);
    if (w->d_cm) (void)hipFree(w->d_cm);
    w->d_cm = nullptr;
    w->cm_cap = 0;
    HIP_TRY(hipMalloc(&w->d_cm, cap * 4));
    w->cm_cap = cap;
  }
  return POST_OK;
}

struct DeviceTables { /* AES tables resident per device */
  uint32_t *d_te = nullptr;
  uint8_t *d_sbox = nullptr;
};
std::mutex g_tables_mu;
std::map<int, DeviceTables> g_tables;

int get_aes_tables(int dev, DeviceTables &out) {
  std::lock_guard<std::mutex> lk(g_tables_mu);
  auto it = g_tables.find(dev);
  if (it != g_tables.end()) {
    out = it->second;
    return POST_OK;
  }
  static uint32_t te[1024];
  static uint8_t sbox[256];
  poste::aes128_tables(te, sbox);
  DeviceTables t;
  HIP_TRY(hipMalloc(&t.d_te, sizeof(te)));
  HIP_TRY(hipMalloc(&t.d_sbox, sizeof(sbox)));
  HIP_TRY(hipMemcpy(t.d_te, te, sizeof(te), hipMemcpyHostToDevice));
  HIP_TRY(hipMemcpy(t.d_sbox, sbox, sizeof(sbox), hipMemcpyHostToDevice));
  g_tables[dev] = t;
  out = t;
  return POST_OK;
}

} // namespace

extern "C" {

const char *post_last_error(void) { return g_last_error.c_str(); }

const char *post_engine_version(void) {
  return "spacemesh-post-hip 0.2 (gfx950)";
}

/* ------------------------- providers ------------------------- */

int post_providers(PostProvider *providers, uint32_t cap, uint32_t *count) {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess) n = 0;
  *count = (uint32_t)n;
  for (uint32_t i = 0; i < (uint32_t)n && i < cap; i++) {
    hipDeviceProp_t prop;
    if (hipGetDeviceProperties(&prop, (int)i) != hipSuccess) continue;
    providers[i].id = i;
    std::snprintf(providers[i].model, sizeof(providers[i].model), "%s",
                  prop.name);
    providers[i].device_type = 0;
    providers[i].memory_bytes = prop.totalGlobalMem;
    providers[i].performance = 0;
  }
  return POST_OK;
}

/* ------------------------- init session ------------------------- */

struct PostInitSession {
  PostInitConfig cfg;
  std::string data_dir;
  uint64_t range_start = 0, range_end = 0;
  std::atomic<uint64_t> written{0};
  std::atomic<bool> cancel{false};
  uint8_t commitment[32];

  std::mutex nonce_mu;
  bool nonce_found = false;
  uint64_t nonce_idx = 0;
  uint8_t nonce_label[32];

  uint64_t lanes = 0;       /* concurrent labels (scratch slots); one slot
                               = one quad of 4 threads in the kernel */
  uint64_t batch_seq = 0;   /* batches processed (self-check period) */
  uint32_t gap_shift = 0;
  uint64_t batch = 0;       /* labels per launch */
  uint32_t *d_scratch = nullptr;
  uint8_t *d_out = nullptr; /* batch output (16 B per label) */
  uint8_t *d_all = nullptr; /* whole-range output kept on device (small) */
  bool keep_all = false;
  PostVrfCandidate *d_cand = nullptr;
  unsigned int *d_cand_count = nullptr;
  uint32_t *d_xbuf = nullptr; /* 128 B per batch task (3-kernel staging) */
  uint8_t *h_batch = nullptr; /* pinned */
  hipStream_t stream = nullptr;
  hipEvent_t ev0 = nullptr, ev1 = nullptr;
  double last_kernel_ms = 0; /* accumulated over the last post_init_step */

  ~PostInitSession() {
    if (ev0) (void)hipEventDestroy(ev0);
    if (ev1) (void)hipEventDestroy(ev1);
    if (stream) (void)hipStreamDestroy(stream);
    if (d_scratch) (void)hipFree(d_scratch);
    if (d_out) (void)hipFree(d_out);
    if (d_all) (void)hipFree(d_all);
    if (d_cand) (void)hipFree(d_cand);
    if (d_cand_count) (void)hipFree(d_cand_count);
    if (d_xbuf) (void)hipFree(d_xbuf);
    if (h_batch) (void)hipHostFree(h_batch);
  }
};

static uint64_t existing_labels(const std::string &dir, uint64_t per_file,
                                uint64_t range_start, uint64_t range_total) {
  /* resume point = number of contiguous complete labels already on disk
   * (StartSession resume, activation/post.go:267-271) */
  uint64_t done = 0;
  for (uint64_t pos = range_start; pos < range_start + range_total;) {
    uint64_t file_i = pos / per_file;
    uint64_t in_file = pos % per_file;
    uint64_t file_cap = std::min(per_file - in_file,
                                 range_start + range_total - pos);
    char path[4096];
    std::snprintf(path, sizeof path, "%s/postdata_%llu.bin", dir.c_str(),
                  (unsigned long long)file_i);
    FILE *f = std::fopen(path, "rb");
    if (!f) break;
    std::fseek(f, 0, SEEK_END);
    long sz = std::ftell(f);
    std::fclose(f);
    uint64_t labels_here = (uint64_t)(sz < 0 ? 0 : sz) / POST_LABEL_SIZE;
    if (in_file > labels_here) break;
    uint64_t usable = std::min(labels_here - in_file, file_cap);
    done += usable;
    pos += usable;
    if (usable < file_cap) break;
  }
  return done;
}

static int check_existing_metadata(PostInitSession *s);
static int write_metadata(PostInitSession *s);

int post_init_new(const PostInitConfig *cfg, PostInitSession **out) {
  if (!cfg || !out) {
    set_error("null args");
    return POST_ERR_INVALID_ARGS;
  }
  if (cfg->scrypt_n < 2 || (cfg->scrypt_n & (cfg->scrypt_n - 1))) {
    set_error("scrypt N must be a power of two >= 2");
    return POST_ERR_INVALID_ARGS;
  }
  int rc = require_gpu(cfg->provider_id);
  if (rc != POST_OK) return rc;

  /* everything below must free the session on failure */
  struct Guard {
    PostInitSession *p;
    ~Guard() { delete p; }
  };
  auto *s = new PostInitSession();
  Guard guard{s};
  s->cfg = *cfg;
  if (cfg->data_dir) {
    s->data_dir = cfg->data_dir;
    /* create the data directory like the reference initializer does */
    std::string path;
    for (size_t i = 0; i <= s->data_dir.size(); i++) {
      if (i == s->data_dir.size() || s->data_dir[i] == '/') {
        path = s->data_dir.substr(0, i);
        if (!path.empty()) (void)::mkdir(path.c_str(), 0755);
      }
    }
    struct stat st;
    if (stat(s->data_dir.c_str(), &st) != 0 || !S_ISDIR(st.st_mode)) {
      set_error("cannot create data dir " + s->data_dir);
      return POST_ERR_IO;
    }
  }
  uint64_t total = (uint64_t)cfg->num_units * cfg->labels_per_unit;
  s->range_start = cfg->index_start;
  s->range_end = cfg->index_end ? cfg->index_end : total;
  if (s->range_end > total || s->range_start >= s->range_end) {
    set_error("bad index range");
    return POST_ERR_INVALID_ARGS;
  }
  poste::commitment(cfg->node_id, cfg->commitment_atx_id, s->commitment);

  /* scratch sizing: lanes * 128 * (N >> gap_shift) bytes */
  s->gap_shift = pick_gap_shift(cfg->scrypt_n);
  size_t free_b = 0, total_b = 0;
  HIP_TRY(hipMemGetInfo(&free_b, &total_b));
  uint64_t budget = cfg->scratch_bytes
                        ? cfg->scratch_bytes
                        : (uint64_t)((double)free_b * 0.80);
  uint64_t per_lane = ((uint64_t)cfg->scrypt_n >> s->gap_shift) * 128;
  uint64_t lanes = budget / per_lane;
  uint64_t range = s->range_end - s->range_start;
  lanes = std::min<uint64_t>(lanes, std::max<uint64_t>(range, 64));
  /* scratch beyond the kernel's resident capacity is wasted: extra
   * workgroups only queue (grid-stride covers the batch regardless) */
  lanes = std::min<uint64_t>(lanes,
                             poste_label_resident_slots(s->gap_shift));
  lanes = (lanes / 128) * 128; /* keep the dual-stream grid block-aligned */
  if (lanes == 0) lanes = 128;
  if (lanes == 0) {
    set_error("not enough device memory for one scratch lane block");
    return POST_ERR_OOM;
  }
  s->lanes = lanes;
  s->batch = std::min<uint64_t>(range, lanes * 4);

  HIP_TRY(hipMalloc(&s->d_scratch, (size_t)lanes * per_lane));
  uint64_t keep_bytes = range * POST_LABEL_SIZE;
  s->keep_all = s->data_dir.empty() && keep_bytes <= (8ull << 30);
  if (s->keep_all) {
    HIP_TRY(hipMalloc(&s->d_all, (size_t)keep_bytes));
  }
  HIP_TRY(hipMalloc(&s->d_out, (size_t)s->batch * POST_LABEL_SIZE));
  HIP_TRY(hipMalloc(&s->d_cand, sizeof(PostVrfCandidate) * CAND_CAP));
  HIP_TRY(hipMalloc(&s->d_xbuf, (size_t)s->batch * 128));
  HIP_TRY(hipMalloc(&s->d_cand_count, sizeof(unsigned int)));
  HIP_TRY(hipHostMalloc(&s->h_batch, (size_t)s->batch * POST_LABEL_SIZE));
  HIP_TRY(hipStreamCreate(&s->stream));

  if (!s->data_dir.empty()) {
    /* the dir is bound to one identity/config (verifyMetadata role) */
    rc = check_existing_metadata(s);
    if (rc != POST_OK) return rc;
    rc = write_metadata(s); /* persisted at session creation, like
                               initialization.SaveMetadata; rewritten with
                               the nonce at completion */
    if (rc != POST_OK) return rc;
    uint64_t per_file = std::max<uint64_t>(
        1, s->cfg.max_file_size / POST_LABEL_SIZE);
    s->written = existing_labels(s->data_dir, per_file, s->range_start, range);
  }
  guard.p = nullptr; /* ownership passes to the caller */
  *out = s;
  return POST_OK;
}

uint64_t post_init_num_labels_written(const PostInitSession *s) {
  return s ? s->written.load() : 0;
}

void post_init_cancel(PostInitSession *s) {
  if (s) s->cancel = true;
}

int post_init_nonce(const PostInitSession *s, uint64_t *index,
                    uint8_t label[32]) {
  auto *m = const_cast<PostInitSession *>(s);
  std::lock_guard<std::mutex> lk(m->nonce_mu);
  if (!s->nonce_found) return POST_ERR;
  *index = s->nonce_idx;
  std::memcpy(label, s->nonce_label, 32);
  return POST_OK;
}

static int write_batch_files(PostInitSession *s, uint64_t pos, uint64_t count,
                             const uint8_t *host_labels) {
  uint64_t per_file =
      std::max<uint64_t>(1, s->cfg.max_file_size / POST_LABEL_SIZE);
  uint64_t off = 0;
  while (off < count) {
    uint64_t gpos = pos + off;
    uint64_t file_i = gpos / per_file;
    uint64_t in_file = gpos % per_file;
    uint64_t take = std::min(per_file - in_file, count - off);
    char path[4096];
    std::snprintf(path, sizeof path, "%s/postdata_%llu.bin",
                  s->data_dir.c_str(), (unsigned long long)file_i);
    FILE *f = std::fopen(path, "r+b");
    if (!f) f = std::fopen(path, "w+b");
    if (!f) {
      set_error(std::string("cannot open ") + path);
      return POST_ERR_IO;
    }
    std::fseek(f, (long)(in_file * POST_LABEL_SIZE), SEEK_SET);
    size_t wr = std::fwrite(host_labels + off * POST_LABEL_SIZE,
                            POST_LABEL_SIZE, take, f);
    std::fclose(f);
    if (wr != take) {
      set_error("short write");
      return POST_ERR_IO;
    }
    off += take;
  }
  return POST_OK;
}

static void b64(const uint8_t *in, size_t n, std::string &out) {
  static const char t[] =
      "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789+/";
  for (size_t i = 0; i < n; i += 3) {
    uint32_t v = (uint32_t)in[i] << 16;
    if (i + 1 < n) v |= (uint32_t)in[i + 1] << 8;
    if (i + 2 < n) v |= in[i + 2];
    out += t[(v >> 18) & 63];
    out += t[(v >> 12) & 63];
    out += i + 1 < n ? t[(v >> 6) & 63] : '=';
    out += i + 2 < n ? t[v & 63] : '=';
  }
}

static bool b64_decode(const std::string &in, uint8_t *out, size_t want) {
  static int8_t rev[256];
  static bool init = false;
  if (!init) {
    static const char t[] =
        "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789+/";
    for (int i = 0; i < 256; i++) rev[i] = -1;
    for (int i = 0; i < 64; i++) rev[(uint8_t)t[i]] = (int8_t)i;
    init = true;
  }
  size_t n = 0;
  uint32_t acc = 0;
  int bits = 0;
  for (char c : in) {
    if (c == '=') break;
    int8_t v = rev[(uint8_t)c];
    if (v < 0) return false;
    acc = (acc << 6) | (uint32_t)v;
    bits += 6;
    if (bits >= 8) {
      bits -= 8;
      if (n >= want) return false;
      out[n++] = (uint8_t)(acc >> bits);
    }
  }
  return n == want;
}

/* Minimal reader for the fields write_metadata emits: enough to implement
 * the reference's metadata guard (initialization.LoadMetadata +
 * verifyMetadata semantics: a data dir is bound to one identity/config —
 * activation/post.go:373-435 reads the persisted commitment). */
static bool md_find_string(const std::string &j, const char *key,
                           std::string &out) {
  std::string pat = std::string("\"") + key + "\": \"";
  size_t p = j.find(pat);
  if (p == std::string::npos) return false;
  p += pat.size();
  size_t e = j.find('"', p);
  if (e == std::string::npos) return false;
  out = j.substr(p, e - p);
  return true;
}

static bool md_find_u64(const std::string &j, const char *key,
                        uint64_t &out) {
  std::string pat = std::string("\"") + key + "\": ";
  size_t p = j.find(pat);
  if (p == std::string::npos) return false;
  out = strtoull(j.c_str() + p + pat.size(), nullptr, 10);
  return true;
}

static int check_existing_metadata(PostInitSession *s) {
  char path[4096];
  std::snprintf(path, sizeof path, "%s/postdata_metadata.json",
                s->data_dir.c_str());
  FILE *f = std::fopen(path, "rb");
  if (!f) return POST_OK; /* fresh dir */
  std::string j;
  char buf[4096];
  size_t rd;
  while ((rd = std::fread(buf, 1, sizeof buf, f)) > 0) j.append(buf, rd);
  std::fclose(f);
  std::string nid_b64, atx_b64, want_nid, want_atx;
  b64(s->cfg.node_id, 32, want_nid);
  b64(s->cfg.commitment_atx_id, 32, want_atx);
  uint64_t lpu = 0, n = 0;
  if (md_find_string(j, "NodeId", nid_b64) && nid_b64 != want_nid) {
    set_error("data dir belongs to a different node id (postdata_metadata"
              ".json mismatch)");
    return POST_ERR_INVALID_ARGS;
  }
  if (md_find_string(j, "CommitmentAtxId", atx_b64) &&
      atx_b64 != want_atx) {
    set_error("data dir was initialized for a different commitment ATX");
    return POST_ERR_INVALID_ARGS;
  }
  if (md_find_u64(j, "LabelsPerUnit", lpu) &&
      lpu != s->cfg.labels_per_unit) {
    set_error("data dir was initialized with a different LabelsPerUnit");
    return POST_ERR_INVALID_ARGS;
  }
  if (md_find_u64(j, "N", n) && n != s->cfg.scrypt_n) {
    set_error("data dir was initialized with different scrypt params");
    return POST_ERR_INVALID_ARGS;
  }
  /* Resume must carry the persisted VRF nonce forward: the reference
   * initializer keeps metadata's Nonce/NonceValue across sessions, and a
   * sharded (index_start/end) session that loses its shard-local minimum
   * would corrupt the cross-GPU min-reduce after kill+resume.  Seed the
   * session's running minimum from the file before write_metadata (called
   * right after us) rewrites it. */
  uint64_t nonce_idx = 0;
  std::string nv_b64;
  if (md_find_u64(j, "Nonce", nonce_idx) &&
      md_find_string(j, "NonceValue", nv_b64)) {
    uint8_t label[32];
    if (!b64_decode(nv_b64, label, 32)) {
      set_error("postdata_metadata.json NonceValue is not 32 base64 bytes");
      return POST_ERR_INVALID_ARGS;
    }
    std::lock_guard<std::mutex> lk(s->nonce_mu);
    s->nonce_found = true;
    s->nonce_idx = nonce_idx;
    std::memcpy(s->nonce_label, label, 32);
  }
  return POST_OK;
}

static int write_metadata(PostInitSession *s) {
  /* postdata_metadata.json, field set per shared.PostMetadata usage
   * (activation/post_test.go:305-309; Go json base64 for byte slices) */
  std::string nid, atx, nv;
  b64(s->cfg.node_id, 32, nid);
  b64(s->cfg.commitment_atx_id, 32, atx);
  char path[4096];
  std::snprintf(path, sizeof path, "%s/postdata_metadata.json",
                s->data_dir.c_str());
  FILE *f = std::fopen(path, "w");
  if (!f) {
    set_error("cannot write metadata");
    return POST_ERR_IO;
  }
  std::fprintf(f,
               "{\n  \"NodeId\": \"%s\",\n  \"CommitmentAtxId\": \"%s\",\n"
               "  \"LabelsPerUnit\": %llu,\n  \"NumUnits\": %u,\n"
               "  \"MaxFileSize\": %llu,\n  \"Scrypt\": {\"N\": %u, \"R\": 1,"
               " \"P\": 1}",
               nid.c_str(), atx.c_str(),
               (unsigned long long)s->cfg.labels_per_unit, s->cfg.num_units,
               (unsigned long long)s->cfg.max_file_size, s->cfg.scrypt_n);
  std::lock_guard<std::mutex> lk(s->nonce_mu);
  if (s->nonce_found) {
    b64(s->nonce_label, 32, nv);
    std::fprintf(f, ",\n  \"Nonce\": %llu,\n  \"NonceValue\": \"%s\"",
                 (unsigned long long)s->nonce_idx, nv.c_str());
  }
  std::fprintf(f, "\n}\n");
  std::fclose(f);
  return POST_OK;
}

int post_init_step(PostInitSession *s, uint64_t max_labels, uint64_t *done) {
  if (!s || !done) return POST_ERR_INVALID_ARGS;
  *done = 0;
  int rc = require_gpu(s->cfg.provider_id);
  if (rc != POST_OK) return rc;

  LabelKernelArgs args;
  std::memset(&args, 0, sizeof(args));
  load_commitment_words(s->commitment, args.commitment_le);
  args.scrypt_n = s->cfg.scrypt_n;
  args.gap_shift = s->gap_shift;
  args.xbuf = s->d_xbuf;
  args.out_full = 0;
  args.scratch = s->d_scratch;
  args.scratch_lanes = s->lanes;
  args.indices = nullptr;
  args.commit_ids = nullptr;
  args.commitments = nullptr;
  args.has_difficulty = 1;
  args.cand = s->d_cand;
  args.cand_count = s->d_cand_count;
  args.cand_cap = CAND_CAP;

  uint8_t cur_difficulty[32];
  std::memset(cur_difficulty, 0xff, 32); /* track the global minimum */
  {
    std::lock_guard<std::mutex> lk(s->nonce_mu);
    if (s->nonce_found) std::memcpy(cur_difficulty, s->nonce_label, 32);
  }

  const uint32_t blocks = (uint32_t)(s->lanes / 64); /* 64 quads/block */
  std::vector<PostVrfCandidate> cands(CAND_CAP);

  if (!s->ev0) {
    HIP_TRY(hipEventCreate(&s->ev0));
    HIP_TRY(hipEventCreate(&s->ev1));
  }
  s->last_kernel_ms = 0;

  uint64_t range = s->range_end - s->range_start;
  uint64_t pos = s->written.load(); /* relative position (resume) */
  uint64_t target = std::min(range, pos + max_labels);
  while (pos < target) {
    if (s->cancel.load()) return POST_ERR_CANCELLED;
    uint64_t count = std::min(s->batch, target - pos);
    args.start = s->range_start + pos;
    args.count = count;
    args.out = s->keep_all ? s->d_all + pos * POST_LABEL_SIZE : s->d_out;
    difficulty_to_be_words(cur_difficulty, args.difficulty_be);
    unsigned int zero = 0;
    HIP_TRY(hipMemcpyAsync(s->d_cand_count, &zero, sizeof(zero),
                           hipMemcpyHostToDevice, s->stream));
    HIP_TRY(hipEventRecord(s->ev0, s->stream));
    HIP_TRY(poste_launch_label_kernel(&args, blocks, s->stream));
    HIP_TRY(hipEventRecord(s->ev1, s->stream));
    HIP_TRY(hipMemcpyAsync(s->h_batch, args.out, count * POST_LABEL_SIZE,
                           hipMemcpyDeviceToHost, s->stream));
    unsigned int n_cand = 0;
    HIP_TRY(hipMemcpyAsync(&n_cand, s->d_cand_count, sizeof(n_cand),
                           hipMemcpyDeviceToHost, s->stream));
    /* the reference label for this batch's self-check computes on the host
     * WHILE the kernel runs (a ~5 ms scrypt that would otherwise sit on
     * the critical path between launches).  POST_SELFCHECK_PERIOD=k checks
     * every k-th batch instead — with 8 ranks per node the per-rank host
     * scrypt otherwise contends for the same host cores (SCALE readiness,
     * VERDICT r01 §next-3); detection latency rises k batches, coverage
     * semantics (ErrReferenceLabelMismatch) are unchanged. */
    const char *sp_env = getenv("POST_SELFCHECK_PERIOD");
    long selfcheck_period = sp_env ? strtol(sp_env, nullptr, 10) : 1;
    if (selfcheck_period < 1) selfcheck_period = 1;
    const bool do_selfcheck = (s->batch_seq++ % (uint64_t)selfcheck_period)
                              == 0;
    const uint64_t probe = count / 2;
    uint8_t ref[32];
    const int ref_rc =
        do_selfcheck ? poste::host_label(s->commitment, args.start + probe,
                                         s->cfg.scrypt_n, ref)
                     : 0;
    HIP_TRY(hipStreamSynchronize(s->stream));
    bool nonce_improved = false;
    if (n_cand > 0) {
      unsigned int take = std::min(n_cand, CAND_CAP);
      HIP_TRY(hipMemcpy(cands.data(), s->d_cand,
                        sizeof(PostVrfCandidate) * take,
                        hipMemcpyDeviceToHost));
      std::lock_guard<std::mutex> lk(s->nonce_mu);
      for (unsigned int i = 0; i < take; i++) {
        uint8_t lab[32];
        for (int k = 0; k < 8; k++) {
          uint32_t w = cands[i].label_be[k];
          lab[4 * k] = (uint8_t)(w >> 24);
          lab[4 * k + 1] = (uint8_t)(w >> 16);
          lab[4 * k + 2] = (uint8_t)(w >> 8);
          lab[4 * k + 3] = (uint8_t)w;
        }
        int c = std::memcmp(lab, cur_difficulty, 32);
        if (!s->nonce_found || c < 0 ||
            (c == 0 && cands[i].index < s->nonce_idx)) {
          s->nonce_found = true;
          s->nonce_idx = cands[i].index;
          std::memcpy(s->nonce_label, lab, 32);
          std::memcpy(cur_difficulty, lab, 32);
          nonce_improved = true;
        }
      }
    }
    /* persist an improved running minimum immediately: metadata is the
     * only carrier of the shard-local minimum across kill+resume (the
     * creation-time write has no nonce yet, and a completed-at-exit-only
     * write loses it on any interrupt — the cfg4 soak caught exactly
     * that).  Improvements are rare (~log of batches), so this is a few
     * small rewrites per session. */
    if (nonce_improved && !s->data_dir.empty()) {
      rc = write_metadata(s);
      if (rc != POST_OK) return rc;
    }

    /* reference-label self-check (ErrReferenceLabelMismatch semantics,
     * activation/post.go:299-312): compare the host label computed above */
    if (ref_rc != 0) {
      set_error("host reference label failed");
      return POST_ERR;
    }
    if (do_selfcheck &&
        std::memcmp(ref, s->h_batch + probe * POST_LABEL_SIZE,
                    POST_LABEL_SIZE) != 0) {
      set_error("reference label mismatch: device labels diverge from the "
                "host reference (ErrReferenceLabelMismatch)");
      return POST_ERR;
    }

    if (!s->data_dir.empty()) {
      rc = write_batch_files(s, s->range_start + pos, count, s->h_batch);
      if (rc != POST_OK) return rc;
    }
    float ms = 0;
    (void)hipEventElapsedTime(&ms, s->ev0, s->ev1);
    s->last_kernel_ms += ms;
    pos += count;
    s->written.store(pos);
    *done += count;
  }
  return POST_OK;
}

double post_init_last_kernel_ms(const PostInitSession *s) {
  return s ? s->last_kernel_ms : 0.0;
}

/* nonce-only batch beyond the configured range (no label output) */
static int nonce_only_batch(PostInitSession *s, uint64_t gstart,
                            uint64_t count) {
  LabelKernelArgs args;
  std::memset(&args, 0, sizeof(args));
  load_commitment_words(s->commitment, args.commitment_le);
  args.scrypt_n = s->cfg.scrypt_n;
  args.gap_shift = s->gap_shift;
  args.xbuf = s->d_xbuf;
  args.scratch = s->d_scratch;
  args.scratch_lanes = s->lanes;
  args.start = gstart;
  args.count = count;
  args.out = nullptr;
  args.has_difficulty = 1;
  args.cand = s->d_cand;
  args.cand_count = s->d_cand_count;
  args.cand_cap = CAND_CAP;
  uint8_t cur[32];
  std::memset(cur, 0xff, 32);
  {
    std::lock_guard<std::mutex> lk(s->nonce_mu);
    if (s->nonce_found) std::memcpy(cur, s->nonce_label, 32);
  }
  difficulty_to_be_words(cur, args.difficulty_be);
  unsigned int zero = 0;
  HIP_TRY(hipMemcpy(s->d_cand_count, &zero, sizeof(zero),
                    hipMemcpyHostToDevice));
  HIP_TRY(poste_launch_label_kernel(&args, (uint32_t)(s->lanes / 64),
                                    s->stream));
  HIP_TRY(hipStreamSynchronize(s->stream));
  unsigned int n_cand = 0;
  HIP_TRY(hipMemcpy(&n_cand, s->d_cand_count, sizeof(n_cand),
                    hipMemcpyDeviceToHost));
  if (n_cand > 0) {
    unsigned int take = std::min(n_cand, CAND_CAP);
    std::vector<PostVrfCandidate> cands(take);
    HIP_TRY(hipMemcpy(cands.data(), s->d_cand,
                      sizeof(PostVrfCandidate) * take,
                      hipMemcpyDeviceToHost));
    bool improved = false;
    {
      std::lock_guard<std::mutex> lk(s->nonce_mu);
      for (unsigned int i = 0; i < take; i++) {
        uint8_t lab[32];
        for (int k = 0; k < 8; k++) {
          uint32_t w = cands[i].label_be[k];
          lab[4 * k] = (uint8_t)(w >> 24);
          lab[4 * k + 1] = (uint8_t)(w >> 16);
          lab[4 * k + 2] = (uint8_t)(w >> 8);
          lab[4 * k + 3] = (uint8_t)w;
        }
        int c = s->nonce_found ? std::memcmp(lab, s->nonce_label, 32) : -1;
        if (c < 0 || (c == 0 && cands[i].index < s->nonce_idx)) {
          s->nonce_found = true;
          s->nonce_idx = cands[i].index;
          std::memcpy(s->nonce_label, lab, 32);
          improved = true;
        }
      }
    }
    if (improved && !s->data_dir.empty()) {
      int rc = write_metadata(s);
      if (rc != POST_OK) return rc;
    }
  }
  return POST_OK;
}

int post_init_run(PostInitSession *s) {
  if (!s) return POST_ERR_INVALID_ARGS;
  uint64_t range = s->range_end - s->range_start;
  while (s->written.load() < range) {
    uint64_t done = 0;
    int rc = post_init_step(s, s->batch * 8, &done);
    if (rc != POST_OK) return rc;
    if (done == 0) break;
  }

  /* When this session covers the whole label space and the tracked minimum
   * does not meet the VRF threshold (p ~ e^-16 with the x16 margin), keep
   * searching past the end — the reference initializer's follow-up nonce
   * search (init completes with Nonce optional; VerifyVRFNonce needs one
   * below threshold, validation.go:261-286). */
  uint64_t total = (uint64_t)s->cfg.num_units * s->cfg.labels_per_unit;
  if (s->range_start == 0 && s->range_end == total) {
    uint8_t diff[32];
    poste::vrf_difficulty(total, diff);
    for (uint64_t ext = total; ext < 2 * total; ext += s->batch) {
      {
        std::lock_guard<std::mutex> lk(s->nonce_mu);
        if (s->nonce_found &&
            std::memcmp(s->nonce_label, diff, 32) < 0)
          break;
      }
      if (s->cancel.load()) return POST_ERR_CANCELLED;
      int rc = nonce_only_batch(s, ext, std::min(s->batch, 2 * total - ext));
      if (rc != POST_OK) return rc;
    }
  }

  if (!s->data_dir.empty()) return write_metadata(s);
  return POST_OK;
}

int post_init_copy_labels(const PostInitSession *s, uint64_t first,
                          uint64_t count, uint8_t *out_host) {
  if (!s || !out_host) return POST_ERR_INVALID_ARGS;
  if (!s->keep_all) {
    set_error("labels were not kept on device (file mode or large range)");
    return POST_ERR_INVALID_ARGS;
  }
  if (first + count > s->range_end - s->range_start) {
    set_error("range out of bounds");
    return POST_ERR_INVALID_ARGS;
  }
  HIP_TRY(hipMemcpy(out_host, s->d_all + first * POST_LABEL_SIZE,
                    count * POST_LABEL_SIZE, hipMemcpyDeviceToHost));
  return POST_OK;
}

void post_init_free(PostInitSession *s) { delete s; }

/* ------------------------- benchmark ------------------------- */

int post_benchmark(uint32_t provider_id, uint32_t scrypt_n,
                   uint64_t *labels_per_sec) {
  int rc = require_gpu(provider_id);
  if (rc != POST_OK) return rc;
  PostInitConfig cfg;
  std::memset(&cfg, 0, sizeof(cfg));
  std::memset(cfg.node_id, 0xA5, 32);
  std::memset(cfg.commitment_atx_id, 0x5A, 32);
  cfg.num_units = 1;
  cfg.labels_per_unit = 1ull << 18;
  cfg.max_file_size = 1ull << 40;
  cfg.scrypt_n = scrypt_n;
  cfg.provider_id = provider_id;
  cfg.scratch_bytes = 8ull << 30;
  PostInitSession *s = nullptr;
  rc = post_init_new(&cfg, &s);
  if (rc != POST_OK) return rc;
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, nullptr);
  rc = post_init_run(s);
  (void)hipEventRecord(t1, nullptr);
  (void)hipEventSynchronize(t1);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, t0, t1);
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  if (rc == POST_OK && ms > 0)
    *labels_per_sec = (uint64_t)((double)cfg.labels_per_unit / (ms / 1e3));
  post_init_free(s);
  return rc;
}

/* ------------------------- proving ------------------------- */

/* label source: fill dst with labels [base, base+cnt) (16 B each) */
using LabelReader =
    std::function<int(uint64_t base, uint64_t cnt, uint8_t *dst)>;

static int prove_core(const LabelReader &read_labels, uint64_t num_labels,
                      const PostProveConfig *cfg, PostProof *out) {
  if (cfg->nonces == 0 || cfg->nonces % POSTE_NONCE_GROUP != 0) {
    set_error("nonces must be a positive multiple of 16");
    return POST_ERR_INVALID_ARGS;
  }
  if (cfg->pow_mode != POST_POW_MODE_BLAKE3) {
    set_error("RandomX k2pow is not supported by this engine (parity-"
              "unpinned; use POST_POW_MODE_BLAKE3)");
    return POST_ERR_UNSUPPORTED;
  }
  int rc = require_gpu(cfg->provider_id);
  if (rc != POST_OK) return rc;
  int dev = (int)cfg->provider_id;
  DeviceTables tbl;
  rc = get_aes_tables(dev, tbl);
  if (rc != POST_OK) return rc;

  const uint32_t n_ciphers = cfg->nonces / POSTE_NONCES_PER_AES;
  const uint32_t n_groups = cfg->nonces / POSTE_NONCE_GROUP;
  std::vector<uint64_t> group_pow(n_groups);
  for (uint32_t g = 0; g < n_groups; g++)
    group_pow[g] = poste::k2pow_search_blake3(cfg->challenge, g,
                                              cfg->pow_difficulty,
                                              cfg->pow_threads);
  std::vector<uint32_t> rk((size_t)n_ciphers * 44);
  for (uint32_t c = 0; c < n_ciphers; c++) {
    uint8_t key[16];
    uint32_t grp = (c * POSTE_NONCES_PER_AES) / POSTE_NONCE_GROUP;
    poste::prove_cipher_key(cfg->challenge, c, group_pow[grp], key);
    poste::aes128_expand(key, rk.data() + (size_t)c * 44);
  }
  uint32_t *d_rk = nullptr;
  HIP_TRY(hipMalloc(&d_rk, rk.size() * 4));
  HIP_TRY(hipMemcpy(d_rk, rk.data(), rk.size() * 4, hipMemcpyHostToDevice));

  /* double-buffered pipeline: disk/host read of chunk i+1 overlaps the
   * device scan of chunk i (the 256-GiB config-4 pass never needs the
   * whole label set in host memory).  Expected total hits = nonces*k1
   * (a few thousand) regardless of label count, so hits accumulate in one
   * device buffer drained once at the end. */
  const uint64_t CHUNK = 1ull << 24; /* 16M labels = 256 MiB per chunk */
  const uint64_t chunk_lab = std::min(CHUNK, num_labels);
  uint8_t *d_labels[2] = {nullptr, nullptr};
  uint8_t *h_labels[2] = {nullptr, nullptr};
  hipStream_t stream = nullptr;
  hipEvent_t done[2] = {nullptr, nullptr};
  const uint32_t HIT_CAP = 1u << 22;
  PostScanHit *d_hits = nullptr;
  unsigned int *d_hit_count = nullptr;
  auto cleanup = [&] {
    for (int p = 0; p < 2; p++) {
      if (d_labels[p]) (void)hipFree(d_labels[p]);
      if (h_labels[p]) (void)hipHostFree(h_labels[p]);
      if (done[p]) (void)hipEventDestroy(done[p]);
    }
    if (stream) (void)hipStreamDestroy(stream);
    if (d_hits) (void)hipFree(d_hits);
    if (d_hit_count) (void)hipFree(d_hit_count);
    (void)hipFree(d_rk);
  };
#undef HIP_TRY
#define HIP_TRY(expr)                                                          \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      cleanup();                                                               \
      return hip_fail(#expr, _e);                                              \
    }                                                                          \
  } while (0)
  for (int p = 0; p < 2; p++) {
    HIP_TRY(hipMalloc(&d_labels[p], chunk_lab * 16));
    HIP_TRY(hipHostMalloc(&h_labels[p], chunk_lab * 16));
    HIP_TRY(hipEventCreate(&done[p]));
  }
  HIP_TRY(hipStreamCreate(&stream));
  HIP_TRY(hipMalloc(&d_hits, sizeof(PostScanHit) * HIT_CAP));
  HIP_TRY(hipMalloc(&d_hit_count, 4));
  unsigned int zero = 0;
  HIP_TRY(hipMemcpy(d_hit_count, &zero, 4, hipMemcpyHostToDevice));

  ScanKernelArgs sa;
  std::memset(&sa, 0, sizeof(sa));
  sa.te = tbl.d_te;
  sa.sbox = tbl.d_sbox;
  sa.rk = d_rk;
  sa.n_ciphers = n_ciphers;
  sa.difficulty = poste::proving_difficulty(cfg->k1, num_labels);
  sa.hits = d_hits;
  sa.hit_count = d_hit_count;
  sa.hit_cap = HIT_CAP;

  int parity = 0;
  bool inflight[2] = {false, false};
  for (uint64_t base = 0; base < num_labels; base += CHUNK, parity ^= 1) {
    uint64_t cnt = std::min(CHUNK, num_labels - base);
    /* wait until this parity's buffers are free, then stage the chunk */
    if (inflight[parity]) HIP_TRY(hipEventSynchronize(done[parity]));
    int rrc = read_labels(base, cnt, h_labels[parity]);
    if (rrc != POST_OK) {
      cleanup();
      return rrc;
    }
    HIP_TRY(hipMemcpyAsync(d_labels[parity], h_labels[parity], cnt * 16,
                           hipMemcpyHostToDevice, stream));
    sa.labels = (const uint4 *)d_labels[parity];
    sa.count = cnt;
    sa.index_base = base;
    uint32_t blocks = (uint32_t)std::min<uint64_t>(
        (cnt + THREADS - 1) / THREADS, 8192);
    HIP_TRY(poste_launch_scan_kernel(&sa, blocks, stream));
    HIP_TRY(hipEventRecord(done[parity], stream));
    inflight[parity] = true;
  }
  HIP_TRY(hipStreamSynchronize(stream));
  unsigned int n_hits = 0;
  HIP_TRY(hipMemcpy(&n_hits, d_hit_count, 4, hipMemcpyDeviceToHost));
  if (n_hits > HIT_CAP) {
    cleanup();
    set_error("scan hit buffer overflow (pathological K1/num_labels "
              "configuration)");
    return POST_ERR;
  }
  std::vector<PostScanHit> all_hits(n_hits);
  if (n_hits)
    HIP_TRY(hipMemcpy(all_hits.data(), d_hits, sizeof(PostScanHit) * n_hits,
                      hipMemcpyDeviceToHost));
#undef HIP_TRY
#define HIP_TRY(expr)                                                          \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) return hip_fail(#expr, _e);                          \
  } while (0)
  cleanup();

  /* winner: nonce whose k2-th smallest passing index is smallest
   * (streaming-order first across the ascending scan); tie -> lowest nonce */
  std::vector<std::vector<uint64_t>> per_nonce(cfg->nonces);
  for (const auto &h : all_hits) per_nonce[h.nonce].push_back(h.index);
  int64_t best_nonce = -1;
  uint64_t best_kth = UINT64_MAX;
  for (uint32_t nn = 0; nn < cfg->nonces; nn++) {
    auto &v = per_nonce[nn];
    if (v.size() < cfg->k2) continue;
    std::sort(v.begin(), v.end());
    uint64_t kth = v[cfg->k2 - 1];
    if (kth < best_kth) {
      best_kth = kth;
      best_nonce = nn;
    }
  }
  if (best_nonce < 0) {
    set_error("no nonce reached k2 passing indices");
    return POST_ERR_NO_NONCE;
  }
  out->nonce = (uint32_t)best_nonce;
  out->pow = group_pow[best_nonce / POSTE_NONCE_GROUP];
  out->num_indices = (uint16_t)cfg->k2;
  uint32_t bpi = poste::bits_per_index(num_labels);
  out->indices_len = poste::pack_indices(per_nonce[best_nonce].data(),
                                         cfg->k2, bpi, out->indices,
                                         POST_MAX_INDICES_BYTES);
  if (!out->indices_len) {
    set_error("indices exceed the 800-byte wire cap");
    return POST_ERR;
  }
  return POST_OK;
}

int post_prove_buffer(const uint8_t *labels, uint64_t num_labels,
                      const uint8_t node_id[32],
                      const uint8_t commitment_atx_id[32],
                      const PostProveConfig *cfg, PostProof *out) {
  (void)node_id;
  (void)commitment_atx_id; /* scan operates on labels only */
  if (!labels || !cfg || !out) return POST_ERR_INVALID_ARGS;
  LabelReader reader = [labels](uint64_t base, uint64_t cnt, uint8_t *dst) {
    std::memcpy(dst, labels + base * 16, cnt * 16);
    return POST_OK;
  };
  return prove_core(reader, num_labels, cfg, out);
}

int post_k2pow_search(const uint8_t challenge[32], uint32_t nonce_group,
                      const uint8_t pow_difficulty[32], uint32_t pow_mode,
                      uint32_t threads, uint64_t *out) {
  if (pow_mode != POST_POW_MODE_BLAKE3) {
    set_error("RandomX k2pow is not supported (use POST_POW_MODE_BLAKE3)");
    return POST_ERR_UNSUPPORTED;
  }
  *out = poste::k2pow_search_blake3(challenge, nonce_group, pow_difficulty,
                                    threads);
  return POST_OK;
}

int post_prove_scan(const uint8_t *labels, uint64_t count,
                    uint64_t index_base, uint64_t total_labels,
                    const PostProveConfig *cfg, const uint64_t *group_pows,
                    PostScanHitOut *hits, uint32_t cap, uint32_t *n_hits) {
  if (!labels || !cfg || !group_pows || !hits || !n_hits)
    return POST_ERR_INVALID_ARGS;
  if (cfg->nonces == 0 || cfg->nonces % POSTE_NONCE_GROUP != 0) {
    set_error("nonces must be a positive multiple of 16");
    return POST_ERR_INVALID_ARGS;
  }
  int rc = require_gpu(cfg->provider_id);
  if (rc != POST_OK) return rc;
  DeviceTables tbl;
  rc = get_aes_tables((int)cfg->provider_id, tbl);
  if (rc != POST_OK) return rc;

  const uint32_t n_ciphers = cfg->nonces / POSTE_NONCES_PER_AES;
  std::vector<uint32_t> rk((size_t)n_ciphers * 44);
  for (uint32_t c = 0; c < n_ciphers; c++) {
    uint8_t key[16];
    uint32_t grp = (c * POSTE_NONCES_PER_AES) / POSTE_NONCE_GROUP;
    poste::prove_cipher_key(cfg->challenge, c, group_pows[grp], key);
    poste::aes128_expand(key, rk.data() + (size_t)c * 44);
  }
  uint32_t *d_rk = nullptr;
  uint8_t *d_labels = nullptr;
  PostScanHit *d_hits = nullptr;
  unsigned int *d_hit_count = nullptr;
  auto cleanup = [&] {
    if (d_rk) (void)hipFree(d_rk);
    if (d_labels) (void)hipFree(d_labels);
    if (d_hits) (void)hipFree(d_hits);
    if (d_hit_count) (void)hipFree(d_hit_count);
  };
#undef HIP_TRY
#define HIP_TRY(expr)                                                          \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      cleanup();                                                               \
      return hip_fail(#expr, _e);                                              \
    }                                                                          \
  } while (0)
  HIP_TRY(hipMalloc(&d_rk, rk.size() * 4));
  HIP_TRY(hipMemcpy(d_rk, rk.data(), rk.size() * 4, hipMemcpyHostToDevice));
  const uint64_t CHUNK = 1ull << 24;
  HIP_TRY(hipMalloc(&d_labels, std::min(CHUNK, count) * 16));
  HIP_TRY(hipMalloc(&d_hits, sizeof(PostScanHit) * cap));
  HIP_TRY(hipMalloc(&d_hit_count, 4));
  unsigned int zero = 0;
  HIP_TRY(hipMemcpy(d_hit_count, &zero, 4, hipMemcpyHostToDevice));

  ScanKernelArgs sa;
  std::memset(&sa, 0, sizeof(sa));
  sa.te = tbl.d_te;
  sa.sbox = tbl.d_sbox;
  sa.rk = d_rk;
  sa.n_ciphers = n_ciphers;
  sa.difficulty = poste::proving_difficulty(cfg->k1, total_labels);
  sa.hits = d_hits;
  sa.hit_count = d_hit_count;
  sa.hit_cap = cap;
  for (uint64_t base = 0; base < count; base += CHUNK) {
    uint64_t cnt = std::min(CHUNK, count - base);
    HIP_TRY(hipMemcpy(d_labels, labels + base * 16, cnt * 16,
                      hipMemcpyHostToDevice));
    sa.labels = (const uint4 *)d_labels;
    sa.count = cnt;
    sa.index_base = index_base + base;
    uint32_t blocks = (uint32_t)std::min<uint64_t>(
        (cnt + THREADS - 1) / THREADS, 8192);
    HIP_TRY(poste_launch_scan_kernel(&sa, blocks, nullptr));
    HIP_TRY(hipDeviceSynchronize());
  }
  unsigned int got = 0;
  HIP_TRY(hipMemcpy(&got, d_hit_count, 4, hipMemcpyDeviceToHost));
  if (got > cap) {
    cleanup();
    set_error("scan hit buffer overflow (raise cap)");
    return POST_ERR;
  }
  std::vector<PostScanHit> tmp(got);
  if (got)
    HIP_TRY(hipMemcpy(tmp.data(), d_hits, sizeof(PostScanHit) * got,
                      hipMemcpyDeviceToHost));
#undef HIP_TRY
#define HIP_TRY(expr)                                                          \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) return hip_fail(#expr, _e);                          \
  } while (0)
  cleanup();
  for (unsigned int i = 0; i < got; i++) {
    hits[i].index = tmp[i].index;
    hits[i].nonce = tmp[i].nonce;
    hits[i].pad = 0;
  }
  *n_hits = got;
  return POST_OK;
}

int post_prove(const char *data_dir, const PostProveConfig *cfg,
               PostProof *out) {
  if (!data_dir || !cfg || !out) return POST_ERR_INVALID_ARGS;
  /* map postdata_*.bin in index order; stream chunks (never the whole set
   * in host memory — config 4 is 256 GiB) */
  struct FileSpan {
    std::string path;
    uint64_t first_label;
    uint64_t labels;
  };
  std::vector<FileSpan> files;
  uint64_t total = 0;
  for (uint64_t i = 0;; i++) {
    char path[4096];
    std::snprintf(path, sizeof path, "%s/postdata_%llu.bin", data_dir,
                  (unsigned long long)i);
    FILE *f = std::fopen(path, "rb");
    if (!f) break;
    std::fseek(f, 0, SEEK_END);
    long sz = std::ftell(f);
    std::fclose(f);
    uint64_t nlab = (uint64_t)(sz < 0 ? 0 : sz) / 16;
    files.push_back({path, total, nlab});
    total += nlab;
  }
  if (total == 0) {
    set_error("no postdata_*.bin in data_dir");
    return POST_ERR_IO;
  }
  LabelReader reader = [&files](uint64_t base, uint64_t cnt, uint8_t *dst)
      -> int {
    uint64_t done = 0;
    for (const auto &fs : files) {
      if (done == cnt) break;
      uint64_t want = base + done;
      if (want < fs.first_label || want >= fs.first_label + fs.labels)
        continue;
      uint64_t in_file = want - fs.first_label;
      uint64_t take = std::min(fs.labels - in_file, cnt - done);
      FILE *f = std::fopen(fs.path.c_str(), "rb");
      if (!f) {
        set_error("cannot reopen " + fs.path);
        return POST_ERR_IO;
      }
      if (std::fseek(f, (long)(in_file * 16), SEEK_SET) != 0 ||
          std::fread(dst + done * 16, 16, take, f) != take) {
        std::fclose(f);
        set_error("short read from " + fs.path);
        return POST_ERR_IO;
      }
      std::fclose(f);
      done += take;
    }
    if (done != cnt) {
      set_error("label range not covered by postdata files");
      return POST_ERR_IO;
    }
    return POST_OK;
  };
  return prove_core(reader, total, cfg, out);
}

/* ------------------------- verification ------------------------- */

int post_verify_batch(const PostProof *proofs, const PostProofMetadata *metas,
                      uint32_t n, const PostVerifyConfig *cfg, int *statuses,
                      uint32_t *invalid_indices) {
  return post_verify_batch_seeded(proofs, metas, n, cfg, nullptr, 0,
                                  statuses, invalid_indices);
}

int post_verify_batch_seeded(const PostProof *proofs,
                             const PostProofMetadata *metas, uint32_t n,
                             const PostVerifyConfig *cfg,
                             const uint8_t *subset_seeds, size_t seed_len,
                             int *statuses, uint32_t *invalid_indices) {
  if (!proofs || !metas || !cfg || !statuses) return POST_ERR_INVALID_ARGS;
  if (cfg->pow_mode != POST_POW_MODE_BLAKE3) {
    set_error("RandomX k2pow is not supported (use POST_POW_MODE_BLAKE3)");
    return POST_ERR_UNSUPPORTED;
  }
  int rc = require_gpu(cfg->provider_id);
  if (rc != POST_OK) return rc;

  struct Task {
    uint64_t label_index;
    uint32_t proof;
    uint32_t position; /* position within the proof's k2 indices */
  };
  std::vector<Task> tasks;
  std::vector<uint32_t> commit_words((size_t)n * 8);
  std::vector<std::vector<uint64_t>> proof_indices(n);
  std::vector<std::vector<Task>> per_proof_tasks(n);

  /* per-proof host prep (commitment blake3, pow verify, index unpack,
   * subset sampling) is independent across proofs and dominates the
   * K3=1 batch wall time — spread it across host threads; every slot
   * written is per-proof disjoint.  The commitment table is indexed by
   * ABSOLUTE proof number from the kernel tasks, so every proof gets a
   * slot — including ones rejected here. */
  auto prep_one = [&](uint32_t p) {
    statuses[p] = POST_OK;
    const PostProof &pr = proofs[p];
    const PostProofMetadata &me = metas[p];
    {
      uint8_t cm[32];
      poste::commitment(me.node_id, me.commitment_atx_id, cm);
      std::memcpy(commit_words.data() + (size_t)p * 8, cm, 32);
    }
    uint64_t num_labels = (uint64_t)me.num_units * me.labels_per_unit;
    if (num_labels == 0) { /* malformed metadata */
      statuses[p] = POST_ERR_INVALID_ARGS;
      return;
    }
    uint32_t bpi = poste::bits_per_index(num_labels);
    if (pr.num_indices != cfg->k2 ||
        pr.indices_len != ((uint64_t)cfg->k2 * bpi + 7) / 8) {
      statuses[p] = POST_ERR_INVALID_ARGS;
      return;
    }
    uint32_t group = pr.nonce / POSTE_NONCE_GROUP;
    if (poste::k2pow_verify_blake3(me.challenge, group, pr.pow,
                                   cfg->pow_difficulty) != 0) {
      statuses[p] = POST_ERR_POW;
      return;
    }
    proof_indices[p].resize(cfg->k2);
    poste::unpack_indices(pr.indices, cfg->k2, bpi,
                          proof_indices[p].data());

    std::vector<uint32_t> positions;
    const uint8_t *seed = subset_seeds ? subset_seeds + (size_t)p * seed_len
                                       : cfg->subset_seed;
    size_t slen = subset_seeds ? seed_len : cfg->subset_seed_len;
    if (cfg->selected_index >= 0) {
      positions.push_back((uint32_t)cfg->selected_index);
    } else if (seed && cfg->k3 < cfg->k2) {
      positions.resize(cfg->k3);
      poste::subset_positions(cfg->k2, cfg->k3, seed, slen,
                              positions.data());
    } else {
      positions.resize(cfg->k2);
      for (uint32_t i = 0; i < cfg->k2; i++) positions[i] = i;
    }
    for (uint32_t pos : positions) {
      if (pos >= cfg->k2) {
        statuses[p] = POST_ERR_INVALID_ARGS;
        return;
      }
      uint64_t li = proof_indices[p][pos];
      if (li >= num_labels) {
        statuses[p] = POST_ERR_INVALID_INDEX;
        if (invalid_indices) invalid_indices[p] = pos;
        return;
      }
      per_proof_tasks[p].push_back({li, p, pos});
    }
  };
  {
    unsigned hw = std::thread::hardware_concurrency();
    unsigned nthreads = std::min<unsigned>(hw ? hw : 1, 32);
    if (n < 64) nthreads = 1; /* not worth spawning for small batches */
    if (nthreads <= 1) {
      for (uint32_t p = 0; p < n; p++) prep_one(p);
    } else {
      std::vector<std::thread> ths;
      std::atomic<uint32_t> next{0};
      for (unsigned t = 0; t < nthreads; t++)
        ths.emplace_back([&] {
          for (;;) {
            uint32_t p = next.fetch_add(64);
            if (p >= n) return;
            uint32_t e = std::min(n, p + 64);
            for (; p < e; p++) prep_one(p);
          }
        });
      for (auto &th : ths) th.join();
    }
  }
  /* concatenate in proof order (deterministic task->index mapping) */
  for (uint32_t p = 0; p < n; p++) {
    if (statuses[p] != POST_OK) continue;
    tasks.insert(tasks.end(), per_proof_tasks[p].begin(),
                 per_proof_tasks[p].end());
  }
  if (tasks.empty()) return POST_OK;

  /* scratch-bounded chunks of label recomputes on the GPU */
  size_t free_b = 0, total_b = 0;
  HIP_TRY(hipMemGetInfo(&free_b, &total_b));
  uint32_t gap_shift = pick_gap_shift(cfg->scrypt_n);
  uint64_t per_lane = ((uint64_t)cfg->scrypt_n >> gap_shift) * 128;
  uint64_t max_lanes = (uint64_t)((double)free_b * 0.75) / per_lane;
  max_lanes = std::min<uint64_t>(max_lanes,
                                 poste_label_resident_slots(gap_shift));
  max_lanes = (max_lanes / 128) * 128;
  if (max_lanes == 0) max_lanes = 128;
  if (max_lanes == 0) {
    set_error("not enough memory for verification scratch");
    return POST_ERR_OOM;
  }

  std::vector<uint64_t> h_idx(tasks.size());
  std::vector<uint32_t> h_cid(tasks.size());
  for (size_t i = 0; i < tasks.size(); i++) {
    h_idx[i] = tasks[i].label_index;
    h_cid[i] = tasks[i].proof;
  }
  uint64_t lanes = std::min<uint64_t>(
      max_lanes, ((tasks.size() + 127) / 128) * 128);

  /* per-proof cipher data for the device predicate (absolute indexing:
   * every proof gets a slot, matching task_proof).  Declared here,
   * COMPUTED after the label kernel launches — the ~10k blake3+AES key
   * expansions overlap the label recompute instead of preceding it. */
  std::vector<uint32_t> vrk((size_t)n * 44, 0);
  std::vector<uint8_t> vhalf(n, 0);
  std::vector<uint64_t> vdiff(n, 0);
  auto cipher_prep = [&] {
    for (uint32_t p = 0; p < n; p++) {
      if (statuses[p] != POST_OK) continue;
      const PostProof &pr = proofs[p];
      const PostProofMetadata &me = metas[p];
      uint8_t key[16];
      poste::prove_cipher_key(me.challenge, pr.nonce / POSTE_NONCES_PER_AES,
                              pr.pow, key);
      poste::aes128_expand(key, vrk.data() + (size_t)p * 44);
      vhalf[p] = (uint8_t)(pr.nonce % POSTE_NONCES_PER_AES);
      uint64_t num_labels = (uint64_t)me.num_units * me.labels_per_unit;
      vdiff[p] = poste::proving_difficulty(cfg->k1, num_labels);
    }
  };
  DeviceTables tbl;
  rc = get_aes_tables((int)cfg->provider_id, tbl);
  if (rc != POST_OK) return rc;

  std::vector<uint8_t> h_pass(tasks.size());
  const bool dbg = getenv("POST_VERIFY_DEBUG") != nullptr;
  auto tick = [] {
    return std::chrono::duration<double>(
               std::chrono::steady_clock::now().time_since_epoch())
        .count();
  };
  double t0 = tick();
  {
    VerifyWorkspace *ws = get_verify_ws((int)cfg->provider_id);
    std::lock_guard<std::mutex> lk(ws->mu);
    rc = ws_reserve(ws, (size_t)lanes * per_lane, tasks.size(),
                    commit_words.size());
    if (rc != POST_OK) return rc;
    if (dbg) {
      std::fprintf(stderr, "[verify] reserve %.3fs (tasks=%zu lanes=%llu)\n",
                   tick() - t0, tasks.size(), (unsigned long long)lanes);
      t0 = tick();
    }
    HIP_TRY(hipMemcpy(ws->d_idx, h_idx.data(), h_idx.size() * 8,
                      hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(ws->d_cid, h_cid.data(), h_cid.size() * 4,
                      hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(ws->d_cm, commit_words.data(),
                      commit_words.size() * 4, hipMemcpyHostToDevice));

    LabelKernelArgs la;
    std::memset(&la, 0, sizeof(la));
    la.scrypt_n = cfg->scrypt_n;
    la.gap_shift = gap_shift;
    la.xbuf = ws->d_xbuf;
    la.out_full = 1;
    la.scratch = ws->d_scratch;
    la.scratch_lanes = lanes;
    la.out = ws->d_out;
    la.indices = h_idx.size() ? ws->d_idx : nullptr;
    la.commit_ids = ws->d_cid;
    la.commitments = ws->d_cm;
    la.count = tasks.size();
    if (dbg) {
      std::fprintf(stderr, "[verify] h2d %.3fs\n", tick() - t0);
      t0 = tick();
    }
    HIP_TRY(poste_launch_label_kernel(&la, (uint32_t)(lanes / 64),
                                      nullptr));
    cipher_prep(); /* overlaps the label kernel (launch is async) */
    if (dbg) {
      (void)hipDeviceSynchronize();
      std::fprintf(stderr, "[verify] label kernel %.3fs\n", tick() - t0);
      t0 = tick();
    }
    /* device-side AES threshold predicate over the recomputed labels */
    rc = ws_reserve_proofs(ws, n);
    if (rc != POST_OK) return rc;
    HIP_TRY(hipMemcpy(ws->d_vrk, vrk.data(), vrk.size() * 4,
                      hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(ws->d_half, vhalf.data(), vhalf.size(),
                      hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(ws->d_vdiff, vdiff.data(), vdiff.size() * 8,
                      hipMemcpyHostToDevice));
    VerifyPredArgs pa;
    std::memset(&pa, 0, sizeof(pa));
    pa.labels2 = (const uint4 *)ws->d_out;
    pa.count = tasks.size();
    pa.te = tbl.d_te;
    pa.sbox = tbl.d_sbox;
    pa.rk = ws->d_vrk;
    pa.task_proof = ws->d_cid; /* task->proof map doubles as commit id */
    pa.half = ws->d_half;
    pa.difficulty = ws->d_vdiff;
    pa.pass = ws->d_pass;
    uint32_t pblocks = (uint32_t)std::min<uint64_t>(
        (tasks.size() + THREADS - 1) / THREADS, 8192);
    HIP_TRY(poste_launch_verify_pred_kernel(&pa, pblocks, nullptr));
    HIP_TRY(hipDeviceSynchronize());
    if (dbg) {
      std::fprintf(stderr, "[verify] predicate %.3fs\n", tick() - t0);
      t0 = tick();
    }
    HIP_TRY(hipMemcpy(h_pass.data(), ws->d_pass, h_pass.size(),
                      hipMemcpyDeviceToHost));
  }

  /* collect per-proof verdicts from the device predicate */
  for (size_t i = 0; i < tasks.size(); i++) {
    const Task &t = tasks[i];
    if (statuses[t.proof] != POST_OK) continue;
    if (!h_pass[i]) {
      statuses[t.proof] = POST_ERR_INVALID_INDEX;
      if (invalid_indices) invalid_indices[t.proof] = t.position;
    }
  }
  return POST_OK;
}

int post_verify(const PostProof *proof, const PostProofMetadata *meta,
                const PostVerifyConfig *cfg, uint32_t *invalid_index) {
  int status = 0;
  uint32_t inv = 0;
  int rc = post_verify_batch(proof, meta, 1, cfg, &status, &inv);
  if (rc != POST_OK) return rc;
  if (invalid_index) *invalid_index = inv;
  return status;
}

int post_verify_vrf_nonce(const PostProofMetadata *meta, uint64_t index,
                          uint32_t scrypt_n, uint32_t provider_id) {
  if (!meta) return POST_ERR_INVALID_ARGS;
  int rc = require_gpu(provider_id);
  if (rc != POST_OK) return rc;
  uint64_t num_labels = (uint64_t)meta->num_units * meta->labels_per_unit;
  uint8_t cm[32];
  poste::commitment(meta->node_id, meta->commitment_atx_id, cm);

  uint32_t gap_shift = pick_gap_shift(scrypt_n);
  uint64_t per_lane = ((uint64_t)scrypt_n >> gap_shift) * 128;
  uint32_t *d_scratch = nullptr;
  uint64_t *d_idx = nullptr;
  uint8_t *d_out = nullptr;
  uint32_t *d_xbuf1 = nullptr;
  HIP_TRY(hipMalloc(&d_scratch, (size_t)128 * per_lane));
  HIP_TRY(hipMalloc(&d_idx, 8));
  HIP_TRY(hipMalloc(&d_out, 32));
  HIP_TRY(hipMalloc(&d_xbuf1, 128));
  HIP_TRY(hipMemcpy(d_idx, &index, 8, hipMemcpyHostToDevice));
  LabelKernelArgs la;
  std::memset(&la, 0, sizeof(la));
  load_commitment_words(cm, la.commitment_le);
  la.scrypt_n = scrypt_n;
  la.gap_shift = gap_shift;
  la.xbuf = d_xbuf1;
  la.out_full = 1;
  la.scratch = d_scratch;
  la.scratch_lanes = 128;
  la.out = d_out;
  la.indices = d_idx;
  la.count = 1;
  HIP_TRY(poste_launch_label_kernel(&la, 1, nullptr));
  HIP_TRY(hipDeviceSynchronize());
  uint8_t full[32];
  HIP_TRY(hipMemcpy(full, d_out, 32, hipMemcpyDeviceToHost));
  (void)hipFree(d_scratch);
  (void)hipFree(d_idx);
  (void)hipFree(d_out);
  (void)hipFree(d_xbuf1);
  uint8_t difficulty[32];
  poste::vrf_difficulty(num_labels, difficulty);
  if (std::memcmp(full, difficulty, 32) < 0) return POST_OK;
  set_error("vrf nonce label above threshold");
  return POST_ERR_INVALID_INDEX;
}

/* ------------------------- self-test exports ------------------------- */

void post_selftest_blake3(const uint8_t *msg, size_t len, uint8_t out[32]) {
  poste::blake3(msg, len, out, 32);
}

void post_selftest_aes128(const uint8_t key[16], const uint8_t in[16],
                          uint8_t out[16]) {
  uint32_t rk[44];
  poste::aes128_expand(key, rk);
  poste::aes128_enc_block(rk, in, out);
}

int post_selftest_label(const uint8_t node_id[32],
                        const uint8_t commitment_atx_id[32], uint64_t index,
                        uint32_t scrypt_n, uint8_t out[32]) {
  uint8_t cm[32];
  poste::commitment(node_id, commitment_atx_id, cm);
  return poste::host_label(cm, index, scrypt_n, out);
}

} /* extern "C" */
