/* oracle_bench — CLI around the CPU oracle.
 * Test/baseline infrastructure only (see oracle.h header): this binary is
 * the bench.py `cpu_baseline` timer (the role of the reference's CPU
 * provider, initialization.CPUProviderID(), activation/post_test.go:356) and
 * the postdata_*.bin writer for BASELINE config 1.  Never part of the
 * product path.
 *
 * Usage:
 *   oracle_bench bench --labels N [--scrypt-n 8192] [--threads T]
 *       -> one JSON line {"labels_per_sec":..., "threads":..., ...}
 *   oracle_bench init --out DIR --node-id HEX32 --atx-id HEX32
 *       --num-units U --labels-per-unit L [--scrypt-n N] [--max-file-size B]
 *       -> postdata_*.bin + postdata_metadata.json + JSON summary line
 */
#define _GNU_SOURCE
#include "oracle.h"
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <time.h>

#ifdef _OPENMP
#include <omp.h>
#endif

static double now_sec(void) {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return ts.tv_sec + ts.tv_nsec * 1e-9;
}

static int hex2bin(const char *hex, uint8_t *out, size_t n) {
  if (strlen(hex) != 2 * n) return -1;
  for (size_t i = 0; i < n; i++) {
    unsigned v;
    if (sscanf(hex + 2 * i, "%2x", &v) != 1) return -1;
    out[i] = (uint8_t)v;
  }
  return 0;
}

static const char *argval(int argc, char **argv, const char *name,
                          const char *dflt) {
  for (int i = 0; i < argc - 1; i++)
    if (!strcmp(argv[i], name)) return argv[i + 1];
  return dflt;
}

static void b64enc(const uint8_t *in, size_t n, char *out) {
  static const char tbl[] =
      "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789+/";
  size_t o = 0;
  for (size_t i = 0; i < n; i += 3) {
    uint32_t v = (uint32_t)in[i] << 16;
    if (i + 1 < n) v |= (uint32_t)in[i + 1] << 8;
    if (i + 2 < n) v |= in[i + 2];
    out[o++] = tbl[(v >> 18) & 63];
    out[o++] = tbl[(v >> 12) & 63];
    out[o++] = i + 1 < n ? tbl[(v >> 6) & 63] : '=';
    out[o++] = i + 2 < n ? tbl[v & 63] : '=';
  }
  out[o] = 0;
}

static int cmd_bench(int argc, char **argv) {
  uint64_t labels = strtoull(argval(argc, argv, "--labels", "1024"), NULL, 0);
  uint32_t n = (uint32_t)strtoul(argval(argc, argv, "--scrypt-n", "8192"),
                                 NULL, 0);
  int threads = atoi(argval(argc, argv, "--threads", "0"));
#ifdef _OPENMP
  if (threads > 0) omp_set_num_threads(threads);
  int used = threads > 0 ? threads : omp_get_max_threads();
#else
  (void)threads;
  int used = 1;
#endif
  uint8_t node_id[32], atx[32], commitment[32];
  memset(node_id, 0xA5, 32);
  memset(atx, 0x5A, 32);
  oracle_commitment(node_id, atx, commitment);
  uint8_t *out = malloc((size_t)labels * ORACLE_LABEL_SIZE);
  uint8_t difficulty[32];
  memset(difficulty, 0xff, 32);
  OracleVrfNonce best = {0, {0}, 0};
  double t0 = now_sec();
  int rc = oracle_init_range(commitment, 0, labels, n, out, difficulty, &best);
  double dt = now_sec() - t0;
  free(out);
  if (rc) {
    fprintf(stderr, "init failed\n");
    return 1;
  }
  printf("{\"labels\": %llu, \"seconds\": %.3f, \"labels_per_sec\": %.1f, "
         "\"threads\": %d, \"scrypt_n\": %u}\n",
         (unsigned long long)labels, dt, labels / dt, used, n);
  return 0;
}

static int cmd_init(int argc, char **argv) {
  const char *outdir = argval(argc, argv, "--out", NULL);
  const char *nid_hex = argval(argc, argv, "--node-id", NULL);
  const char *atx_hex = argval(argc, argv, "--atx-id", NULL);
  if (!outdir || !nid_hex || !atx_hex) {
    fprintf(stderr, "missing --out/--node-id/--atx-id\n");
    return 2;
  }
  uint32_t num_units =
      (uint32_t)strtoul(argval(argc, argv, "--num-units", "1"), NULL, 0);
  uint64_t lpu = strtoull(
      argval(argc, argv, "--labels-per-unit", "4194304"), NULL, 0);
  uint32_t n = (uint32_t)strtoul(argval(argc, argv, "--scrypt-n", "8192"),
                                 NULL, 0);
  uint64_t max_file = strtoull(
      argval(argc, argv, "--max-file-size", "4294967296"), NULL, 0);
  uint8_t node_id[32], atx[32], commitment[32];
  if (hex2bin(nid_hex, node_id, 32) || hex2bin(atx_hex, atx, 32)) {
    fprintf(stderr, "bad hex id\n");
    return 2;
  }
  oracle_commitment(node_id, atx, commitment);
  uint64_t total = (uint64_t)num_units * lpu;
  uint64_t labels_per_file = max_file / ORACLE_LABEL_SIZE;
  if (labels_per_file == 0) labels_per_file = 1;
  uint8_t difficulty[32];
  memset(difficulty, 0xff, 32);
  OracleVrfNonce best = {0, {0}, 0};
  uint64_t chunk = 1 << 18;
  uint8_t *buf = malloc((size_t)chunk * ORACLE_LABEL_SIZE);
  double t0 = now_sec();
  for (uint64_t file_i = 0, pos = 0; pos < total; file_i++) {
    uint64_t file_end = pos + labels_per_file;
    if (file_end > total) file_end = total;
    char path[4096];
    snprintf(path, sizeof path, "%s/postdata_%llu.bin", outdir,
             (unsigned long long)file_i);
    FILE *f = fopen(path, "wb");
    if (!f) {
      perror("fopen");
      free(buf);
      return 1;
    }
    while (pos < file_end) {
      uint64_t endc = pos + chunk < file_end ? pos + chunk : file_end;
      if (oracle_init_range(commitment, pos, endc, n, buf, difficulty,
                            &best)) {
        fclose(f);
        free(buf);
        return 1;
      }
      fwrite(buf, ORACLE_LABEL_SIZE, endc - pos, f);
      pos = endc;
    }
    fclose(f);
  }
  double dt = now_sec() - t0;
  free(buf);
  /* postdata_metadata.json — field set per shared.PostMetadata usage,
   * activation/post_test.go:305-309 (Go json: base64 byte slices). */
  char path[4096], nid_b64[64], atx_b64[64], nv_b64[64];
  snprintf(path, sizeof path, "%s/postdata_metadata.json", outdir);
  b64enc(node_id, 32, nid_b64);
  b64enc(atx, 32, atx_b64);
  b64enc(best.label, 32, nv_b64);
  FILE *mf = fopen(path, "w");
  if (!mf) {
    perror("fopen metadata");
    return 1;
  }
  fprintf(mf,
          "{\n  \"NodeId\": \"%s\",\n  \"CommitmentAtxId\": \"%s\",\n"
          "  \"LabelsPerUnit\": %llu,\n  \"NumUnits\": %u,\n"
          "  \"MaxFileSize\": %llu,\n  \"Scrypt\": {\"N\": %u, \"R\": 1, "
          "\"P\": 1}",
          nid_b64, atx_b64, (unsigned long long)lpu, num_units,
          (unsigned long long)max_file, n);
  if (best.found)
    fprintf(mf, ",\n  \"Nonce\": %llu,\n  \"NonceValue\": \"%s\"",
            (unsigned long long)best.index, nv_b64);
  fprintf(mf, "\n}\n");
  fclose(mf);
  printf("{\"labels\": %llu, \"seconds\": %.3f, \"labels_per_sec\": %.1f, "
         "\"nonce\": %llu, \"nonce_found\": %d}\n",
         (unsigned long long)total, dt, total / dt,
         (unsigned long long)best.index, best.found);
  return 0;
}

int main(int argc, char **argv) {
  if (argc < 2) {
    fprintf(stderr, "usage: oracle_bench bench|init ...\n");
    return 2;
  }
  if (!strcmp(argv[1], "bench")) return cmd_bench(argc - 2, argv + 2);
  if (!strcmp(argv[1], "init")) return cmd_init(argc - 2, argv + 2);
  fprintf(stderr, "unknown command %s\n", argv[1]);
  return 2;
}
