/* cubefs_amd/csrc/gfrs_host.cpp — C-ABI host runtime of the MI355X EC/CRC
 * engine (include/gfrs.h).
 *
 * Mirrors the blobstore/common/ec host semantics:
 *   ec.NewEncoder / encoder / lrcEncoder     (encoder.go:78-112, lrcencoder.go)
 *   Encode / Verify / Reconstruct[Data]      (encoder.go:114-151)
 *   LRC layering: global RS(n,m) + per-AZ RS((n+m)/az, l/az)
 *                                            (lrcencoder.go:35-82,133-207)
 *   inversionTree decode-matrix cache keyed by the missing-index set
 *                                            (inversion_tree.go:16-164) —
 *                                            here a bitmask-keyed map
 *   codemode local stripe layout             (codemode.go:301-318)
 *
 * GPU-only by design: no CPU compute fallback exists.  When no HIP device
 * is visible every compute call fails with GFRS_ERR_NO_GPU.
 */
#include "../../include/gfrs.h"
#include "gfrs_internal.h"

#include <cstdarg>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <map>
#include <mutex>
#include <set>
#include <string>
#include <algorithm>
#include <array>
#include <atomic>
#include <memory>
#include <vector>

namespace gfrs {

/* 256-bit shard bitmask + tag: collision-free plan-cache key for every
 * engine tactic_valid admits (n+m <= 256). */
typedef std::array<uint64_t, 5> PlanKey;

static inline PlanKey plan_key_tag(uint64_t tag) {
  PlanKey k{};
  k[4] = tag;
  return k;
}

static inline void plan_key_set(PlanKey &k, int i) {
  k[size_t(i) >> 6] |= 1ull << (unsigned(i) & 63u);
}

static thread_local std::string g_err;

static void seterr(const char *fmt, ...) {
  char buf[512];
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(buf, sizeof(buf), fmt, ap);
  va_end(ap);
  g_err = buf;
}

static int hip_fail(const char *what, hipError_t e) {
  seterr("%s: %s", what, hipGetErrorString(e));
  return GFRS_ERR_HIP;
}

#define HIP_TRY(call)                                    \
  do {                                                   \
    hipError_t e_ = (call);                              \
    if (e_ != hipSuccess) return hip_fail(#call, e_);    \
  } while (0)

/* per-device one-time CRC table init */
static std::mutex g_dev_mu;
static std::set<int> g_dev_inited;

static int ensure_device_init(int device) {
  std::lock_guard<std::mutex> lk(g_dev_mu);
  if (g_dev_inited.count(device)) return GFRS_OK;
  HIP_TRY(hipSetDevice(device));
  int rc = crc_device_init_current();
  if (rc != 0) {
    seterr("crc_device_init_current failed");
    return GFRS_ERR_HIP;
  }
  g_dev_inited.insert(device);
  return GFRS_OK;
}

/* Grow-only device buffer. */
struct DevBuf {
  void *p = nullptr;
  size_t cap = 0;
  int ensure(size_t n) {
    if (n <= cap) return GFRS_OK;
    if (p) hipFree(p);
    p = nullptr;
    cap = 0;
    HIP_TRY(hipMalloc(&p, n));
    cap = n;
    return GFRS_OK;
  }
  ~DevBuf() {
    if (p) hipFree(p);
  }
};

struct PinBuf {
  void *p = nullptr;
  size_t cap = 0;
  int ensure(size_t n) {
    if (n <= cap) return GFRS_OK;
    if (p) hipHostFree(p);
    p = nullptr;
    cap = 0;
    HIP_TRY(hipHostMalloc(&p, n));
    cap = n;
    return GFRS_OK;
  }
  ~PinBuf() {
    if (p) hipHostFree(p);
  }
};

/* An uploaded coding plan: index lists + per-coefficient nibble tables. */
struct DevPlan {
  DevBuf in_idx, out_idx, tabs;
  int k = 0, nout = 0;
  int upload(const std::vector<int32_t> &in, const std::vector<int32_t> &out,
             const std::vector<uint8_t> &rows /* nout*k coefficients */,
             hipStream_t s, int rowk = -1 /* row width; in may carry extra
             (cmp) indices after the first rowk inputs */) {
    const GfTables &t = gft();
    k = rowk >= 0 ? rowk : int(in.size());
    nout = int(out.size());
    /* 3-way linear-split tables, 32 B per coefficient (A8|B8|C4|pad):
     * GF(2^8) multiply-by-constant is GF(2)-linear, so mul_c(b) =
     * A[b&7] ^ B[(b>>3)&7] ^ C[b>>6] with A[i]=mul(c,i), B[i]=mul(c,8i),
     * C[i]=mul(c,64i).  Every kernel looks a packed dword up as
     * 3 v_perm + xor3 (+xor into acc) with selectors shared across
     * output rows — measured ~25% fewer VALU than the lo|hi nibble
     * form's 4 v_perm + 2 blends + 2 mask broadcasts. */
    std::vector<uint8_t> tb(size_t(nout) * k * 32, 0);
    for (int r = 0; r < nout; r++)
      for (int c = 0; c < k; c++) {
        uint8_t coef = rows[size_t(r) * k + c];
        uint8_t *lt = &tb[(size_t(r) * k + c) * 32];
        for (int i = 0; i < 8; i++) {
          lt[i] = t.mul[coef][i];
          lt[8 + i] = t.mul[coef][8 * i];
        }
        for (int i = 0; i < 4; i++) lt[16 + i] = t.mul[coef][64 * i];
      }
    int rc;
    if ((rc = in_idx.ensure(in.size() * 4)) != GFRS_OK) return rc;
    if ((rc = out_idx.ensure(out.size() * 4)) != GFRS_OK) return rc;
    if ((rc = tabs.ensure(tb.size())) != GFRS_OK) return rc;
    HIP_TRY(hipMemcpyAsync(in_idx.p, in.data(), in.size() * 4,
                           hipMemcpyHostToDevice, s));
    HIP_TRY(hipMemcpyAsync(out_idx.p, out.data(), out.size() * 4,
                           hipMemcpyHostToDevice, s));
    HIP_TRY(hipMemcpyAsync(tabs.p, tb.data(), tb.size(),
                           hipMemcpyHostToDevice, s));
    HIP_TRY(hipStreamSynchronize(s)); /* plans are built once, reused many times */
    return GFRS_OK;
  }
};

/* One independent execution lane: own stream + own scratch, so
 * concurrent single-stripe callers on one context (the reference shares
 * one ec.Encoder across ~100 goroutines behind a counting semaphore,
 * encoder.go:29,115) issue to the GPU in parallel instead of
 * serializing on one stream. */
struct Lane {
  hipStream_t stream = nullptr;
  DevBuf ptr_buf, fail_buf, stage_dev;
  PinBuf stage_pin;
  std::mutex mu;
  ~Lane() {
    if (stream) hipStreamDestroy(stream);
  }
};

/* view over either a lane's or the context's stream+scratch */
struct LaneView {
  hipStream_t stream;
  DevBuf *ptr_buf, *fail_buf, *stage_dev;
  PinBuf *stage_pin;
};

struct gfrs_ctx_impl {
  gfrs_tactic t{};
  int total = 0;   /* n+m+l */
  int local_n = 0; /* (n+m)/az when l>0 */
  int local_m = 0; /* l/az */
  int device = 0;
  hipStream_t own_stream = nullptr;
  hipStream_t stream = nullptr;
  std::mutex mu;

  std::vector<uint8_t> enc_matrix;   /* (n+m)×n */
  std::vector<uint8_t> local_matrix; /* (local_n+local_m)×local_n */
  DevPlan enc_plan;                  /* global encode */
  std::vector<DevPlan *> local_enc;  /* per-AZ local encode (global idx) */
  DevPlan local_plan;                /* local engine, identity idx (one
                                        local-stripe set, lrcencoder.go:94) */
  DevPlan fused_lrc;                 /* m+l output rows composed over the
                                        n data shards (locals are linear in
                                        the data through the parity rows) */
  bool fused_lrc_ok = false;
  /* Plan-cache key: full 256-bit missing/present bitmask (words 0..3) plus
   * an engine/namespace tag (word 4).  tactic_valid admits engines up to
   * n+m=256 shards, so a packed-u64 key would silently collide distinct
   * missing sets above 58 shards (and 1ull<<i is UB at i>=64); the wide
   * key makes every reachable engine collision-free. */
  std::map<PlanKey, DevPlan *> dec_cache; /* missing-bitmask → data decode */
  std::map<PlanKey, DevPlan *> par_cache; /* missing-bitmask → parity rows */

  DevBuf ptr_buf;   /* pointer tables for pointer-mode launches */
  DevBuf fail_buf;  /* verify flags / crc bad blocks */
  DevBuf stage_dev; /* host-mode staging */
  PinBuf stage_pin;

  /* finalize-pipeline resources (repair path): shard_finalize of chunk i
   * runs on aux_stream behind an event while chunk i+1's repair kernel
   * streams on the main stream */
  hipStream_t aux_stream = nullptr;
  hipEvent_t aux_ev[2] = {nullptr, nullptr};
  hipEvent_t aux_done = nullptr;
  int ensure_aux() {
    if (aux_stream) return GFRS_OK;
    if (hipStreamCreateWithFlags(&aux_stream, hipStreamNonBlocking) !=
        hipSuccess)
      return GFRS_ERR_HIP;
    for (int i = 0; i < 2; i++)
      if (hipEventCreateWithFlags(&aux_ev[i], hipEventDisableTiming) !=
          hipSuccess)
        return GFRS_ERR_HIP;
    if (hipEventCreateWithFlags(&aux_done, hipEventDisableTiming) !=
        hipSuccess)
      return GFRS_ERR_HIP;
    return GFRS_OK;
  }

  /* stream-lane pool for concurrent foreground (single-stripe) calls;
   * empty when GFRS_LANES=0 or after gfrs_set_stream pins a caller
   * stream (user_stream) */
  std::vector<std::unique_ptr<Lane>> lanes;
  std::atomic<uint32_t> lane_rr{0};
  bool user_stream = false;
  std::mutex cache_mu; /* plan caches (dec/par) — lanes bypass this->mu */

  Lane *pick_lane() {
    if (user_stream || lanes.empty()) return nullptr;
    return lanes[lane_rr.fetch_add(1) % lanes.size()].get();
  }

  ~gfrs_ctx_impl() {
    for (auto *p : local_enc) delete p;
    for (auto &kv : dec_cache) delete kv.second;
    for (auto &kv : par_cache) delete kv.second;
    if (own_stream) hipStreamDestroy(own_stream);
    if (aux_stream) hipStreamDestroy(aux_stream);
    for (int i = 0; i < 2; i++)
      if (aux_ev[i]) hipEventDestroy(aux_ev[i]);
    if (aux_done) hipEventDestroy(aux_done);
  }
};

struct StreamGuard {
  int prev = -1;
  explicit StreamGuard(const gfrs_ctx_impl *c) {
    hipGetDevice(&prev);
    hipSetDevice(c->device);
  }
  ~StreamGuard() {
    if (prev >= 0) hipSetDevice(prev);
  }
};

static inline LaneView view_of(gfrs_ctx_impl *c) {
  return LaneView{c->stream, &c->ptr_buf, &c->fail_buf, &c->stage_dev,
                  &c->stage_pin};
}
static inline LaneView view_of(Lane *L) {
  return LaneView{L->stream, &L->ptr_buf, &L->fail_buf, &L->stage_dev,
                  &L->stage_pin};
}

/* local stripe global indices for one AZ (codemode.go:301-318) */
static std::vector<int32_t> local_stripe(const gfrs_tactic &t, int az_idx) {
  int ln = t.n / t.az_count, lm = t.m / t.az_count, ll = t.l / t.az_count;
  std::vector<int32_t> idx;
  idx.reserve(ln + lm + ll);
  for (int i = 0; i < ln; i++) idx.push_back(az_idx * ln + i);
  for (int i = 0; i < lm; i++) idx.push_back(t.n + az_idx * lm + i);
  for (int i = 0; i < ll; i++) idx.push_back(t.n + t.m + az_idx * ll + i);
  return idx;
}

static bool tactic_valid(const gfrs_tactic *t) {
  /* codemode.go:291-299 IsValid; replicate modes (m==0, l==0) are legal —
   * ec.NewEncoder accepts them and Encode is a no-op (reedsolomon.go:442) */
  if (!t) return false;
  if (t->m == 0 && t->l == 0)
    return t->n > 0 && t->az_count > 0 && t->n % t->az_count == 0;
  if (t->n <= 0 || t->m <= 0 || t->l < 0 || t->az_count <= 0) return false;
  if (t->n % t->az_count || t->m % t->az_count || t->l % t->az_count)
    return false;
  if (t->n + t->m > 256) return false;
  if (t->l > 0 && (t->n + t->m) / t->az_count + t->l / t->az_count > 256)
    return false;
  return true;
}

}  // namespace gfrs

using namespace gfrs;

/* unlocked bodies of the public batch entry points (defined below);
 * already-locked callers reuse these instead of re-locking */
static int encode_batch_impl(gfrs_ctx_impl *c, void *base, size_t shard_len,
                             size_t stripe_stride, int nstripes,
                             hipStream_t st);
static int verify_batch_impl(gfrs_ctx_impl *c, const void *base,
                             size_t shard_len, size_t stripe_stride,
                             int nstripes, uint64_t *fail_bitmap,
                             hipStream_t st, DevBuf *fail);
static int crc32b_encode_batch_impl(gfrs_ctx_impl *c, void *dst,
                                    size_t dst_stride, const void *src,
                                    size_t src_stride, int64_t n,
                                    int64_t block_len, int nshards);

extern "C" {

const char *gfrs_last_error(void) { return g_err.c_str(); }
const char *gfrs_version(void) { return "gfrs 0.1.0 (gfx950)"; }

int gfrs_device_count(void) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

int gfrs_buffer_sizes(const gfrs_tactic *t, int64_t data_size,
                      int64_t *shard_size, int64_t *ec_data_size,
                      int64_t *ec_size) {
  /* buf.go:67-133 */
  if (!t || t->n <= 0) return GFRS_ERR_INVALID_CODEMODE;
  if (data_size <= 0) return GFRS_ERR_SHORT_DATA;
  int64_t ss = (data_size + t->n - 1) / t->n;
  if (ss < t->min_shard_size) ss = t->min_shard_size;
  *shard_size = ss;
  *ec_data_size = ss * t->n;
  *ec_size = ss * (t->n + t->m + t->l);
  return GFRS_OK;
}

gfrs_ctx *gfrs_create(const gfrs_tactic *t, int device) {
  if (!tactic_valid(t)) {
    seterr("invalid code mode tactic");
    return nullptr;
  }
  int ndev = gfrs_device_count();
  if (ndev == 0) {
    seterr("no HIP device visible (the gfrs engine has no CPU fallback)");
    return nullptr;
  }
  if (device < 0) hipGetDevice(&device);
  if (device >= ndev) {
    seterr("device %d out of range (%d visible)", device, ndev);
    return nullptr;
  }
  if (ensure_device_init(device) != GFRS_OK) return nullptr;

  auto *c = new gfrs_ctx_impl();
  c->t = *t;
  c->total = t->n + t->m + t->l;
  c->device = device;

  int prev = -1;
  hipGetDevice(&prev);
  hipSetDevice(device);
  bool ok = hipStreamCreate(&c->own_stream) == hipSuccess;
  c->stream = c->own_stream;
  if (ok) {
    const char *le = getenv("GFRS_LANES");
    int nl = le ? atoi(le) : 4;
    if (nl < 0) nl = 0;
    if (nl > 16) nl = 16;
    for (int i = 0; i < nl && ok; i++) {
      auto L = std::make_unique<Lane>();
      ok = hipStreamCreate(&L->stream) == hipSuccess;
      if (ok) c->lanes.push_back(std::move(L));
    }
  }

  /* global encode matrix (reedsolomon.go:220-244); replicate modes have
   * no parity engine (reedsolomon.go:442 parityShards==0 early return) */
  if (ok && t->m > 0) {
    c->enc_matrix.resize(size_t(t->n + t->m) * t->n);
    ok = gf_build_matrix(t->n, t->n + t->m, c->enc_matrix.data());
    if (ok) {
      std::vector<int32_t> in(t->n), out(t->m);
      for (int i = 0; i < t->n; i++) in[i] = i;
      for (int i = 0; i < t->m; i++) out[i] = t->n + i;
      std::vector<uint8_t> rows(c->enc_matrix.begin() + size_t(t->n) * t->n,
                                c->enc_matrix.end());
      ok = c->enc_plan.upload(in, out, rows, c->stream) == GFRS_OK;
    }
  }
  /* local engines (encoder.go:92-104) */
  if (ok && t->l > 0) {
    c->local_n = (t->n + t->m) / t->az_count;
    c->local_m = t->l / t->az_count;
    c->local_matrix.resize(size_t(c->local_n + c->local_m) * c->local_n);
    ok = gf_build_matrix(c->local_n, c->local_n + c->local_m,
                         c->local_matrix.data());
    if (ok) {
      std::vector<uint8_t> lrows(
          c->local_matrix.begin() + size_t(c->local_n) * c->local_n,
          c->local_matrix.end());
      for (int az = 0; az < t->az_count && ok; az++) {
        auto idx = local_stripe(*t, az);
        std::vector<int32_t> in(idx.begin(), idx.begin() + c->local_n);
        std::vector<int32_t> out(idx.begin() + c->local_n, idx.end());
        auto *p = new DevPlan();
        ok = p->upload(in, out, lrows, c->stream) == GFRS_OK;
        c->local_enc.push_back(p);
      }
      if (ok) {
        std::vector<int32_t> in(c->local_n), out(c->local_m);
        for (int i = 0; i < c->local_n; i++) in[i] = i;
        for (int i = 0; i < c->local_m; i++) out[i] = c->local_n + i;
        ok = c->local_plan.upload(in, out, lrows, c->stream) == GFRS_OK;
      }
      if (ok && t->m + t->l <= 4 && t->n + t->m + t->l <= 16) {
        /* compose every output (global parity AND local parity) as a row
         * over the n data shards: a local input that is a data shard
         * contributes its coefficient directly; one that is a global
         * parity contributes localcoef x its parity row (GF linearity) */
        const GfTables &gt2 = gft();
        std::vector<int32_t> in(t->n), out;
        std::vector<uint8_t> rows;
        for (int i = 0; i < t->n; i++) in[i] = i;
        for (int p2 = 0; p2 < t->m; p2++) {
          out.push_back(t->n + p2);
          const uint8_t *er = &c->enc_matrix[size_t(t->n + p2) * t->n];
          rows.insert(rows.end(), er, er + t->n);
        }
        for (int az = 0; az < t->az_count; az++) {
          auto idx = local_stripe(*t, az);
          for (int lp = 0; lp < c->local_m; lp++) {
            std::vector<uint8_t> row(t->n, 0);
            const uint8_t *lr =
                &c->local_matrix[size_t(c->local_n + lp) * c->local_n];
            for (int j = 0; j < c->local_n; j++) {
              const int g2i = idx[j];
              if (g2i < t->n) {
                row[g2i] ^= lr[j];
              } else { /* global parity input */
                const uint8_t *er = &c->enc_matrix[size_t(g2i) * t->n];
                for (int d = 0; d < t->n; d++)
                  row[d] ^= gt2.mul[lr[j]][er[d]];
              }
            }
            out.push_back(t->n + t->m + az * c->local_m + lp);
            rows.insert(rows.end(), row.begin(), row.end());
          }
        }
        ok = c->fused_lrc.upload(in, out, rows, c->stream) == GFRS_OK;
        c->fused_lrc_ok = ok;
      }
    }
  }
  if (prev >= 0) hipSetDevice(prev);
  if (!ok) {
    seterr("gfrs_create: init failed (%s)", g_err.c_str());
    delete c;
    return nullptr;
  }
  return reinterpret_cast<gfrs_ctx *>(c);
}

void gfrs_destroy(gfrs_ctx *ctx) { delete reinterpret_cast<gfrs_ctx_impl *>(ctx); }

int gfrs_set_stream(gfrs_ctx *ctx, void *hip_stream) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  std::lock_guard<std::mutex> lk(c->mu);
  c->stream = hip_stream ? reinterpret_cast<hipStream_t>(hip_stream)
                         : c->own_stream;
  /* a caller-pinned stream implies caller-managed ordering: foreground
   * calls then issue on that one stream instead of fanning out */
  c->user_stream = hip_stream != nullptr;
  return GFRS_OK;
}

int gfrs_synchronize(gfrs_ctx *ctx) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  StreamGuard g(c);
  HIP_TRY(hipStreamSynchronize(c->stream));
  for (auto &L : c->lanes) HIP_TRY(hipStreamSynchronize(L->stream));
  return GFRS_OK;
}

int gfrs_encode_matrix(gfrs_ctx *ctx, uint8_t *out) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  memcpy(out, c->enc_matrix.data(), c->enc_matrix.size());
  return GFRS_OK;
}

int gfrs_compute_encode_matrix(int k, int total, uint8_t *out) {
  if (!gf_build_matrix(k, total, out)) {
    seterr("encode matrix build failed (k=%d total=%d)", k, total);
    return GFRS_ERR_SINGULAR;
  }
  return GFRS_OK;
}

int gfrs_probe_perm(void) {
  if (gfrs_device_count() == 0) return GFRS_ERR_NO_GPU;
  return probe_perm_device();
}

}  /* extern "C" (continued in this file below) */

/* ---------------- internal helpers for the compute entry points ------- */

namespace gfrs {

/* Upload a pointer table for pointer-mode launches (stream-ordered, so
 * reuse of the scratch buffer is safe across calls on one stream). */
static int upload_ptrs(const LaneView &lv, void *const *shards, int nshards) {
  int rc = lv.ptr_buf->ensure(size_t(nshards) * 8);
  if (rc != GFRS_OK) return rc;
  /* small, use pinned staging for async copy */
  if ((rc = lv.stage_pin->ensure(size_t(nshards) * 8)) != GFRS_OK) return rc;
  memcpy(lv.stage_pin->p, shards, size_t(nshards) * 8);
  HIP_TRY(hipMemcpyAsync(lv.ptr_buf->p, lv.stage_pin->p, size_t(nshards) * 8,
                         hipMemcpyHostToDevice, lv.stream));
  return GFRS_OK;
}

/* decode plan for a missing pattern (bitmask over all shards of the
 * engine's k+m space).  Mirrors inversion_tree caching. */
static int get_decode_plan(gfrs_ctx_impl *c, int k, int m,
                           const std::vector<uint8_t> &matrix,
                           const std::vector<int> &engine_idx /* global ids */,
                           const std::vector<uint8_t> &present,
                           DevPlan **out_plan, int tag, hipStream_t s) {
  std::lock_guard<std::mutex> cache_lk(c->cache_mu);
  /* key = missing bitmask + engine tag (global=1, az-local=2+az, ...):
   * global and local engines can share k, first index AND mask, so the
   * tag is load-bearing (a collision here once wrote a local parity row
   * over a global one — caught by the randomized fuzz test) */
  PlanKey key = plan_key_tag(uint64_t(tag));
  for (int i = 0; i < k + m; i++)
    if (!present[i]) plan_key_set(key, i);
  auto it = c->dec_cache.find(key);
  if (it != c->dec_cache.end()) {
    *out_plan = it->second;
    return GFRS_OK;
  }
  /* build: first k valid rows (reedsolomon.go:1453-1466) */
  std::vector<int> valid;
  for (int i = 0; i < k + m && int(valid.size()) < k; i++)
    if (present[i]) valid.push_back(i);
  if (int(valid.size()) < k) return GFRS_ERR_TOO_FEW_SHARDS;
  std::vector<uint8_t> sub(size_t(k) * k), inv(size_t(k) * k);
  for (int r = 0; r < k; r++)
    memcpy(&sub[size_t(r) * k], &matrix[size_t(valid[r]) * k], k);
  if (!gf_invert(sub.data(), k, inv.data())) return GFRS_ERR_SINGULAR;

  std::vector<int32_t> in, out;
  std::vector<uint8_t> rows;
  for (int r = 0; r < k; r++) in.push_back(engine_idx[valid[r]]);
  for (int i = 0; i < k; i++)
    if (!present[i]) {
      out.push_back(engine_idx[i]);
      rows.insert(rows.end(), &inv[size_t(i) * k], &inv[size_t(i) * k + k]);
    }
  auto *p = new DevPlan();
  int rc = p->upload(in, out, rows, s);
  if (rc != GFRS_OK) {
    delete p;
    return rc;
  }
  c->dec_cache[key] = p;
  *out_plan = p;
  return GFRS_OK;
}

/* parity-regeneration plan for missing parity rows */
static int get_parity_plan(gfrs_ctx_impl *c, int k, int m,
                           const std::vector<uint8_t> &matrix,
                           const std::vector<int> &engine_idx,
                           const std::vector<uint8_t> &present,
                           DevPlan **out_plan, int tag, hipStream_t s) {
  std::lock_guard<std::mutex> cache_lk(c->cache_mu);
  PlanKey key = plan_key_tag(uint64_t(tag));
  for (int i = 0; i < k + m; i++)
    if (!present[i]) plan_key_set(key, i);
  auto it = c->par_cache.find(key);
  if (it != c->par_cache.end()) {
    *out_plan = it->second;
    return GFRS_OK;
  }
  std::vector<int32_t> in, out;
  std::vector<uint8_t> rows;
  for (int i = 0; i < k; i++) in.push_back(engine_idx[i]);
  for (int i = k; i < k + m; i++)
    if (!present[i]) {
      out.push_back(engine_idx[i]);
      rows.insert(rows.end(), &matrix[size_t(i) * k],
                  &matrix[size_t(i) * k + k]);
    }
  if (out.empty()) {
    *out_plan = nullptr;
    return GFRS_OK;
  }
  auto *p = new DevPlan();
  int rc = p->upload(in, out, rows, s);
  if (rc != GFRS_OK) {
    delete p;
    return rc;
  }
  c->par_cache[key] = p;
  *out_plan = p;
  return GFRS_OK;
}

/* Run one engine's reconstruct over pointer-mode shards.
 * engine_idx maps engine-local 0..k+m-1 to global shard slots. */
static int reconstruct_engine(gfrs_ctx_impl *c, const LaneView &lv, int k,
                              int m, const std::vector<uint8_t> &matrix,
                              const std::vector<int> &engine_idx,
                              const std::vector<uint8_t> &present,
                              size_t shard_len, int nstripes, int data_only,
                              uint64_t strided_base, uint64_t stripe_stride,
                              int nptr_or_0, int tag) {
  int npresent = 0, dpresent = 0;
  for (int i = 0; i < k + m; i++)
    if (present[i]) {
      npresent++;
      if (i < k) dpresent++;
    }
  if (npresent == k + m || (data_only && dpresent == k)) return GFRS_OK;
  if (npresent < k) return GFRS_ERR_TOO_FEW_SHARDS;

  DevPlan *dec = nullptr;
  int rc = GFRS_OK;
  bool have_missing_data = dpresent < k;
  if (have_missing_data) {
    rc = get_decode_plan(c, k, m, matrix, engine_idx, present, &dec, tag,
                         lv.stream);
    if (rc != GFRS_OK) return rc;
    if (nptr_or_0 > 0)
      launch_rs_apply(reinterpret_cast<const uint64_t *>(lv.ptr_buf->p),
                      nptr_or_0, (const int32_t *)dec->in_idx.p, dec->k,
                      (const int32_t *)dec->out_idx.p, dec->nout,
                      (const uint8_t *)dec->tabs.p, shard_len, nstripes,
                      lv.stream);
    else
      launch_rs_apply_strided(strided_base, stripe_stride,
                              (const int32_t *)dec->in_idx.p, dec->k,
                              (const int32_t *)dec->out_idx.p, dec->nout,
                              (const uint8_t *)dec->tabs.p, shard_len,
                              nstripes, lv.stream);
  }
  if (!data_only) {
    DevPlan *par = nullptr;
    rc = get_parity_plan(c, k, m, matrix, engine_idx, present, &par, tag,
                         lv.stream);
    if (rc != GFRS_OK) return rc;
    if (par) {
      if (nptr_or_0 > 0)
        launch_rs_apply(reinterpret_cast<const uint64_t *>(lv.ptr_buf->p),
                        nptr_or_0, (const int32_t *)par->in_idx.p, par->k,
                        (const int32_t *)par->out_idx.p, par->nout,
                        (const uint8_t *)par->tabs.p, shard_len, nstripes,
                        lv.stream);
      else
        launch_rs_apply_strided(strided_base, stripe_stride,
                                (const int32_t *)par->in_idx.p, par->k,
                                (const int32_t *)par->out_idx.p, par->nout,
                                (const uint8_t *)par->tabs.p, shard_len,
                                nstripes, lv.stream);
    }
  }
  return GFRS_OK;
}

/* Core reconstruct across global + local engines.  present covers all
 * n+m+l global slots. */
static int reconstruct_all(gfrs_ctx_impl *c, const LaneView &lv,
                           std::vector<uint8_t> &present, size_t shard_len,
                           int nstripes, int data_only, uint64_t base,
                           uint64_t stride, int nptr_or_0) {
  const gfrs_tactic &t = c->t;
  std::vector<int> gidx(t.n + t.m);
  for (int i = 0; i < t.n + t.m; i++) gidx[i] = i;
  std::vector<uint8_t> gpresent(present.begin(), present.begin() + t.n + t.m);
  int rc = reconstruct_engine(c, lv, t.n, t.m, c->enc_matrix, gidx, gpresent,
                              shard_len, nstripes, data_only, base, stride,
                              nptr_or_0, /*tag=*/1);
  if (rc != GFRS_OK) return rc;
  if (t.l == 0 || data_only) return GFRS_OK;
  /* regenerate bad local parities per AZ (lrcencoder.go:160-184); after
   * the global pass everything below n+m is intact */
  for (int az = 0; az < t.az_count; az++) {
    auto idx = local_stripe(t, az);
    bool need = false;
    std::vector<uint8_t> lp(idx.size());
    std::vector<int> li(idx.begin(), idx.end());
    for (size_t i = 0; i < idx.size(); i++) {
      lp[i] = idx[i] < t.n + t.m ? 1 : present[idx[i]];
      if (!lp[i]) need = true;
    }
    if (!need) continue;
    rc = reconstruct_engine(c, lv, c->local_n, c->local_m, c->local_matrix,
                            li, lp, shard_len, nstripes, /*data_only=*/0,
                            base, stride, nptr_or_0, /*tag=*/2 + az);
    if (rc != GFRS_OK) return rc;
  }
  return GFRS_OK;
}

}  // namespace gfrs

/* ---------------- compute entry points ---------------- */

extern "C" {

int gfrs_encode(gfrs_ctx *ctx, void *const *shards, size_t shard_len,
                int nshards, int memloc) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  if (c->t.m == 0) return nshards == c->total ? GFRS_OK : GFRS_ERR_INVALID_SHARDS;
  if (nshards != c->total) {
    seterr("encode: want %d shards, got %d", c->total, nshards);
    return GFRS_ERR_INVALID_SHARDS;
  }
  /* foreground call: fan out over the stream-lane pool so concurrent
   * callers on one context overlap on the GPU (the reference shares one
   * encoder across ~100 goroutines, encoder.go:29) */
  Lane *L = c->pick_lane();
  std::unique_lock<std::mutex> lk(L ? L->mu : c->mu);
  StreamGuard g(c);
  const LaneView lv = L ? view_of(L) : view_of(c);
  int rc;
  if (memloc == GFRS_MEM_HOST) {
    /* stage contiguous stripe, run strided, copy parity back */
    size_t tot = size_t(c->total) * shard_len;
    if ((rc = lv.stage_dev->ensure(tot)) != GFRS_OK) return rc;
    if ((rc = lv.stage_pin->ensure(tot)) != GFRS_OK) return rc;
    uint8_t *pin = (uint8_t *)lv.stage_pin->p;
    for (int i = 0; i < c->t.n; i++)
      memcpy(pin + size_t(i) * shard_len, shards[i], shard_len);
    uint8_t *dev = (uint8_t *)lv.stage_dev->p;
    HIP_TRY(hipMemcpyAsync(dev, pin, size_t(c->t.n) * shard_len,
                           hipMemcpyHostToDevice, lv.stream));
    rc = encode_batch_impl(c, dev, shard_len, tot, 1, lv.stream);
    if (rc != GFRS_OK) return rc;
    HIP_TRY(hipMemcpyAsync(pin + size_t(c->t.n) * shard_len,
                           dev + size_t(c->t.n) * shard_len,
                           size_t(c->t.m + c->t.l) * shard_len,
                           hipMemcpyDeviceToHost, lv.stream));
    HIP_TRY(hipStreamSynchronize(lv.stream));
    for (int i = c->t.n; i < c->total; i++)
      memcpy(shards[i], pin + size_t(i) * shard_len, shard_len);
    return GFRS_OK;
  }
  /* ec.Buffer lays shards out contiguously (buf.go:24-35): detect that
   * and take the strided path — no pointer-table upload, one less
   * dependency on the foreground latency path */
  bool contig = true;
  for (int i = 1; i < nshards && contig; i++)
    contig = (const uint8_t *)shards[i] ==
             (const uint8_t *)shards[0] + size_t(i) * shard_len;
  if (contig) {
    rc = encode_batch_impl(c, shards[0], shard_len,
                           size_t(nshards) * shard_len, 1, lv.stream);
    if (rc != GFRS_OK) return rc;
    HIP_TRY(hipStreamSynchronize(lv.stream));
    return GFRS_OK;
  }
  if ((rc = upload_ptrs(lv, shards, nshards)) != GFRS_OK) return rc;
  launch_rs_apply((const uint64_t *)lv.ptr_buf->p, c->total,
                  (const int32_t *)c->enc_plan.in_idx.p, c->enc_plan.k,
                  (const int32_t *)c->enc_plan.out_idx.p, c->enc_plan.nout,
                  (const uint8_t *)c->enc_plan.tabs.p, shard_len, 1,
                  lv.stream);
  for (auto *lp : c->local_enc)
    launch_rs_apply((const uint64_t *)lv.ptr_buf->p, c->total,
                    (const int32_t *)lp->in_idx.p, lp->k,
                    (const int32_t *)lp->out_idx.p, lp->nout,
                    (const uint8_t *)lp->tabs.p, shard_len, 1, lv.stream);
  HIP_TRY(hipStreamSynchronize(lv.stream));
  return GFRS_OK;
}

/* unlocked bodies: public entry points take c->mu (the reference encoder
 * is share-safe behind its limiter, encoder.go:29,115) */
static int encode_batch_impl(gfrs_ctx_impl *c, void *base, size_t shard_len,
                             size_t stripe_stride, int nstripes,
                             hipStream_t st) {
  StreamGuard g(c);
  launch_rs_apply_strided((uint64_t)base, stripe_stride,
                          (const int32_t *)c->enc_plan.in_idx.p,
                          c->enc_plan.k, (const int32_t *)c->enc_plan.out_idx.p,
                          c->enc_plan.nout, (const uint8_t *)c->enc_plan.tabs.p,
                          shard_len, nstripes, st);
  for (auto *lp : c->local_enc)
    launch_rs_apply_strided((uint64_t)base, stripe_stride,
                            (const int32_t *)lp->in_idx.p, lp->k,
                            (const int32_t *)lp->out_idx.p, lp->nout,
                            (const uint8_t *)lp->tabs.p, shard_len, nstripes,
                            st);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return hip_fail("encode_batch launch", e);
  return GFRS_OK;
}

int gfrs_encode_batch(gfrs_ctx *ctx, void *base, size_t shard_len,
                      size_t stripe_stride, int nstripes) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  std::lock_guard<std::mutex> lk(c->mu);
  return encode_batch_impl(c, base, shard_len, stripe_stride, nstripes,
                           c->stream);
}

int gfrs_verify(gfrs_ctx *ctx, void *const *shards, size_t shard_len,
                int nshards, int memloc, int *ok) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  if (c->t.m == 0) { /* zero parity rows: vacuously true (reedsolomon.go:784) */
    *ok = 1;
    return nshards == c->total ? GFRS_OK : GFRS_ERR_INVALID_SHARDS;
  }
  bool local_form = c->t.l > 0 && nshards == c->total / c->t.az_count;
  if (nshards != c->total && !local_form) return GFRS_ERR_INVALID_SHARDS;
  Lane *L = c->pick_lane();
  std::unique_lock<std::mutex> lk(L ? L->mu : c->mu);
  StreamGuard g(c);
  const LaneView lv = L ? view_of(L) : view_of(c);
  if (local_form) {
    /* one local stripe set (lrcencoder.go:94-99) */
    int rc;
    if (memloc == GFRS_MEM_HOST) {
      size_t tot = size_t(nshards) * shard_len;
      if ((rc = lv.stage_dev->ensure(tot)) != GFRS_OK) return rc;
      if ((rc = lv.stage_pin->ensure(tot)) != GFRS_OK) return rc;
      uint8_t *pin = (uint8_t *)lv.stage_pin->p;
      for (int i = 0; i < nshards; i++)
        memcpy(pin + size_t(i) * shard_len, shards[i], shard_len);
      HIP_TRY(hipMemcpyAsync(lv.stage_dev->p, pin, tot,
                             hipMemcpyHostToDevice, lv.stream));
      if ((rc = lv.fail_buf->ensure(4)) != GFRS_OK) return rc;
      HIP_TRY(hipMemsetAsync(lv.fail_buf->p, 0, 4, lv.stream));
      launch_rs_verify_strided((uint64_t)lv.stage_dev->p, tot,
                               (const int32_t *)c->local_plan.in_idx.p,
                               c->local_plan.k,
                               (const int32_t *)c->local_plan.out_idx.p,
                               c->local_plan.nout,
                               (const uint8_t *)c->local_plan.tabs.p,
                               shard_len, 1, (uint32_t *)lv.fail_buf->p,
                               lv.stream);
    } else {
      if ((rc = upload_ptrs(lv, shards, nshards)) != GFRS_OK) return rc;
      if ((rc = lv.fail_buf->ensure(4)) != GFRS_OK) return rc;
      HIP_TRY(hipMemsetAsync(lv.fail_buf->p, 0, 4, lv.stream));
      launch_rs_verify((const uint64_t *)lv.ptr_buf->p, nshards,
                       (const int32_t *)c->local_plan.in_idx.p,
                       c->local_plan.k,
                       (const int32_t *)c->local_plan.out_idx.p,
                       c->local_plan.nout,
                       (const uint8_t *)c->local_plan.tabs.p, shard_len, 1,
                       (uint32_t *)lv.fail_buf->p, lv.stream);
    }
    uint32_t fail = 0;
    HIP_TRY(hipMemcpyAsync(&fail, lv.fail_buf->p, 4, hipMemcpyDeviceToHost,
                           lv.stream));
    HIP_TRY(hipStreamSynchronize(lv.stream));
    *ok = fail == 0;
    return GFRS_OK;
  }
  int rc;
  const uint64_t *pt;
  if (memloc == GFRS_MEM_HOST) {
    size_t tot = size_t(c->total) * shard_len;
    if ((rc = lv.stage_dev->ensure(tot + 64)) != GFRS_OK) return rc;
    if ((rc = lv.stage_pin->ensure(tot)) != GFRS_OK) return rc;
    uint8_t *pin = (uint8_t *)lv.stage_pin->p;
    for (int i = 0; i < c->total; i++)
      memcpy(pin + size_t(i) * shard_len, shards[i], shard_len);
    HIP_TRY(hipMemcpyAsync(lv.stage_dev->p, pin, tot, hipMemcpyHostToDevice,
                           lv.stream));
    uint64_t fb = 0;
    rc = verify_batch_impl(c, lv.stage_dev->p, shard_len, tot, 1, &fb,
                           lv.stream, lv.fail_buf);
    if (rc != GFRS_OK) return rc;
    *ok = fb == 0;
    return GFRS_OK;
  }
  if ((rc = upload_ptrs(lv, shards, nshards)) != GFRS_OK) return rc;
  if ((rc = lv.fail_buf->ensure(4)) != GFRS_OK) return rc;
  HIP_TRY(hipMemsetAsync(lv.fail_buf->p, 0, 4, lv.stream));
  pt = (const uint64_t *)lv.ptr_buf->p;
  launch_rs_verify(pt, c->total, (const int32_t *)c->enc_plan.in_idx.p,
                   c->enc_plan.k, (const int32_t *)c->enc_plan.out_idx.p,
                   c->enc_plan.nout, (const uint8_t *)c->enc_plan.tabs.p,
                   shard_len, 1, (uint32_t *)lv.fail_buf->p, lv.stream);
  for (auto *lp : c->local_enc)
    launch_rs_verify(pt, c->total, (const int32_t *)lp->in_idx.p, lp->k,
                     (const int32_t *)lp->out_idx.p, lp->nout,
                     (const uint8_t *)lp->tabs.p, shard_len, 1,
                     (uint32_t *)lv.fail_buf->p, lv.stream);
  uint32_t fail = 0;
  HIP_TRY(hipMemcpyAsync(&fail, lv.fail_buf->p, 4, hipMemcpyDeviceToHost,
                         lv.stream));
  HIP_TRY(hipStreamSynchronize(lv.stream));
  *ok = fail == 0;
  return GFRS_OK;
}

static int verify_batch_impl(gfrs_ctx_impl *c, const void *base,
                             size_t shard_len, size_t stripe_stride,
                             int nstripes, uint64_t *fail_bitmap,
                             hipStream_t st, DevBuf *fail) {
  StreamGuard g(c);
  int rc;
  if ((rc = fail->ensure(size_t(nstripes) * 4)) != GFRS_OK) return rc;
  HIP_TRY(hipMemsetAsync(fail->p, 0, size_t(nstripes) * 4, st));
  launch_rs_verify_strided((uint64_t)base, stripe_stride,
                           (const int32_t *)c->enc_plan.in_idx.p,
                           c->enc_plan.k,
                           (const int32_t *)c->enc_plan.out_idx.p,
                           c->enc_plan.nout,
                           (const uint8_t *)c->enc_plan.tabs.p, shard_len,
                           nstripes, (uint32_t *)fail->p, st);
  for (auto *lp : c->local_enc)
    launch_rs_verify_strided((uint64_t)base, stripe_stride,
                             (const int32_t *)lp->in_idx.p, lp->k,
                             (const int32_t *)lp->out_idx.p, lp->nout,
                             (const uint8_t *)lp->tabs.p, shard_len, nstripes,
                             (uint32_t *)fail->p, st);
  std::vector<uint32_t> fails(nstripes);
  HIP_TRY(hipMemcpyAsync(fails.data(), fail->p, size_t(nstripes) * 4,
                         hipMemcpyDeviceToHost, st));
  HIP_TRY(hipStreamSynchronize(st));
  if (fail_bitmap) {
    memset(fail_bitmap, 0, ((nstripes + 63) / 64) * 8);
    for (int s = 0; s < nstripes; s++)
      if (fails[s]) fail_bitmap[s / 64] |= 1ull << (s % 64);
  }
  return GFRS_OK;
}

int gfrs_verify_batch(gfrs_ctx *ctx, const void *base, size_t shard_len,
                      size_t stripe_stride, int nstripes,
                      uint64_t *fail_bitmap) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  std::lock_guard<std::mutex> lk(c->mu);
  return verify_batch_impl(c, base, shard_len, stripe_stride, nstripes,
                           fail_bitmap, c->stream, &c->fail_buf);
}

int gfrs_reconstruct(gfrs_ctx *ctx, void *const *shards, size_t shard_len,
                     int nshards, int memloc, const int32_t *bad_idx,
                     int nbad, int data_only) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  const gfrs_tactic &t = c->t;
  Lane *L = c->pick_lane();
  std::unique_lock<std::mutex> lk(L ? L->mu : c->mu);
  StreamGuard g(c);
  const LaneView lv = L ? view_of(L) : view_of(c);
  int rc;

  if (t.m == 0) /* no parity: any missing shard is unrecoverable */
    return nbad == 0 ? GFRS_OK : GFRS_ERR_TOO_FEW_SHARDS;
  /* full set or, for LRC, a single local stripe (lrcencoder.go:147-153) */
  bool local_form = t.l > 0 && nshards == c->total / t.az_count;
  if (nshards != c->total && !local_form) return GFRS_ERR_INVALID_SHARDS;

  std::vector<uint8_t> present(nshards, 1);
  for (int i = 0; i < nbad; i++) {
    if (bad_idx[i] < 0 || bad_idx[i] >= nshards) return GFRS_ERR_INVALID_SHARDS;
    present[bad_idx[i]] = 0;
  }

  if (memloc == GFRS_MEM_HOST) {
    /* stage the whole stripe contiguously and run in strided mode */
    size_t tot = size_t(nshards) * shard_len;
    if ((rc = lv.stage_dev->ensure(tot)) != GFRS_OK) return rc;
    if ((rc = lv.stage_pin->ensure(tot)) != GFRS_OK) return rc;
    uint8_t *pin = (uint8_t *)lv.stage_pin->p;
    uint8_t *dev = (uint8_t *)lv.stage_dev->p;
    for (int i = 0; i < nshards; i++)
      if (present[i]) memcpy(pin + size_t(i) * shard_len, shards[i], shard_len);
    HIP_TRY(hipMemcpyAsync(dev, pin, tot, hipMemcpyHostToDevice, lv.stream));
    if (local_form) {
      std::vector<int> li(nshards);
      for (int i = 0; i < nshards; i++) li[i] = i;
      rc = reconstruct_engine(c, lv, c->local_n, c->local_m, c->local_matrix,
                              li, present, shard_len, 1, data_only,
                              (uint64_t)dev, tot, 0, /*tag=*/50);
    } else {
      rc = reconstruct_all(c, lv, present, shard_len, 1, data_only,
                           (uint64_t)dev, tot, 0);
    }
    if (rc != GFRS_OK) return rc;
    HIP_TRY(hipMemcpyAsync(pin, dev, tot, hipMemcpyDeviceToHost, lv.stream));
    HIP_TRY(hipStreamSynchronize(lv.stream));
    /* copy back only shards actually rebuilt: with data_only the engine
     * leaves missing parity untouched (ReconstructData semantics,
     * reedsolomon.go:1441-1444) — overwriting the caller's buffer with
     * stale staging bytes would be wrong */
    const int ndata = local_form ? c->local_n : t.n;
    for (int i = 0; i < nshards; i++)
      if (!present[i] && (!data_only || i < ndata))
        memcpy(shards[i], pin + size_t(i) * shard_len, shard_len);
    return GFRS_OK;
  }

  if ((rc = upload_ptrs(lv, shards, nshards)) != GFRS_OK) return rc;

  if (local_form) {
    std::vector<int> li(nshards);
    for (int i = 0; i < nshards; i++) li[i] = i;
    rc = reconstruct_engine(c, lv, c->local_n, c->local_m, c->local_matrix,
                            li, present, shard_len, 1, data_only, 0, 0,
                            nshards, /*tag=*/50);
  } else {
    rc = reconstruct_all(c, lv, present, shard_len, 1, data_only, 0, 0,
                         nshards);
  }
  if (rc != GFRS_OK) return rc;
  HIP_TRY(hipStreamSynchronize(lv.stream));
  return GFRS_OK;
}

int gfrs_reconstruct_batch(gfrs_ctx *ctx, void *base, size_t shard_len,
                           size_t stripe_stride, int nstripes,
                           const int32_t *bad_idx, int nbad, int data_only) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  std::lock_guard<std::mutex> lk(c->mu);
  StreamGuard g(c);
  std::vector<uint8_t> present(c->total, 1);
  for (int i = 0; i < nbad; i++) {
    if (bad_idx[i] < 0 || bad_idx[i] >= c->total)
      return GFRS_ERR_INVALID_SHARDS;
    present[bad_idx[i]] = 0;
  }
  int rc = reconstruct_all(c, view_of(c), present, shard_len, nstripes,
                           data_only, (uint64_t)base, stripe_stride, 0);
  if (rc != GFRS_OK) return rc;
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return hip_fail("reconstruct_batch launch", e);
  return GFRS_OK;
}

int gfrs_encode_frame_batch(gfrs_ctx *ctx, void *framed,
                            size_t framed_stride, void *base,
                            size_t shard_len, size_t stripe_stride,
                            int nstripes, int64_t block_len) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  const gfrs_tactic &t = c->t;
  if (block_len <= 0 || block_len % 4096) return GFRS_ERR_INVALID_BLOCK;
  if (nstripes <= 0 || shard_len == 0) return GFRS_ERR_INVALID_SHARDS;
  std::lock_guard<std::mutex> lk(c->mu);
  StreamGuard g(c);
  /* the fused kernel gives a workgroup a whole 64 KiB frame; below a few
   * frames per shard the two-kernel composition (whose rs_apply packs
   * small stripes per tile) is the faster shape.  LRC runs fused through
   * the composed plan: every global AND local parity is a row over the n
   * data shards. */
  const int gm_all = t.m + t.l;
  /* Measured crossover (pipelined fused kernel, RS(6+3), fixed 4 GiB
   * source): fused wins from 8 KiB shards up (837 vs 306 GiB/s at 8 KiB,
   * ~1180 vs ~780 at 16 KiB-8 MiB); the two-kernel path with rs_apply
   * stripe packing wins at 2-4 KiB (684 vs 320 at 2 KiB).
   * GFRS_FUSED_MIN overrides the threshold (bytes); shards
   * <= 4096 take the wave-per-stripe small kernel, and at
   * 5 KiB the big fused kernel beats two-kernel 622 vs 392. */
  static const size_t fused_min = []() {
    const char *e = getenv("GFRS_FUSED_MIN");
    const long v = e ? atol(e) : 0;
    return v > 0 ? size_t(v) : size_t(4097);
  }();
  const bool shapes_ok = block_len == 65536 && t.m >= 1 && gm_all <= 4 &&
                         t.n + gm_all <= 16 && framed_stride % 4 == 0 &&
                         (t.l == 0 || c->fused_lrc_ok);
  /* wave-per-stripe crossover (measured r02): the small kernel wins the
   * whole 4-8 KiB band (5K 1353 / 6K 1427 / 7K 1056 / 8K 863 vs
   * 514-833 GiB/s for the workgroup-per-frame form) */
  static const size_t small_max = []() {
    const char *e = getenv("GFRS_SMALL_MAX");
    const long v = e ? atol(e) : 0;
    return v > 0 ? size_t(v) : size_t(8192);
  }();
  const bool small_ok =
      shard_len <= 4096 || (shard_len <= small_max && gm_all <= 3);
  if (shapes_ok && small_ok) {
    /* MinShardSize-class shapes: wave-per-stripe fused kernel */
    const DevPlan &pl = t.l == 0 ? c->enc_plan : c->fused_lrc;
    launch_rs_encode_frame_small((uint8_t *)framed, framed_stride,
                                 (uint64_t)base, stripe_stride, shard_len,
                                 t.n, gm_all, (const uint8_t *)pl.tabs.p,
                                 nstripes, c->stream);
    hipError_t e = hipGetLastError();
    if (e != hipSuccess) return hip_fail("encode_frame_small launch", e);
    return GFRS_OK;
  }
  const bool fused = shapes_ok && shard_len >= fused_min;
  if (fused) {
    const DevPlan &pl = t.l == 0 ? c->enc_plan : c->fused_lrc;
    launch_rs_encode_frame((uint8_t *)framed, framed_stride, (uint64_t)base,
                           stripe_stride, shard_len, t.n, gm_all,
                           (const uint8_t *)pl.tabs.p, nstripes, c->stream);
    hipError_t e = hipGetLastError();
    if (e != hipSuccess) return hip_fail("encode_frame launch", e);
    return GFRS_OK;
  }
  /* fallback composition: needs the contiguous ec.Buffer batch layout */
  if (stripe_stride != size_t(c->total) * shard_len)
    return GFRS_ERR_UNSUPPORTED;
  int rc = encode_batch_impl(c, base, shard_len, stripe_stride, nstripes,
                             c->stream);
  if (rc != GFRS_OK) return rc;
  return crc32b_encode_batch_impl(c, framed, framed_stride, base, shard_len,
                                  int64_t(shard_len), block_len,
                                  nstripes * c->total);
}

/* ---------------- sized coder (rpc2 body framing) ---------------- */

int gfrs_sized_encode_size(int64_t actual_size, int64_t block_len,
                           int64_t *size, int64_t *tail) {
  /* PartialEncodeSizeWith with stableSize == 0 (util.go:73-80),
   * _alignment = 512 (rpc2/transport/allocator.go:12-15) */
  int64_t enc = gfrs_crc32b_encode_size(actual_size, block_len);
  if (enc < 0) return int(enc);
  int64_t t = (512 - (enc & 511)) & 511;
  *size = enc + t;
  *tail = t;
  return GFRS_OK;
}

int64_t gfrs_sized_decode_size(int64_t total, int64_t tail,
                               int64_t block_len) {
  /* PartialDecodeSizeWith, stableSize == 0 (util.go:86-94) */
  return gfrs_crc32b_decode_size(total - tail, block_len);
}

int64_t gfrs_sized_encode(gfrs_ctx *ctx, void *dst, const void *src,
                          int64_t n, int64_t block_len) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  if (block_len <= 0 || block_len % 4096) return GFRS_ERR_INVALID_BLOCK;
  if (n <= 0) return GFRS_ERR_INVALID_SHARDS;
  int64_t size = 0, tail = 0;
  int rc = gfrs_sized_encode_size(n, block_len, &size, &tail);
  if (rc != GFRS_OK) return rc;
  std::lock_guard<std::mutex> lk(c->mu);
  StreamGuard g(c);
  launch_sized_encode((uint8_t *)dst, 0, (const uint8_t *)src, 0, n,
                      block_len, 1, c->stream);
  if (tail)
    HIP_TRY(hipMemsetAsync((uint8_t *)dst + size - tail, 0, size_t(tail),
                           c->stream));
  HIP_TRY(hipStreamSynchronize(c->stream));
  return size;
}

int gfrs_sized_verify(gfrs_ctx *ctx, const void *framed, int64_t total,
                      int64_t tail, int64_t block_len, int64_t *bad_block) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  if (block_len <= 0 || block_len % 4096) return GFRS_ERR_INVALID_BLOCK;
  std::lock_guard<std::mutex> lk(c->mu);
  StreamGuard g(c);
  int rc;
  if ((rc = c->fail_buf.ensure(8)) != GFRS_OK) return rc;
  int64_t bad = INT64_MAX;
  HIP_TRY(hipMemcpyAsync(c->fail_buf.p, &bad, 8, hipMemcpyHostToDevice,
                         c->stream));
  launch_sized_verify((const uint8_t *)framed, 0, total - tail, block_len, 1,
                      (int64_t *)c->fail_buf.p, c->stream);
  HIP_TRY(hipMemcpyAsync(&bad, c->fail_buf.p, 8, hipMemcpyDeviceToHost,
                         c->stream));
  HIP_TRY(hipStreamSynchronize(c->stream));
  *bad_block = bad == INT64_MAX ? -1 : bad;
  return GFRS_OK;
}

int64_t gfrs_sized_decode(gfrs_ctx *ctx, void *dst, const void *framed,
                          int64_t total, int64_t tail, int64_t block_len) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  if (block_len <= 0 || block_len % 4096) return GFRS_ERR_INVALID_BLOCK;
  std::lock_guard<std::mutex> lk(c->mu);
  StreamGuard g(c);
  int rc;
  if ((rc = c->fail_buf.ensure(8)) != GFRS_OK) return rc;
  int64_t bad = INT64_MAX;
  HIP_TRY(hipMemcpyAsync(c->fail_buf.p, &bad, 8, hipMemcpyHostToDevice,
                         c->stream));
  launch_sized_decode((uint8_t *)dst, 0, (const uint8_t *)framed, 0,
                      total - tail, block_len, 1, (int64_t *)c->fail_buf.p,
                      c->stream);
  HIP_TRY(hipMemcpyAsync(&bad, c->fail_buf.p, 8, hipMemcpyDeviceToHost,
                         c->stream));
  HIP_TRY(hipStreamSynchronize(c->stream));
  if (bad != INT64_MAX) return GFRS_ERR_MISMATCHED_CRC;
  return gfrs_sized_decode_size(total, tail, block_len);
}

/* ---------------- blobnode shard images ---------------- */

/* (the 32 B headers - magic, BE ids, CRC - are built on device by
 * shard_finalize_k; see shard.go:241-261 for the format) */

int64_t gfrs_shard_disk_size(int64_t size, int64_t block_len) {
  int64_t body = gfrs_crc32b_encode_size(size, block_len);
  if (body < 0) return body;
  return 32 + body + 8; /* header + framed body + footer */
}

int gfrs_shard_write_batch(gfrs_ctx *ctx, void *dst, size_t dst_stride,
                           const void *src, size_t src_stride, int64_t size,
                           int64_t block_len, const uint64_t *bids,
                           const uint64_t *vuids, int nshards) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  if (block_len <= 0 || block_len % 4096) return GFRS_ERR_INVALID_BLOCK;
  if (size <= 0 || nshards <= 0) return GFRS_ERR_INVALID_SHARDS;
  std::lock_guard<std::mutex> lk(c->mu);
  StreamGuard g(c);
  int rc;
  /* ship the raw id arrays; the finalize kernel builds the 32 B headers
   * (shard.go:241-261) on device - host-side construction was the
   * bottleneck at millions of shards per call */
  const size_t idbytes = size_t(nshards) * 8;
  if ((rc = c->stage_pin.ensure(idbytes * 2)) != GFRS_OK) return rc;
  memcpy(c->stage_pin.p, bids, idbytes);
  memcpy((uint8_t *)c->stage_pin.p + idbytes, vuids, idbytes);
  DevBuf &hbuf = c->ptr_buf; /* reuse scratch (stream-ordered) */
  if ((rc = hbuf.ensure(idbytes * 2)) != GFRS_OK) return rc;
  HIP_TRY(hipMemcpyAsync(hbuf.p, c->stage_pin.p, idbytes * 2,
                         hipMemcpyHostToDevice, c->stream));
  /* framed body at +32 */
  launch_crc_encode((uint8_t *)dst + 32, dst_stride, (const uint8_t *)src,
                    src_stride, size, block_len, nshards, c->stream);
  launch_shard_finalize((uint8_t *)dst, dst_stride,
                        (const uint64_t *)hbuf.p,
                        (const uint64_t *)hbuf.p + nshards, size, block_len,
                        nshards, c->stream);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return hip_fail("shard_write launch", e);
  return GFRS_OK;
}

int gfrs_shard_parse_batch(gfrs_ctx *ctx, const void *img, size_t stride,
                           int64_t size, int64_t block_len,
                           uint64_t *out_meta, int64_t *bad_block_per_shard,
                           int nshards) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  if (block_len <= 0 || block_len % 4096) return GFRS_ERR_INVALID_BLOCK;
  if (size <= 0 || nshards <= 0) return GFRS_ERR_INVALID_SHARDS;
  int64_t body = gfrs_crc32b_encode_size(size, block_len);
  std::lock_guard<std::mutex> lk(c->mu);
  StreamGuard g(c);
  int rc;
  if ((rc = c->fail_buf.ensure(size_t(nshards) * (8 + 32))) != GFRS_OK)
    return rc;
  int64_t *bad_dev = (int64_t *)c->fail_buf.p;
  uint64_t *meta_dev = (uint64_t *)((uint8_t *)c->fail_buf.p + 8 * nshards);
  std::vector<int64_t> bad(nshards, INT64_MAX);
  HIP_TRY(hipMemcpyAsync(bad_dev, bad.data(), 8 * size_t(nshards),
                         hipMemcpyHostToDevice, c->stream));
  /* body block CRCs */
  launch_crc_verify((const uint8_t *)img + 32, stride, body, block_len,
                    nshards, bad_dev, c->stream);
  /* header/footer */
  launch_shard_parse((const uint8_t *)img, stride, size, block_len, nshards,
                     meta_dev, c->stream);
  HIP_TRY(hipMemcpyAsync(bad.data(), bad_dev, 8 * size_t(nshards),
                         hipMemcpyDeviceToHost, c->stream));
  HIP_TRY(hipMemcpyAsync(out_meta, meta_dev, 32 * size_t(nshards),
                         hipMemcpyDeviceToHost, c->stream));
  HIP_TRY(hipStreamSynchronize(c->stream));
  for (int j = 0; j < nshards; j++) {
    bad_block_per_shard[j] = bad[j] == INT64_MAX ? -1 : bad[j];
    /* body corruption (caught by the block-CRC recompute) must surface in
     * err too: the frame-header fold in the footer check cannot see
     * payload flips that keep the stored header */
    if (bad_block_per_shard[j] >= 0 && int64_t(out_meta[4 * j + 3]) == 0)
      out_meta[4 * j + 3] = uint64_t(int64_t(GFRS_ERR_MISMATCHED_CRC));
  }
  return GFRS_OK;
}

int gfrs_encode_idx(gfrs_ctx *ctx, const void *data_shard, int idx,
                    void *const *parity, size_t shard_len, int nparity) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  if (idx < 0 || idx >= c->t.n || nparity != c->t.m)
    return GFRS_ERR_INVALID_SHARDS;
  std::lock_guard<std::mutex> lk(c->mu);
  StreamGuard g(c);
  int rc;
  /* pointer table: [data, parity...]; plan: k=1 input col idx */
  std::vector<void *> ptrs(1 + nparity);
  ptrs[0] = const_cast<void *>(data_shard);
  for (int r = 0; r < nparity; r++) ptrs[1 + r] = parity[r];
  if ((rc = upload_ptrs(view_of(c), ptrs.data(), int(ptrs.size()))) !=
      GFRS_OK)
    return rc;
  /* per-idx plan cached in dec_cache keyed off a synthetic mask */
  PlanKey key = plan_key_tag(62); /* EncodeIdx namespace */
  plan_key_set(key, idx);
  DevPlan *p;
  std::unique_lock<std::mutex> cache_lk(c->cache_mu);
  auto it = c->dec_cache.find(key);
  if (it != c->dec_cache.end()) {
    p = it->second;
  } else {
    std::vector<int32_t> in{0};
    std::vector<int32_t> out(nparity);
    std::vector<uint8_t> rows(nparity);
    for (int r = 0; r < nparity; r++) {
      out[r] = 1 + r;
      rows[r] = c->enc_matrix[size_t(c->t.n + r) * c->t.n + idx];
    }
    p = new DevPlan();
    rc = p->upload(in, out, rows, c->stream);
    if (rc != GFRS_OK) {
      delete p;
      return rc;
    }
    c->dec_cache[key] = p;
  }
  launch_rs_apply_xor((const uint64_t *)c->ptr_buf.p, 1 + nparity,
                      (const int32_t *)p->in_idx.p, 1,
                      (const int32_t *)p->out_idx.p, p->nout,
                      (const uint8_t *)p->tabs.p, shard_len, 1, c->stream);
  HIP_TRY(hipStreamSynchronize(c->stream));
  return GFRS_OK;
}

/* LRC reconstruct+verify in ONE rs_apply_mixed pass: write rows for the
 * bad shards (data, global or local parity, composed over the k global
 * decode inputs), compare rows for every surviving global parity not
 * already an input and every surviving local parity.  Requires the bad
 * set to be globally decodable; returns GFRS_ERR_UNSUPPORTED otherwise
 * so the caller can fall back to local-stripe decode + full verify. */
static int lrc_mixed_reconstruct_verify(gfrs_ctx_impl *cc, void *base,
                                        size_t shard_len,
                                        size_t stripe_stride, int nstripes,
                                        const int32_t *bad_idx, int nbad,
                                        uint64_t *fail_bitmap) {
  const gfrs_tactic &t = cc->t;
  const int k = t.n, m = t.m;
  const int total_sh = k + m + t.l;
  std::vector<uint8_t> present(k + m, 1);
  std::vector<int> badv(bad_idx, bad_idx + nbad);
  for (int i = 0; i < nbad; i++) {
    if (badv[i] < k + m) present[badv[i]] = 0;
    for (int j = 0; j < i; j++)
      if (badv[j] == badv[i]) return GFRS_ERR_INVALID_SHARDS;
  }
  std::lock_guard<std::mutex> lk(cc->mu);
  StreamGuard g(cc);
  PlanKey key = plan_key_tag(60); /* LRC mixed reconstruct+verify */
  for (int b : badv) plan_key_set(key, b);
  DevPlan *plan = nullptr;
  uint32_t cmp_mask = 0;
  {
    std::lock_guard<std::mutex> cache_lk(cc->cache_mu);
    auto it = cc->dec_cache.find(key);
    if (it != cc->dec_cache.end()) plan = it->second;
  }
  /* cmp_mask is derivable from the (cached) row order: rows after nbad
   * are compares */
  if (!plan) {
    std::vector<int> valid;
    for (int i = 0; i < k + m && int(valid.size()) < k; i++)
      if (present[i]) valid.push_back(i);
    if (int(valid.size()) < k) return GFRS_ERR_UNSUPPORTED;
    std::vector<uint8_t> sub(size_t(k) * k), dec(size_t(k) * k);
    for (int r = 0; r < k; r++)
      memcpy(&sub[size_t(r) * k], &cc->enc_matrix[size_t(valid[r]) * k], k);
    if (!gf_invert(sub.data(), k, dec.data())) return GFRS_ERR_SINGULAR;
    std::vector<int> slot(k, -1);
    for (int j = 0; j < k; j++)
      if (valid[j] < k) slot[valid[j]] = j;
    const GfTables &gt2 = gft();
    auto dspace_row = [&](int sh, std::vector<uint8_t> &drow) {
      drow.assign(k, 0);
      if (sh < k) {
        drow[sh] = 1;
      } else if (sh < k + m) {
        memcpy(drow.data(), &cc->enc_matrix[size_t(sh) * k], k);
      } else {
        const int az = (sh - k - m) / cc->local_m;
        const int lp = (sh - k - m) % cc->local_m;
        auto idx = local_stripe(t, az);
        const uint8_t *lr =
            &cc->local_matrix[size_t(cc->local_n + lp) * cc->local_n];
        for (int j = 0; j < cc->local_n; j++) {
          const int g2i = idx[j];
          if (g2i < k) {
            drow[g2i] ^= lr[j];
          } else {
            const uint8_t *er = &cc->enc_matrix[size_t(g2i) * k];
            for (int d = 0; d < k; d++)
              drow[d] ^= gt2.mul[lr[j]][er[d]];
          }
        }
      }
    };
    auto xform = [&](const std::vector<uint8_t> &drow,
                     std::vector<uint8_t> &row) {
      row.assign(k, 0);
      for (int d = 0; d < k; d++) {
        const uint8_t coef = drow[d];
        if (!coef) continue;
        if (present[d]) {
          row[slot[d]] ^= coef;
        } else {
          for (int j = 0; j < k; j++)
            row[j] ^= gt2.mul[coef][dec[size_t(d) * k + j]];
        }
      }
    };
    std::vector<int32_t> in, out;
    std::vector<uint8_t> rows, row, drow;
    for (int j = 0; j < k; j++) in.push_back(valid[j]);
    for (int b : badv) { /* write rows */
      out.push_back(b);
      dspace_row(b, drow);
      xform(drow, row);
      rows.insert(rows.end(), row.begin(), row.end());
    }
    auto add_check = [&](int sh) {
      out.push_back(sh);
      dspace_row(sh, drow);
      xform(drow, row);
      rows.insert(rows.end(), row.begin(), row.end());
    };
    for (int p2 = k; p2 < k + m; p2++) {
      if (!present[p2]) continue;
      if (std::find(valid.begin(), valid.end(), p2) != valid.end()) continue;
      add_check(p2);
    }
    for (int q = k + m; q < total_sh; q++)
      if (std::find(badv.begin(), badv.end(), q) == badv.end())
        add_check(q);
    if (out.size() > 32) return GFRS_ERR_UNSUPPORTED; /* cmp_mask width */
    plan = new DevPlan();
    int rc = plan->upload(in, out, rows, cc->stream, k);
    if (rc != GFRS_OK) {
      delete plan;
      return rc;
    }
    {
      std::lock_guard<std::mutex> cache_lk(cc->cache_mu);
      cc->dec_cache[key] = plan;
    }
  }
  for (int r = nbad; r < plan->nout; r++) cmp_mask |= 1u << r;
  int rc;
  if ((rc = cc->fail_buf.ensure(size_t(nstripes) * 4)) != GFRS_OK) return rc;
  HIP_TRY(hipMemsetAsync(cc->fail_buf.p, 0, size_t(nstripes) * 4,
                         cc->stream));
  launch_rs_apply_mixed_strided((uint64_t)base, stripe_stride,
                                (const int32_t *)plan->in_idx.p, plan->k,
                                (const int32_t *)plan->out_idx.p, plan->nout,
                                (const uint8_t *)plan->tabs.p, cmp_mask,
                                shard_len, nstripes,
                                (uint32_t *)cc->fail_buf.p, cc->stream);
  std::vector<uint32_t> fails(nstripes);
  HIP_TRY(hipMemcpyAsync(fails.data(), cc->fail_buf.p, size_t(nstripes) * 4,
                         hipMemcpyDeviceToHost, cc->stream));
  HIP_TRY(hipStreamSynchronize(cc->stream));
  if (fail_bitmap) {
    memset(fail_bitmap, 0, ((nstripes + 63) / 64) * 8);
    for (int s2 = 0; s2 < nstripes; s2++)
      if (fails[s2]) fail_bitmap[s2 / 64] |= 1ull << (s2 % 64);
  }
  return GFRS_OK;
}

int gfrs_reconstruct_verify_batch(gfrs_ctx *ctx, void *base,
                                  size_t shard_len, size_t stripe_stride,
                                  int nstripes, const int32_t *bad_idx,
                                  int nbad, uint64_t *fail_bitmap) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  const gfrs_tactic &t = c->t;
  if (nstripes <= 0) return GFRS_ERR_INVALID_SHARDS;
  if (t.l != 0) {
    /* LRC single pass: every bad shard (data, global or local parity)
     * is a write row composed over the k global-decode inputs, and
     * every surviving global/local parity is a compare row - ONE
     * rs_apply_mixed pass instead of reconstruct + full re-read verify.
     * Falls back to two passes only when the bad set needs local-stripe
     * decode (globally undecodable). */
    int nglobad = 0;
    bool ok_range = true;
    for (int i = 0; i < nbad; i++) {
      if (bad_idx[i] < 0 || bad_idx[i] >= t.n + t.m + t.l) ok_range = false;
      else if (bad_idx[i] < t.n + t.m) nglobad++;
    }
    if (!ok_range) return GFRS_ERR_INVALID_SHARDS;
    if (nglobad <= t.m && c->fused_lrc_ok) {
      int rc = lrc_mixed_reconstruct_verify(c, base, shard_len,
                                            stripe_stride, nstripes,
                                            bad_idx, nbad, fail_bitmap);
      if (rc != GFRS_ERR_UNSUPPORTED) return rc;
    }
    int rc = gfrs_reconstruct_batch(ctx, base, shard_len, stripe_stride,
                                    nstripes, bad_idx, nbad, 0);
    if (rc != GFRS_OK) return rc;
    return gfrs_verify_batch(ctx, base, shard_len, stripe_stride, nstripes,
                             fail_bitmap);
  }
  const int k = t.n, m = t.m;
  std::vector<uint8_t> present(k + m, 1);
  for (int i = 0; i < nbad; i++) {
    if (bad_idx[i] < 0 || bad_idx[i] >= k + m) return GFRS_ERR_INVALID_SHARDS;
    present[bad_idx[i]] = 0;
  }
  std::lock_guard<std::mutex> lk(c->mu);
  StreamGuard g(c);

  PlanKey key = plan_key_tag(63); /* reconstruct+verify namespace */
  for (int i = 0; i < k + m; i++)
    if (!present[i]) plan_key_set(key, i);
  DevPlan *plan = nullptr;
  uint32_t cmp_mask = 0;
  {
    std::lock_guard<std::mutex> cache_lk(c->cache_mu);
    auto it = c->dec_cache.find(key);
    if (it != c->dec_cache.end()) plan = it->second;
  }
  if (!plan) {
    /* valid inputs: first k present in index order (reedsolomon.go:1453) */
    std::vector<int> valid;
    for (int i = 0; i < k + m && int(valid.size()) < k; i++)
      if (present[i]) valid.push_back(i);
    if (int(valid.size()) < k) return GFRS_ERR_TOO_FEW_SHARDS;
    std::vector<uint8_t> sub(size_t(k) * k), dec(size_t(k) * k);
    for (int r = 0; r < k; r++)
      memcpy(&sub[size_t(r) * k], &c->enc_matrix[size_t(valid[r]) * k], k);
    if (!gf_invert(sub.data(), k, dec.data())) return GFRS_ERR_SINGULAR;
    /* input slot of each present data column */
    std::vector<int> slot(k, -1);
    for (int j = 0; j < k; j++)
      if (valid[j] < k) slot[valid[j]] = j;

    const GfTables &gt2 = gft();
    std::vector<int32_t> in, out;
    std::vector<uint8_t> rows;
    for (int j = 0; j < k; j++) in.push_back(valid[j]);
    /* 1) missing data: decode rows (write) */
    for (int i = 0; i < k; i++)
      if (!present[i]) {
        out.push_back(i);
        rows.insert(rows.end(), &dec[size_t(i) * k], &dec[size_t(i) * k + k]);
      }
    /* 2) every parity row composed over the valid inputs:
     *    ver_p[j] = enc_p[d]·1[slot(d)==j] (+) Σ_missing enc_p[d]·dec[d][j] */
    for (int p2 = k; p2 < k + m; p2++) {
      std::vector<uint8_t> row(k, 0);
      for (int d = 0; d < k; d++) {
        const uint8_t coef = c->enc_matrix[size_t(p2) * k + d];
        if (present[d]) {
          row[slot[d]] ^= coef;
        } else {
          for (int j = 0; j < k; j++)
            row[j] ^= gt2.mul[coef][dec[size_t(d) * k + j]];
        }
      }
      out.push_back(p2);
      rows.insert(rows.end(), row.begin(), row.end());
    }
    plan = new DevPlan();
    int rc = plan->upload(in, out, rows, c->stream);
    if (rc != GFRS_OK) {
      delete plan;
      return rc;
    }
    {
      std::lock_guard<std::mutex> cache_lk(c->cache_mu);
      c->dec_cache[key] = plan;
    }
  }
  /* compare bits: present parity rows; missing (data or parity) written */
  {
    int nmissdata = 0;
    for (int i = 0; i < k; i++)
      if (!present[i]) nmissdata++;
    for (int p2 = k; p2 < k + m; p2++)
      if (present[p2]) cmp_mask |= 1u << (nmissdata + (p2 - k));
  }
  int rc;
  if ((rc = c->fail_buf.ensure(size_t(nstripes) * 4)) != GFRS_OK) return rc;
  HIP_TRY(hipMemsetAsync(c->fail_buf.p, 0, size_t(nstripes) * 4, c->stream));
  launch_rs_apply_mixed_strided((uint64_t)base, stripe_stride,
                                (const int32_t *)plan->in_idx.p, plan->k,
                                (const int32_t *)plan->out_idx.p, plan->nout,
                                (const uint8_t *)plan->tabs.p, cmp_mask,
                                shard_len, nstripes,
                                (uint32_t *)c->fail_buf.p, c->stream);
  std::vector<uint32_t> fails(nstripes);
  HIP_TRY(hipMemcpyAsync(fails.data(), c->fail_buf.p, size_t(nstripes) * 4,
                         hipMemcpyDeviceToHost, c->stream));
  HIP_TRY(hipStreamSynchronize(c->stream));
  if (fail_bitmap) {
    memset(fail_bitmap, 0, ((nstripes + 63) / 64) * 8);
    for (int s2 = 0; s2 < nstripes; s2++)
      if (fails[s2]) fail_bitmap[s2 / 64] |= 1ull << (s2 % 64);
  }
  return GFRS_OK;
}

int gfrs_update_idx(gfrs_ctx *ctx, const void *old_shard,
                    const void *new_shard, int idx, void *const *parity,
                    size_t shard_len, int nparity) {
  int rc = gfrs_encode_idx(ctx, old_shard, idx, parity, shard_len, nparity);
  if (rc != GFRS_OK) return rc;
  return gfrs_encode_idx(ctx, new_shard, idx, parity, shard_len, nparity);
}

int gfrs_repair_batch(gfrs_ctx *ctx, void *base, size_t shard_len,
                      size_t stripe_stride, int nstripes,
                      const int32_t *bad_idx, int nbad, void *disk_dst,
                      size_t dst_stride, int64_t block_len,
                      const uint64_t *bids, const uint64_t *vuids,
                      uint64_t *fail_bitmap) {
  if (nbad <= 0) return GFRS_ERR_INVALID_SHARDS;
  auto *cc = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  {
    const gfrs_tactic &t = cc->t;
    /* Fused repair: one kernel reads the k inputs + surviving-parity
     * check shards and writes ONLY the repaired shards' framed bodies -
     * the raw reconstruction never touches HBM and `base` is NOT
     * mutated (the legacy path below reconstructs in place). */
    static const size_t fmin = []() {
      const char *e = getenv("GFRS_FUSED_MIN");
      const long v = e ? atol(e) : 0;
      return v > 0 ? size_t(v) : size_t(4097);
    }();
    int nglobad = 0;
    for (int i = 0; i < nbad; i++)
      if (bad_idx[i] < t.n + t.m) nglobad++;
    if ((t.l == 0 || cc->fused_lrc_ok) && block_len == 65536 && t.m >= 1 &&
        t.n + t.m + t.l <= 16 && nglobad <= t.m && nbad <= 4 &&
        (shard_len >= fmin || shard_len <= 4096) && dst_stride % 4 == 0 &&
        nstripes > 0) {
      const int k = t.n, m = t.m;
      const int total_sh = k + m + t.l;
      std::vector<uint8_t> present(k + m, 1);
      std::vector<int> badv(bad_idx, bad_idx + nbad);
      bool badrange = false;
      for (int i = 0; i < nbad; i++) {
        if (badv[i] < 0 || badv[i] >= total_sh) return GFRS_ERR_INVALID_SHARDS;
        if (badv[i] < k + m) present[badv[i]] = 0;
        for (int j = 0; j < i; j++)
          if (badv[j] == badv[i]) badrange = true;
      }
      if (badrange) return GFRS_ERR_INVALID_SHARDS;
      std::vector<int> sbad = badv;
      std::sort(sbad.begin(), sbad.end());
      uint32_t colpack = 0;
      for (int r = 0; r < nbad; r++) {
        int col = int(std::find(badv.begin(), badv.end(), sbad[r]) -
                      badv.begin());
        colpack |= uint32_t(col) << (4 * r);
      }
      std::lock_guard<std::mutex> lk(cc->mu);
      StreamGuard g(cc);
      PlanKey key = plan_key_tag(61); /* fused-repair namespace */
      for (int b : badv) plan_key_set(key, b); /* incl. local-parity bads */
      DevPlan *plan = nullptr;
      {
        std::lock_guard<std::mutex> cache_lk(cc->cache_mu);
        auto it = cc->dec_cache.find(key);
        if (it != cc->dec_cache.end()) plan = it->second;
      }
      if (!plan) {
        std::vector<int> valid;
        for (int i = 0; i < k + m && int(valid.size()) < k; i++)
          if (present[i]) valid.push_back(i);
        if (int(valid.size()) < k) return GFRS_ERR_TOO_FEW_SHARDS;
        std::vector<uint8_t> sub(size_t(k) * k), dec(size_t(k) * k);
        for (int r = 0; r < k; r++)
          memcpy(&sub[size_t(r) * k], &cc->enc_matrix[size_t(valid[r]) * k],
                 k);
        if (!gf_invert(sub.data(), k, dec.data()))
          return GFRS_ERR_SINGULAR;
        std::vector<int> slot(k, -1);
        for (int j = 0; j < k; j++)
          if (valid[j] < k) slot[valid[j]] = j;
        const GfTables &gt2 = gft();
        /* shard idx -> its row over the n data shards (locals compose
         * through the global-parity rows, as in the fused_lrc plan) */
        auto dspace_row = [&](int sh, std::vector<uint8_t> &drow) {
          drow.assign(k, 0);
          if (sh < k) {
            drow[sh] = 1;
          } else if (sh < k + m) {
            memcpy(drow.data(), &cc->enc_matrix[size_t(sh) * k], k);
          } else {
            const int az = (sh - k - m) / cc->local_m;
            const int lp = (sh - k - m) % cc->local_m;
            auto idx = local_stripe(t, az);
            const uint8_t *lr =
                &cc->local_matrix[size_t(cc->local_n + lp) * cc->local_n];
            for (int j = 0; j < cc->local_n; j++) {
              const int g2i = idx[j];
              if (g2i < k) {
                drow[g2i] ^= lr[j];
              } else {
                const uint8_t *er = &cc->enc_matrix[size_t(g2i) * k];
                for (int d = 0; d < k; d++)
                  drow[d] ^= gt2.mul[lr[j]][er[d]];
              }
            }
          }
        };
        /* data-space row -> row over the k chosen inputs */
        auto xform = [&](const std::vector<uint8_t> &drow,
                         std::vector<uint8_t> &row) {
          row.assign(k, 0);
          for (int d = 0; d < k; d++) {
            const uint8_t coef = drow[d];
            if (!coef) continue;
            if (present[d]) {
              row[slot[d]] ^= coef;
            } else {
              for (int j = 0; j < k; j++)
                row[j] ^= gt2.mul[coef][dec[size_t(d) * k + j]];
            }
          }
        };
        std::vector<int32_t> in, out;
        std::vector<uint8_t> rows, row, drow;
        for (int j = 0; j < k; j++) in.push_back(valid[j]);
        /* rebuild rows in sorted-bad order (data, global or local) */
        for (int b : sbad) {
          out.push_back(b);
          dspace_row(b, drow);
          xform(drow, row);
          rows.insert(rows.end(), row.begin(), row.end());
        }
        /* check rows while they fit in the 4-row kernel budget:
         * surviving globals not already inputs, then surviving locals
         * (the locals restore detection when npresent == k globally) */
        auto add_check = [&](int sh) {
          if (int(out.size()) >= 4) return;
          out.push_back(sh);
          dspace_row(sh, drow);
          xform(drow, row);
          rows.insert(rows.end(), row.begin(), row.end());
          in.push_back(sh); /* cmp shard index rides after the k inputs */
        };
        for (int p2 = k; p2 < k + m; p2++) {
          if (!present[p2]) continue;
          if (std::find(valid.begin(), valid.end(), p2) != valid.end())
            continue; /* input parity: its check is the identity */
          add_check(p2);
        }
        for (int q = k + m; q < total_sh; q++)
          if (std::find(badv.begin(), badv.end(), q) == badv.end())
            add_check(q);
        plan = new DevPlan();
        int rc2 = plan->upload(in, out, rows, cc->stream, k);
        if (rc2 != GFRS_OK) {
          delete plan;
          return rc2;
        }
        {
          std::lock_guard<std::mutex> cache_lk(cc->cache_mu);
          cc->dec_cache[key] = plan;
        }
      }
      const int gm = plan->nout;
      int rc2;
      if ((rc2 = cc->fail_buf.ensure(size_t(nstripes) * 4)) != GFRS_OK)
        return rc2;
      HIP_TRY(hipMemsetAsync(cc->fail_buf.p, 0, size_t(nstripes) * 4,
                             cc->stream));
      /* id arrays for every (stripe, bad) image, caller's column order;
       * headers themselves are built by the finalize kernel */
      const size_t nimg = size_t(nstripes) * nbad, idbytes = nimg * 8;
      if ((rc2 = cc->stage_pin.ensure(idbytes * 2)) != GFRS_OK) return rc2;
      memcpy(cc->stage_pin.p, bids, idbytes);
      memcpy((uint8_t *)cc->stage_pin.p + idbytes, vuids, idbytes);
      DevBuf &hbuf = cc->ptr_buf;
      if ((rc2 = hbuf.ensure(idbytes * 2)) != GFRS_OK) return rc2;
      HIP_TRY(hipMemcpyAsync(hbuf.p, cc->stage_pin.p, idbytes * 2,
                             hipMemcpyHostToDevice, cc->stream));
      /* Chunked finalize pipeline (GFRS_REPAIR_CHUNK=N, DEFAULT OFF):
       * the header/footer pass of chunk c only needs chunk c's images
       * (the footer CRC is GF(2)-combined from the frame CRCs the
       * repair kernel just wrote), so it can run on an aux stream
       * behind an event while chunk c+1's repair kernel streams on the
       * main stream.  Measured @512x4MiB RS(6+3) bad=[2,6]: the single
       * launch wins — 5.66 ms serial vs 6.48-7.71 chunked (CH=64/128/
       * 256): each repair sub-launch pays a fill/drain ramp and the
       * finalize competes for bandwidth, costing more than the ~0.9 ms
       * finalize pass hides.  Kept as a measured variant; parity-tested
       * (test_repair_batch_chunk_pipeline). */
      int chunk = nstripes;
      if (shard_len > 4096) {
        const char *e = getenv("GFRS_REPAIR_CHUNK");
        const long v = e ? atol(e) : 0;
        if (v > 0 && v < (long)nstripes) chunk = (int)v;
      }
      const bool pipe = chunk < nstripes && cc->ensure_aux() == GFRS_OK;
      if (!pipe) {
        if (shard_len <= 4096)
          launch_rs_repair_frame_small((uint8_t *)disk_dst + 32, dst_stride,
                                       (uint64_t)base, stripe_stride,
                                       shard_len, plan->k, gm, nbad,
                                       (const int32_t *)plan->in_idx.p,
                                       (const uint8_t *)plan->tabs.p,
                                       colpack, (uint32_t *)cc->fail_buf.p,
                                       nstripes, cc->stream);
        else
          launch_rs_repair_frame((uint8_t *)disk_dst + 32, dst_stride,
                                 (uint64_t)base, stripe_stride, shard_len,
                                 plan->k, gm, nbad,
                                 (const int32_t *)plan->in_idx.p,
                                 (const uint8_t *)plan->tabs.p, colpack,
                                 (uint32_t *)cc->fail_buf.p, nstripes,
                                 cc->stream);
        launch_shard_finalize((uint8_t *)disk_dst, dst_stride,
                              (const uint64_t *)hbuf.p,
                              (const uint64_t *)hbuf.p + nimg,
                              int64_t(shard_len), block_len, nstripes * nbad,
                              cc->stream);
      } else {
        int ci = 0;
        for (int lo = 0; lo < nstripes; lo += chunk, ci++) {
          const int cn = std::min(chunk, nstripes - lo);
          uint8_t *cd =
              (uint8_t *)disk_dst + size_t(lo) * nbad * dst_stride;
          launch_rs_repair_frame(
              cd + 32, dst_stride,
              (uint64_t)base + (uint64_t)lo * stripe_stride, stripe_stride,
              shard_len, plan->k, gm, nbad, (const int32_t *)plan->in_idx.p,
              (const uint8_t *)plan->tabs.p, colpack,
              (uint32_t *)cc->fail_buf.p + lo, cn, cc->stream);
          hipEvent_t ev = cc->aux_ev[ci & 1];
          HIP_TRY(hipEventRecord(ev, cc->stream));
          HIP_TRY(hipStreamWaitEvent(cc->aux_stream, ev, 0));
          launch_shard_finalize(
              cd, dst_stride, (const uint64_t *)hbuf.p + size_t(lo) * nbad,
              (const uint64_t *)hbuf.p + nimg + size_t(lo) * nbad,
              int64_t(shard_len), block_len, cn * nbad, cc->aux_stream);
        }
        HIP_TRY(hipEventRecord(cc->aux_done, cc->aux_stream));
        HIP_TRY(hipStreamWaitEvent(cc->stream, cc->aux_done, 0));
      }
      hipError_t e = hipGetLastError();
      if (e != hipSuccess) return hip_fail("repair_frame launch", e);
      std::vector<uint32_t> fails(nstripes);
      HIP_TRY(hipMemcpyAsync(fails.data(), cc->fail_buf.p,
                             size_t(nstripes) * 4, hipMemcpyDeviceToHost,
                             cc->stream));
      HIP_TRY(hipStreamSynchronize(cc->stream));
      if (fail_bitmap) {
        memset(fail_bitmap, 0, ((nstripes + 63) / 64) * 8);
        for (int s2 = 0; s2 < nstripes; s2++)
          if (fails[s2]) fail_bitmap[s2 / 64] |= 1ull << (s2 % 64);
      }
      return GFRS_OK;
    }
  }
  /* legacy path: reconstruct + the mandatory verify in one data pass */
  int rc = gfrs_reconstruct_verify_batch(ctx, base, shard_len, stripe_stride,
                                         nstripes, bad_idx, nbad,
                                         fail_bitmap);
  if (rc != GFRS_OK) return rc;
  /* frame each repaired shard; images laid out (stripe, bad) row-major.
   * For bad shard b the raw source is strided across stripes.  The
   * sub-calls take the ctx lock themselves; stream ordering serializes
   * the kernels. */
  std::vector<uint64_t> bb(nstripes), vv(nstripes);
  for (int b = 0; b < nbad; b++) {
    /* headers differ per stripe: gather this bad-shard column's ids */
    for (int s2 = 0; s2 < nstripes; s2++) {
      bb[s2] = bids[size_t(s2) * nbad + b];
      vv[s2] = vuids[size_t(s2) * nbad + b];
    }
    /* write images for column b: dst row stride = nbad*dst_stride */
    rc = gfrs_shard_write_batch(
        ctx, (uint8_t *)disk_dst + size_t(b) * dst_stride,
        size_t(nbad) * dst_stride,
        (const uint8_t *)base + size_t(bad_idx[b]) * shard_len,
        stripe_stride, int64_t(shard_len), block_len, bb.data(), vv.data(),
        nstripes);
    if (rc != GFRS_OK) return rc;
  }
  return GFRS_OK;
}

/* ---------------- crc32block ---------------- */

/* Host CRC32-IEEE (reflected 0xEDB88320), slice-by-8, Update semantics
 * like hash/crc32.Update (util.go:22 ChecksumIEEE call sites).  Host-side
 * on purpose: it backs the per-block streaming request-body wrapper
 * (request_body.go:57-127), which the reference also runs on host CPUs;
 * all bulk framing goes through the device kernels. */
static uint32_t g_crc8tab[8][256];
static std::once_flag g_crc8_once;

static void crc8tab_init() {
  for (uint32_t i = 0; i < 256; i++) {
    uint32_t c = i;
    for (int k = 0; k < 8; k++) c = (c >> 1) ^ (0xEDB88320u & (0u - (c & 1)));
    g_crc8tab[0][i] = c;
  }
  for (int t = 1; t < 8; t++)
    for (uint32_t i = 0; i < 256; i++)
      g_crc8tab[t][i] =
          (g_crc8tab[t - 1][i] >> 8) ^ g_crc8tab[0][g_crc8tab[t - 1][i] & 0xFF];
}

uint32_t gfrs_crc32_host(uint32_t crc, const void *data, int64_t n) {
  std::call_once(g_crc8_once, crc8tab_init);
  const uint8_t *p = (const uint8_t *)data;
  uint32_t c = ~crc;
  while (n >= 8) {
    uint32_t lo, hi;
    memcpy(&lo, p, 4);
    memcpy(&hi, p + 4, 4);
    lo ^= c;
    c = g_crc8tab[7][lo & 0xFF] ^ g_crc8tab[6][(lo >> 8) & 0xFF] ^
        g_crc8tab[5][(lo >> 16) & 0xFF] ^ g_crc8tab[4][lo >> 24] ^
        g_crc8tab[3][hi & 0xFF] ^ g_crc8tab[2][(hi >> 8) & 0xFF] ^
        g_crc8tab[1][(hi >> 16) & 0xFF] ^ g_crc8tab[0][hi >> 24];
    p += 8;
    n -= 8;
  }
  while (n-- > 0) c = (c >> 8) ^ g_crc8tab[0][(c ^ *p++) & 0xFF];
  return ~c;
}

int64_t gfrs_crc32b_encode_size(int64_t size, int64_t block_len) {
  if (block_len <= 0 || block_len % 4096) return GFRS_ERR_INVALID_BLOCK;
  int64_t payload = block_len - 4;
  return size + 4 * ((size + payload - 1) / payload);
}

int64_t gfrs_crc32b_decode_size(int64_t size, int64_t block_len) {
  if (block_len <= 0 || block_len % 4096) return GFRS_ERR_INVALID_BLOCK;
  return size - 4 * ((size + block_len - 1) / block_len);
}

static int crc32b_encode_batch_impl(gfrs_ctx_impl *c, void *dst,
                                    size_t dst_stride, const void *src,
                                    size_t src_stride, int64_t n,
                                    int64_t block_len, int nshards) {
  if (block_len <= 0 || block_len % 4096) return GFRS_ERR_INVALID_BLOCK;
  if (n <= 0 || nshards <= 0) return GFRS_ERR_INVALID_SHARDS;
  StreamGuard g(c);
  launch_crc_encode((uint8_t *)dst, dst_stride, (const uint8_t *)src,
                    src_stride, n, block_len, nshards, c->stream);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return hip_fail("crc encode launch", e);
  return GFRS_OK;
}

int gfrs_crc32b_encode_batch(gfrs_ctx *ctx, void *dst, size_t dst_stride,
                             const void *src, size_t src_stride, int64_t n,
                             int64_t block_len, int nshards) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  std::lock_guard<std::mutex> lk(c->mu);
  return crc32b_encode_batch_impl(c, dst, dst_stride, src, src_stride, n,
                                  block_len, nshards);
}

int64_t gfrs_crc32b_encode(gfrs_ctx *ctx, void *dst, const void *src,
                           int64_t n, int64_t block_len) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  std::lock_guard<std::mutex> lk(c->mu);
  int rc = crc32b_encode_batch_impl(c, dst, 0, src, 0, n, block_len, 1);
  if (rc != GFRS_OK) return rc;
  StreamGuard g(c);
  HIP_TRY(hipStreamSynchronize(c->stream));
  return gfrs_crc32b_encode_size(n, block_len);
}

int gfrs_crc32b_verify_batch(gfrs_ctx *ctx, const void *framed, size_t stride,
                             int64_t framed_len, int64_t block_len,
                             int nshards, int64_t *bad_block_per_shard) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  if (block_len <= 0 || block_len % 4096) return GFRS_ERR_INVALID_BLOCK;
  if (framed_len <= 0 || nshards <= 0) return GFRS_ERR_INVALID_SHARDS;
  std::lock_guard<std::mutex> lk(c->mu);
  StreamGuard g(c);
  int rc;
  if ((rc = c->fail_buf.ensure(size_t(nshards) * 8)) != GFRS_OK) return rc;
  std::vector<int64_t> bad(nshards, INT64_MAX);
  HIP_TRY(hipMemcpyAsync(c->fail_buf.p, bad.data(), size_t(nshards) * 8,
                         hipMemcpyHostToDevice, c->stream));
  launch_crc_verify((const uint8_t *)framed, stride, framed_len, block_len,
                    nshards, (int64_t *)c->fail_buf.p, c->stream);
  HIP_TRY(hipMemcpyAsync(bad.data(), c->fail_buf.p, size_t(nshards) * 8,
                         hipMemcpyDeviceToHost, c->stream));
  HIP_TRY(hipStreamSynchronize(c->stream));
  for (int i = 0; i < nshards; i++)
    bad_block_per_shard[i] = bad[i] == INT64_MAX ? -1 : bad[i];
  return GFRS_OK;
}

int gfrs_crc32b_verify(gfrs_ctx *ctx, const void *framed, int64_t framed_len,
                       int64_t block_len, int64_t *bad_block) {
  return gfrs_crc32b_verify_batch(ctx, framed, 0, framed_len, block_len, 1,
                                  bad_block);
}

int64_t gfrs_crc32b_decode(gfrs_ctx *ctx, void *dst, const void *framed,
                           int64_t framed_len, int64_t block_len) {
  auto *c = reinterpret_cast<gfrs_ctx_impl *>(ctx);
  if (block_len <= 0 || block_len % 4096) return GFRS_ERR_INVALID_BLOCK;
  std::lock_guard<std::mutex> lk(c->mu);
  StreamGuard g(c);
  int rc;
  if ((rc = c->fail_buf.ensure(8)) != GFRS_OK) return rc;
  int64_t bad = INT64_MAX;
  HIP_TRY(hipMemcpyAsync(c->fail_buf.p, &bad, 8, hipMemcpyHostToDevice,
                         c->stream));
  launch_crc_decode((uint8_t *)dst, 0, (const uint8_t *)framed, 0, framed_len,
                    block_len, 1, (int64_t *)c->fail_buf.p, c->stream);
  HIP_TRY(hipMemcpyAsync(&bad, c->fail_buf.p, 8, hipMemcpyDeviceToHost,
                         c->stream));
  HIP_TRY(hipStreamSynchronize(c->stream));
  if (bad != INT64_MAX) return GFRS_ERR_MISMATCHED_CRC;
  return gfrs_crc32b_decode_size(framed_len, block_len);
}

} /* extern "C" */
