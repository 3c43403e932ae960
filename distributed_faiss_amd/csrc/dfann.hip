// dfann — MI355X-native ANN engine: host runtime + C-ABI (include/dfann.h).
//
// Self-contained HIP/C++ shared library: no torch, no BLAS — every FLOP of
// the hot path runs in the hand-written gfx950 kernels of kernels.hip.
// Python binds via ctypes (distributed_faiss_amd/hip_engine.py).
//
// Index model:
//  * trained artifacts: coarse centroids (+norms), PQ codebooks / SQ ranges
//  * codes live twice: an arrival-order staging arena (append target) and
//    a CSR image grouped by inverted list (search layout), rebuilt lazily
//    ("finalize") via a host counting sort + device gather — the stable
//    sort preserves arrival order inside each list, so CSR position order
//    == ascending arrival id, which the scan's tie-break relies on.
//  * ids are implicit arrival positions (reference quirk).

#include "kernels.hip"
#include "hnsw.hip"

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>
#include <stdexcept>
#include <algorithm>

#include "../../include/dfann.h"

// ---------------------------------------------------------------------------
// error plumbing
// ---------------------------------------------------------------------------

static thread_local std::string g_err;

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      throw std::runtime_error(std::string("HIP error: ") +                    \
                               hipGetErrorString(_e) + " at " #expr);          \
    }                                                                          \
  } while (0)

extern "C" const char *dfann_last_error(void) { return g_err.c_str(); }

// DFANN_TRACE=1: synchronize + log after each build/search phase (debug)
static bool trace_on() {
  static int v = -1;
  if (v < 0) {
    const char *e = getenv("DFANN_TRACE");
    v = (e && e[0] == '1') ? 1 : 0;
  }
  return v == 1;
}

static void trace_point(const char *what, hipStream_t s) {
  if (!trace_on()) return;
  hipError_t e = hipStreamSynchronize(s);
  fprintf(stderr, "[dfann] %s: %s\n", what, hipGetErrorString(e));
  fflush(stderr);
}

// ---------------------------------------------------------------------------
// tiny JSON (flat dict of scalars; produced by our own Python side)
// ---------------------------------------------------------------------------

static bool json_find(const std::string &js, const char *key, std::string &out) {
  std::string pat = std::string("\"") + key + "\"";
  size_t p = js.find(pat);
  if (p == std::string::npos) return false;
  p = js.find(':', p + pat.size());
  if (p == std::string::npos) return false;
  ++p;
  while (p < js.size() && (js[p] == ' ' || js[p] == '\t')) ++p;
  if (p >= js.size()) return false;
  if (js[p] == '"') {
    size_t e = js.find('"', p + 1);
    out = js.substr(p + 1, e - p - 1);
  } else {
    size_t e = p;
    while (e < js.size() && (isdigit(js[e]) || js[e] == '-' || js[e] == '.' ||
                             js[e] == 'e' || js[e] == 'E' || js[e] == '+'))
      ++e;
    out = js.substr(p, e - p);
  }
  return true;
}

static long long json_int(const std::string &js, const char *key, long long dflt) {
  std::string v;
  if (!json_find(js, key, v) || v.empty()) return dflt;
  return atoll(v.c_str());
}

static std::string json_str(const std::string &js, const char *key,
                            const char *dflt) {
  std::string v;
  if (!json_find(js, key, v)) return dflt;
  return v;
}

// ---------------------------------------------------------------------------
// device buffer
// ---------------------------------------------------------------------------

struct DevBuf {
  void *p = nullptr;
  size_t cap = 0;
  DevBuf() = default;
  DevBuf(const DevBuf &) = delete;
  DevBuf &operator=(const DevBuf &) = delete;
  ~DevBuf() { free(); }
  void ensure(size_t bytes) {
    if (bytes <= cap) return;
    void *np = nullptr;
    HIP_CHECK(hipMalloc(&np, bytes));
    if (p) HIP_CHECK(hipFree(p));
    p = np;
    cap = bytes;
  }
  // grow preserving `keep` bytes of content
  void grow_keep(size_t bytes, size_t keep) {
    if (bytes <= cap) return;
    size_t nb = std::max(bytes, cap * 2);
    void *np = nullptr;
    HIP_CHECK(hipMalloc(&np, nb));
    if (p && keep) HIP_CHECK(hipMemcpy(np, p, keep, hipMemcpyDeviceToDevice));
    if (p) HIP_CHECK(hipFree(p));
    p = np;
    cap = nb;
  }
  void free() {
    if (p) (void)hipFree(p);  // best-effort in destructor paths
    p = nullptr;
    cap = 0;
  }
  template <typename T> T *as() { return reinterpret_cast<T *>(p); }
};

// ---------------------------------------------------------------------------
// slab arena: a growable row store made of fixed-size slabs (1<<rlog rows
// each). The CSR rebuild recycles old slabs behind its write window
// (rebuild_csr), so code images never need a 2x contiguous transient —
// the round-1 staging-arena + CSR duplication is gone.
// Kernels address rows through a device slab-pointer table (kernels.hip
// slab_row).
// ---------------------------------------------------------------------------

struct SlabArena {
  int stride = 0;
  int rlog = 0;                 // rows per slab = 1 << rlog
  std::vector<void *> slabs;    // nullptr = freed
  int64_t rows = 0;             // rows in use (bump)
  DevBuf table;                 // device copy of `slabs`
  bool table_dirty = true;

  void init(int stride_, size_t target_slab_bytes = (size_t)512 << 20) {
    reset();
    stride = stride_;
    rlog = 0;
    while (((size_t)2 << rlog) * stride <= target_slab_bytes) ++rlog;
  }
  int64_t rps() const { return (int64_t)1 << rlog; }
  size_t slab_bytes() const { return (size_t)rps() * stride; }
  void ensure_rows(int64_t n) {
    size_t need = (size_t)((n + rps() - 1) >> rlog);
    while (slabs.size() < need) {
      void *p = nullptr;
      HIP_CHECK(hipMalloc(&p, slab_bytes()));
      slabs.push_back(p);
      table_dirty = true;
    }
  }
  void free_slab(size_t i) {
    if (i < slabs.size() && slabs[i]) {
      (void)hipFree(slabs[i]);
      slabs[i] = nullptr;  // kernels past the window never touch it
    }
  }
  const uint8_t *const *dev_table(hipStream_t s) {
    if (table_dirty) {
      table.ensure(std::max(slabs.size() * 8, (size_t)8));
      if (!slabs.empty())
        HIP_CHECK(hipMemcpyAsync(table.p, slabs.data(), slabs.size() * 8,
                                 hipMemcpyHostToDevice, s));
      HIP_CHECK(hipStreamSynchronize(s));
      table_dirty = false;
    }
    return table.as<const uint8_t *const>();
  }
  uint8_t *const *dev_table_mut(hipStream_t s) {
    return const_cast<uint8_t *const *>(
        reinterpret_cast<const uint8_t *const *>(dev_table(s)));
  }
  // contiguous-host <-> arena row-range copies (persistence, get_lists)
  void copy_rows_to_host(void *dst, int64_t row0, int64_t n) const {
    char *d = (char *)dst;
    int64_t r = row0;
    while (n > 0) {
      int64_t in_slab = std::min<int64_t>(n, rps() - (r & (rps() - 1)));
      const char *src = (const char *)slabs[r >> rlog] +
                        (size_t)(r & (rps() - 1)) * stride;
      HIP_CHECK(hipMemcpy(d, src, (size_t)in_slab * stride,
                          hipMemcpyDeviceToHost));
      d += (size_t)in_slab * stride;
      r += in_slab;
      n -= in_slab;
    }
  }
  void copy_rows_from_host(const void *src, int64_t row0, int64_t n) {
    ensure_rows(row0 + n);
    const char *s = (const char *)src;
    int64_t r = row0;
    while (n > 0) {
      int64_t in_slab = std::min<int64_t>(n, rps() - (r & (rps() - 1)));
      char *dst = (char *)slabs[r >> rlog] + (size_t)(r & (rps() - 1)) * stride;
      HIP_CHECK(hipMemcpy(dst, s, (size_t)in_slab * stride,
                          hipMemcpyHostToDevice));
      s += (size_t)in_slab * stride;
      r += in_slab;
      n -= in_slab;
    }
  }
  void reset() {
    for (auto *p : slabs)
      if (p) (void)hipFree(p);
    slabs.clear();
    rows = 0;
    table.free();
    table_dirty = true;
  }
  ~SlabArena() { reset(); }
};

// ---------------------------------------------------------------------------
// splitmix64 (identical to oracle/core.py)
// ---------------------------------------------------------------------------

struct SplitMix64 {
  uint64_t x;
  explicit SplitMix64(uint64_t seed) : x(seed) {}
  uint64_t next() {
    uint64_t z = (x += 0x9E3779B97F4A7C15ULL);
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
    return z ^ (z >> 31);
  }
};

// first k of a partial Fisher-Yates over range(n) (== oracle
// partial_shuffle_indices)
static std::vector<int64_t> pick_init(int64_t n, int64_t k, uint64_t seed) {
  k = std::min(k, n);
  std::vector<int64_t> idx(n);
  for (int64_t i = 0; i < n; ++i) idx[i] = i;
  SplitMix64 rng(seed);
  for (int64_t i = 0; i < k; ++i) {
    int64_t j = i + (int64_t)(rng.next() % (uint64_t)(n - i));
    std::swap(idx[i], idx[j]);
  }
  idx.resize(k);
  return idx;
}

// ---------------------------------------------------------------------------
// index object
// ---------------------------------------------------------------------------

enum IdxType { T_FLAT = 0, T_IVFFLAT = 1, T_IVFPQ = 2, T_IVFSQ = 3, T_HNSW = 4 };
enum { M_IP = 0, M_L2 = 1 };

static int round16(int b) { return (b + 15) & ~15; }

struct TimingEv {
  hipEvent_t a, b;
};

struct dfann_index {
  std::string spec_json;
  int type = T_FLAT;
  int metric = M_IP;
  int d = 0, nlist = 0, m = 0, nbits = 8, dsub = 0, nprobe = 1;
  bool sq8 = false;  // ivfsq: true = 8bit, false = fp16
  uint64_t seed = 1234;
  bool trained = false;
  int64_t ntotal = 0;
  int code_bytes = 0, stride = 0;
  int ws_mb = 512;  // chunk budget for key matrices (spec "ws_mb"; tests
                    // shrink it to force the multi-chunk paths)
  int max_ppc = 256;  // k-means subsample cap per centroid (spec "max_ppc")
  bool coarse_bf16 = false;
  bool pq_pre = false;  // spec "pq_precomputed": PQ-L2 term2/term3 tables
  int pq_lut_global = -1;  // spec "pq_lut_global": ADC LUTs built to HBM
                           // by k_pq_lut, scan stages them coalesced
                           // (-1 auto: on at m >= 32; 0 off; 1 force)
  int pq_lut_mb = 2048;  // spec "pq_lut_mb": LUT chunk budget (measured:
                         // bigger chunks win — launch concurrency beats
                         // LLC residency; sweep in BASELINE.md ladder)
  bool pq_lut_f16 = false;  // spec "pq_lut_f16": __half ADC tables —
                            // halves the LUT HBM round-trip AND the scan
                            // LDS (4 blocks/CU at m=64). APPROXIMATION
                            // (faiss GpuIndexIVFPQ useFloat16LookupTables
                            // equivalent): documented tolerance path,
                            // off by default; the exact path stays the
                            // parity contract.
  int scan_fan = 1;     // spec "scan_fan": list-segment fan (experiment)  // spec "coarse_bf16": assign/coarse GEMMs on
                             // bf16 MFMA (~16x f32 rate) — approximate
                             // ranking path for huge nlist

  DevBuf centroids, cnorm, codebooks, sq_vmin, sq_vdiff, sq_scale;
  // CSR code image (slab arena; csr_rows rows grouped by list) + pending
  // appends (slab arena in arrival order, merged in by rebuild_csr)
  SlabArena csr_arena, pend_arena;
  std::vector<int32_t> h_pend_assign;  // pending rows' list assignment
  int64_t csr_rows = 0;                // rows in the CSR image
  int merge_mb = 2048;  // spec "merge_mb": pending bytes that trigger an
                        // incremental merge during add (bounds the
                        // rebuild transient to ~2x this)
  DevBuf cr_ids, cr_off, id2pos, cr_ids_new;
  std::vector<int64_t> h_off;
  bool dirty = false;
  // HNSW (type "hnswsq"): graph over non-residual SQ8 codes in csr_arena
  // (arrival order, never rebuilt). M = h->m, deg0 = 2M.
  int hnsw_efc = 100;
  int hnsw_entry = -1, hnsw_maxlevel = -1, hnsw_nslots = 0;
  std::vector<int32_t> h_levels, h_upslot;
  DevBuf hn_levels, hn_nbr0, hn_cnt0, hn_upslot, hn_nbrU, hn_cntU;
  DevBuf hn_req, hn_reqcnt, hn_runoff, hn_u;
  // flat arena
  DevBuf flat;
  // workspace
  DevBuf ws1, ws2, ws3, ws4, ws5, ws_bf16a, ws_bf16b;
  DevBuf cent_bf16;
  DevBuf term2, term3_ws, qn_ws;  // PQ-L2 precomputed tables
  DevBuf pq_lut_ws;  // HBM ADC LUTs for the GLUT scan path

  // timing
  bool timing = false;
  std::vector<TimingEv> ev_scan, ev_gemm, ev_merge, ev_lut;
  int64_t scan_rows = 0, scan_bytes = 0, gemm_flops = 0;

  ~dfann_index() {
    for (auto &v : {ev_scan, ev_gemm, ev_merge, ev_lut})
      for (auto &e : v) {
        (void)hipEventDestroy(e.a);
        (void)hipEventDestroy(e.b);
      }
  }

  TimingEv ev_begin(hipStream_t s) {
    TimingEv e;
    HIP_CHECK(hipEventCreate(&e.a));
    HIP_CHECK(hipEventCreate(&e.b));
    HIP_CHECK(hipEventRecord(e.a, s));
    return e;
  }
  void ev_end(TimingEv e, hipStream_t s, std::vector<TimingEv> &dst) {
    HIP_CHECK(hipEventRecord(e.b, s));
    dst.push_back(e);
  }
};

// ---------------------------------------------------------------------------
// launch helpers
// ---------------------------------------------------------------------------

static bool use_regsel(int k) { return k <= 16; }

static dim3 grid1d(int64_t total, int block = 256, int64_t cap = 65535LL * 8) {
  int64_t g = (total + block - 1) / block;
  if (g > cap) g = cap;
  if (g < 1) g = 1;
  return dim3((unsigned)g);
}

// C[i][j] keys for rows of A vs rows of B under `metric`.
// mode: 0 -ip, 1 bn-2ip, 2 qn+bn-2ip (needs qn). Result in `keys` (rows x N).
// B_bf16: pre-converted B for the approximate bf16 path (used only when
// h->coarse_bf16 and the caller provides it).
static void gemm_keys_b(dfann_index *h, const float *A, int64_t Mrows,
                        const float *B, const unsigned short *B_bf16,
                        int64_t N, int K, const float *bn, const float *qn,
                        int mode, float *keys, hipStream_t stream) {
  dim3 g((unsigned)((N + GT - 1) / GT), (unsigned)((Mrows + GT - 1) / GT));
  TimingEv e;
  if (h && h->timing) e = h->ev_begin(stream);
  if (h && h->coarse_bf16 && B_bf16) {
    h->ws_bf16a.ensure((size_t)Mrows * K * 2);
    hipLaunchKernelGGL(k_f32_to_bf16, grid1d(Mrows * K), dim3(256), 0, stream,
                       A, Mrows * K, h->ws_bf16a.as<unsigned short>());
    if (K % 8 == 0 && N >= 2048 && Mrows >= 256 && K >= 64) {
      // big coarse/assign shapes: 256^2-tile glds kernel (512 threads).
      // DFANN_GEMM_P3=1 selects the BK=32 3-buffer counted-vmcnt ring
      // (k_gemm_bf16_256_p3) where K % 32 == 0 — experiment knob until
      // measured on hardware.
      dim3 g2((unsigned)((N + 255) / 256), (unsigned)((Mrows + 255) / 256));
      bool p3 = false;
      if (const char *e = getenv("DFANN_GEMM_P3"))
        p3 = atoi(e) == 1 && (K % 32) == 0;
      hipLaunchKernelGGL(p3 ? k_gemm_bf16_256_p3 : k_gemm_bf16_256, g2,
                         dim3(512), 0, stream,
                         h->ws_bf16a.as<unsigned short>(), B_bf16, keys,
                         (int)Mrows, (int)N, K, K, K, (int)N, qn, bn, mode);
    } else {
      auto bk = (K % 8 == 0) ? k_gemm_bf16_glds : k_gemm_bf16_nt;
      hipLaunchKernelGGL(bk, g, dim3(256), 0, stream,
                         h->ws_bf16a.as<unsigned short>(), B_bf16, keys,
                         (int)Mrows, (int)N, K, K, K, (int)N, qn, bn, mode);
    }
  } else {
    hipLaunchKernelGGL(k_gemm_nt, g, dim3(256), 0, stream, A, B, keys,
                       (int)Mrows, (int)N, K, K, K, (int)N, qn, bn, mode);
  }
  if (h && h->timing) {
    h->ev_end(e, stream, h->ev_gemm);
    h->gemm_flops += 2LL * Mrows * N * K;
  }
  HIP_CHECK(hipGetLastError());
}

static void gemm_keys(dfann_index *h, const float *A, int64_t Mrows,
                      const float *B, int64_t N, int K, const float *bn,
                      const float *qn, int mode, float *keys,
                      hipStream_t stream) {
  gemm_keys_b(h, A, Mrows, B, nullptr, N, K, bn, qn, mode, keys, stream);
}

// refresh derived images after (re)training / loading: bf16 centroids,
// PQ-L2 precomputed term2
static void refresh_cent_bf16(dfann_index *h, hipStream_t stream) {
  if (h->coarse_bf16 && h->centroids.p) {
    h->cent_bf16.ensure((size_t)h->nlist * h->d * 2);
    hipLaunchKernelGGL(k_f32_to_bf16, grid1d((int64_t)h->nlist * h->d),
                       dim3(256), 0, stream, h->centroids.as<float>(),
                       (long long)h->nlist * h->d,
                       h->cent_bf16.as<unsigned short>());
  }
  if (h->pq_pre && h->type == T_IVFPQ && h->metric == M_L2 &&
      h->centroids.p && h->codebooks.p) {
    h->term2.ensure((size_t)h->nlist * h->m * 256 * 4);
    hipLaunchKernelGGL(k_pq_term2, grid1d((int64_t)h->nlist * h->m * 256),
                       dim3(256), 0, stream, h->centroids.as<float>(),
                       h->codebooks.as<float>(), h->nlist, h->m, h->dsub,
                       h->term2.as<float>());
  }
  HIP_CHECK(hipGetLastError());
}

static void rownorms(const float *x, int64_t n, int d, float *out,
                     hipStream_t stream) {
  if (n == 0) return;
  hipLaunchKernelGGL(k_rownorm, dim3((unsigned)n), dim3(256), 0, stream, x, n, d,
                     out);
  HIP_CHECK(hipGetLastError());
}

// nearest-centroid assignment of (n x d) rows against h->centroids
static void assign_rows(dfann_index *h, const float *x, int64_t n,
                        int32_t *assign_dev, hipStream_t stream) {
  int nlist = h->nlist;
  // chunk points so the key matrix stays within the ws budget
  int64_t chunk = std::max<int64_t>(1, ((int64_t)h->ws_mb << 20) / ((int64_t)nlist * 4));
  chunk = std::min<int64_t>(chunk, n);
  h->ws1.ensure((size_t)chunk * nlist * 4);
  float *keys = h->ws1.as<float>();
  h->ws2.ensure((size_t)n * 4);
  float *bestv = h->ws2.as<float>();
  hipLaunchKernelGGL(k_assign_init, grid1d(n), dim3(256), 0, stream, bestv,
                     assign_dev, n);
  for (int64_t s = 0; s < n; s += chunk) {
    int64_t c = std::min(chunk, n - s);
    gemm_keys_b(h, x + s * h->d, c, h->centroids.as<float>(),
                h->cent_bf16.as<unsigned short>(), nlist, h->d,
                h->cnorm.as<float>(), nullptr, h->metric == M_IP ? 0 : 1, keys,
                stream);
    hipLaunchKernelGGL(k_assign_rowblock, dim3((unsigned)c), dim3(256), 0,
                       stream, keys, c, (long long)nlist, (long long)nlist, 0,
                       bestv + s, assign_dev + s);
  }
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// k-means (restating faiss Clustering; deviations shared with the oracle —
// oracle/__init__.py header): niter 25, strided subsample to k*256, seeded
// partial-FY init, metric-driven assignment, mean update, deterministic
// largest-donor empty-cluster split.
// ---------------------------------------------------------------------------

static void kmeans_device(dfann_index *h, const float *x, int64_t n, int kcent,
                          int d, int metric, uint64_t seed, float *cent_out,
                          hipStream_t stream) {
  const int NITER = 25;
  int64_t cap = (int64_t)kcent * (h ? h->max_ppc : 256);
  DevBuf xt_buf;
  const float *xt = x;
  int64_t nt = n;
  if (n > cap) {
    xt_buf.ensure((size_t)cap * d * 4);
    hipLaunchKernelGGL(k_gather_strided, grid1d(cap * d), dim3(256), 0, stream,
                       x, n, cap, d, xt_buf.as<float>());
    xt = xt_buf.as<float>();
    nt = cap;
  }
  trace_point("kmeans:subsample", stream);
  if (nt < kcent) throw std::runtime_error("kmeans: n < k");

  // init centroids
  auto idx = pick_init(nt, kcent, seed);
  DevBuf idx_buf;
  idx_buf.ensure(idx.size() * 8);
  HIP_CHECK(hipMemcpyAsync(idx_buf.p, idx.data(), idx.size() * 8,
                           hipMemcpyHostToDevice, stream));
  hipLaunchKernelGGL(k_gather_rows, grid1d((int64_t)kcent * d), dim3(256), 0,
                     stream, xt, idx_buf.as<int64_t>(), (long long)kcent, d,
                     cent_out);
  trace_point("kmeans:init", stream);

  DevBuf cn, asg, bestv, keys, sums, counts;
  cn.ensure((size_t)kcent * 4);
  asg.ensure((size_t)nt * 4);
  bestv.ensure((size_t)nt * 4);
  sums.ensure((size_t)kcent * d * 4);
  counts.ensure((size_t)kcent * 4);
  int64_t chunk = std::max<int64_t>(
      1, ((int64_t)(h ? h->ws_mb : 512) << 20) / ((int64_t)kcent * 4));
  chunk = std::min(chunk, nt);
  keys.ensure((size_t)chunk * kcent * 4);
  std::vector<int> h_counts(kcent);
  std::vector<float> h_cent;

  DevBuf cent_b16;
  bool use_b16 = h && h->coarse_bf16;
  if (use_b16) cent_b16.ensure((size_t)kcent * d * 2);
  for (int it = 0; it < NITER; ++it) {
    rownorms(cent_out, kcent, d, cn.as<float>(), stream);
    if (use_b16)
      hipLaunchKernelGGL(k_f32_to_bf16, grid1d((int64_t)kcent * d), dim3(256),
                         0, stream, cent_out, (long long)kcent * d,
                         cent_b16.as<unsigned short>());
    hipLaunchKernelGGL(k_assign_init, grid1d(nt), dim3(256), 0, stream,
                       bestv.as<float>(), asg.as<int>(), nt);
    for (int64_t s = 0; s < nt; s += chunk) {
      int64_t c = std::min(chunk, nt - s);
      gemm_keys_b(h, xt + s * d, c, cent_out,
                  use_b16 ? cent_b16.as<unsigned short>() : nullptr, kcent, d,
                  cn.as<float>(), nullptr, metric == M_IP ? 0 : 1,
                  keys.as<float>(), stream);
      hipLaunchKernelGGL(k_assign_rowblock, dim3((unsigned)c), dim3(256), 0,
                         stream, keys.as<float>(), c, (long long)kcent,
                         (long long)kcent, 0, bestv.as<float>() + s,
                         asg.as<int>() + s);
    }
    HIP_CHECK(hipMemsetAsync(sums.p, 0, (size_t)kcent * d * 4, stream));
    HIP_CHECK(hipMemsetAsync(counts.p, 0, (size_t)kcent * 4, stream));
    trace_point("kmeans:assign", stream);
    hipLaunchKernelGGL(k_centroid_accum, grid1d(nt * d), dim3(256), 0, stream,
                       xt, asg.as<int>(), nt, d, sums.as<float>(),
                       counts.as<int>());
    trace_point("kmeans:accum", stream);
    hipLaunchKernelGGL(k_centroid_div, grid1d((int64_t)kcent * d), dim3(256), 0,
                       stream, cent_out, sums.as<float>(), counts.as<int>(),
                       (long long)kcent, d);
    // empty clusters: deterministic largest-donor split (== oracle)
    HIP_CHECK(hipMemcpyAsync(h_counts.data(), counts.p, (size_t)kcent * 4,
                             hipMemcpyDeviceToHost, stream));
    HIP_CHECK(hipStreamSynchronize(stream));
    bool any_empty = false;
    for (int c = 0; c < kcent; ++c)
      if (h_counts[c] == 0) { any_empty = true; break; }
    if (any_empty) {
      h_cent.resize((size_t)kcent * d);
      HIP_CHECK(hipMemcpy(h_cent.data(), cent_out, (size_t)kcent * d * 4,
                          hipMemcpyDeviceToHost));
      std::vector<int> cw(h_counts.begin(), h_counts.end());
      const float eps = 1.0f / 1024.0f;
      for (int ci = 0; ci < kcent; ++ci) {
        if (h_counts[ci] != 0) continue;
        int cj = (int)(std::max_element(cw.begin(), cw.end()) - cw.begin());
        for (int t = 0; t < d; ++t) {
          float v = h_cent[(size_t)cj * d + t];
          h_cent[(size_t)ci * d + t] = v * (1.0f + eps);
          h_cent[(size_t)cj * d + t] = v * (1.0f - eps);
        }
        cw[ci] = cw[cj] / 2;
        cw[cj] -= cw[cj] / 2;
      }
      HIP_CHECK(hipMemcpy(cent_out, h_cent.data(), (size_t)kcent * d * 4,
                          hipMemcpyHostToDevice));
    }
  }
  xt_buf.free();
}

// ---------------------------------------------------------------------------
// create / train / add / finalize
// ---------------------------------------------------------------------------

static dfann_index *create_from_spec(const std::string &js) {
  auto *h = new dfann_index();
  h->spec_json = js;
  std::string t = json_str(js, "type", "");
  if (t == "flat") h->type = T_FLAT;
  else if (t == "ivf_flat") h->type = T_IVFFLAT;
  else if (t == "ivfpq") h->type = T_IVFPQ;
  else if (t == "ivfsq") h->type = T_IVFSQ;
  else if (t == "hnswsq") h->type = T_HNSW;
  else { delete h; throw std::runtime_error("unknown index type '" + t + "'"); }
  h->d = (int)json_int(js, "dim", 0);
  h->metric = (int)json_int(js, "metric", M_IP);
  h->nlist = (int)json_int(js, "nlist", 0);
  h->m = (int)json_int(js, "m", 0);
  h->nbits = (int)json_int(js, "nbits", 8);
  h->nprobe = (int)json_int(js, "nprobe", 1);
  h->seed = (uint64_t)json_int(js, "seed", 1234);
  h->sq8 = json_str(js, "sq_type", "fp16") == "8bit";
  h->ws_mb = (int)json_int(js, "ws_mb", 512);
  if (h->ws_mb < 1) h->ws_mb = 1;
  h->coarse_bf16 = json_int(js, "coarse_bf16", 0) != 0;
  h->max_ppc = (int)json_int(js, "max_ppc", 256);
  h->pq_pre = json_int(js, "pq_precomputed", 0) != 0;
  h->pq_lut_global = (int)json_int(js, "pq_lut_global", -1);
  h->pq_lut_mb = (int)json_int(js, "pq_lut_mb", 2048);
  if (h->pq_lut_mb < 1) h->pq_lut_mb = 1;
  h->pq_lut_f16 = json_int(js, "pq_lut_f16", 0) != 0;
  h->scan_fan = (int)json_int(js, "scan_fan", 1);
  if (h->scan_fan < 1) h->scan_fan = 1;
  if (h->scan_fan > 16) h->scan_fan = 16;
  if (h->d <= 0) { delete h; throw std::runtime_error("bad dim"); }
  if (h->type != T_FLAT && h->type != T_HNSW && h->nlist <= 0) {
    delete h;
    throw std::runtime_error("bad nlist");
  }
  if (h->type == T_IVFPQ) {
    if (h->nbits != 8) { delete h; throw std::runtime_error("only nbits=8"); }
    if (h->m <= 0 || h->d % h->m) {
      delete h;
      throw std::runtime_error("bad m (dim % m != 0)");
    }
    h->dsub = h->d / h->m;
    h->code_bytes = h->m;
  } else if (h->type == T_IVFSQ) {
    h->code_bytes = h->sq8 ? h->d : 2 * h->d;
  } else if (h->type == T_IVFFLAT) {
    h->code_bytes = 4 * h->d;
  } else if (h->type == T_HNSW) {
    // reference index.py:51-60: faiss.IndexHNSWSQ(dim, QT_8bit, store_n)
    // with L2 asserted; spec m = store_n (link cap M), nprobe = efSearch
    if (h->metric != M_L2)
      { delete h; throw std::runtime_error("hnswsq requires L2 (ref index.py:52)"); }
    if (h->m <= 0) h->m = 128;  // reference store_n default
    if (h->m > 256) {
      delete h;
      throw std::runtime_error(
          "hnswsq store_n > 256 unsupported (level-0 degree cap 2M is "
          "sized 512 in the merge kernels)");
    }
    h->hnsw_efc = (int)json_int(js, "ef_construction", 100);
    if (h->hnsw_efc < 1) h->hnsw_efc = 1;
    if (h->hnsw_efc > 512) h->hnsw_efc = 512;
    h->sq8 = true;
    h->code_bytes = h->d;
  } else {
    h->code_bytes = 4 * h->d;
    h->trained = true;  // flat needs no training
  }
  h->stride = round16(h->code_bytes);
  h->merge_mb = (int)json_int(js, "merge_mb", 2048);
  if (h->merge_mb < 1) h->merge_mb = 1;
  // 64 MB slabs: small enough that an empty index costs little, large
  // enough that the 1B-row table stays a few thousand L1-hot entries
  h->csr_arena.init(h->stride, (size_t)64 << 20);
  h->pend_arena.init(h->stride, (size_t)64 << 20);
  if (h->nlist > 0) h->h_off.assign(h->nlist + 1, 0);
  return h;
}

static void hnsw_train_ranges(dfann_index *h, int64_t n, const float *x,
                              hipStream_t stream) {
  DevBuf mn, mx;
  mn.ensure((size_t)h->d * 4);
  mx.ensure((size_t)h->d * 4);
  h->sq_vmin.ensure((size_t)h->d * 4);
  h->sq_vdiff.ensure((size_t)h->d * 4);
  h->sq_scale.ensure((size_t)h->d * 4);
  hipLaunchKernelGGL(k_minmax_init, grid1d(h->d), dim3(256), 0, stream,
                     mn.as<unsigned>(), mx.as<unsigned>(), h->d);
  hipLaunchKernelGGL(k_minmax_dims, grid1d(n * h->d), dim3(256), 0, stream, x,
                     n, h->d, mn.as<unsigned>(), mx.as<unsigned>());
  hipLaunchKernelGGL(k_minmax_decode, grid1d(h->d), dim3(256), 0, stream,
                     mn.as<unsigned>(), mx.as<unsigned>(), h->d,
                     h->sq_vmin.as<float>(), h->sq_vdiff.as<float>(),
                     h->sq_scale.as<float>());
  HIP_CHECK(hipStreamSynchronize(stream));
}

static void train_impl(dfann_index *h, int64_t n, const float *x,
                       hipStream_t stream) {
  if (h->type == T_FLAT) { h->trained = true; return; }
  if (h->trained) return;  // faiss semantics: re-train of trained is a no-op here
  if (h->type == T_HNSW) {
    // non-residual SQ8 codec: per-dim ranges over the raw training set
    // (faiss IndexHNSWSQ trains its storage IndexScalarQuantizer)
    hnsw_train_ranges(h, n, x, stream);
    h->trained = true;
    return;
  }
  h->centroids.ensure((size_t)h->nlist * h->d * 4);
  h->cnorm.ensure((size_t)h->nlist * 4);
  trace_point("train:begin", stream);
  kmeans_device(h, x, n, h->nlist, h->d, h->metric, h->seed,
                h->centroids.as<float>(), stream);
  trace_point("train:coarse-kmeans", stream);
  rownorms(h->centroids.as<float>(), h->nlist, h->d, h->cnorm.as<float>(),
           stream);
  refresh_cent_bf16(h, stream);  // bf16 image for the assign below
  if (h->type == T_IVFPQ || (h->type == T_IVFSQ && h->sq8)) {
    DevBuf asg, resid;
    asg.ensure((size_t)n * 4);
    resid.ensure((size_t)n * h->d * 4);
    assign_rows(h, x, n, asg.as<int>(), stream);
    trace_point("train:assign", stream);
    hipLaunchKernelGGL(k_residual, grid1d(n * h->d), dim3(256), 0, stream, x,
                       h->centroids.as<float>(), asg.as<int>(), n, h->d,
                       resid.as<float>());
    trace_point("train:residual", stream);
    if (h->type == T_IVFPQ) {
      h->codebooks.ensure((size_t)h->m * 256 * h->dsub * 4);
      DevBuf sub;
      sub.ensure((size_t)n * h->dsub * 4);
      for (int j = 0; j < h->m; ++j) {
        hipLaunchKernelGGL(k_subspace_slice, grid1d(n * h->dsub), dim3(256), 0,
                           stream, resid.as<float>(), n, h->d, j * h->dsub,
                           h->dsub, sub.as<float>());
        kmeans_device(h, sub.as<float>(), n, 256, h->dsub, M_L2,
                      h->seed + 1 + j,
                      h->codebooks.as<float>() + (size_t)j * 256 * h->dsub,
                      stream);
        trace_point("train:pq-sub", stream);
      }
    } else {  // SQ8 ranges on residuals
      DevBuf mn, mx;
      mn.ensure((size_t)h->d * 4);
      mx.ensure((size_t)h->d * 4);
      h->sq_vmin.ensure((size_t)h->d * 4);
      h->sq_vdiff.ensure((size_t)h->d * 4);
      h->sq_scale.ensure((size_t)h->d * 4);
      hipLaunchKernelGGL(k_minmax_init, grid1d(h->d), dim3(256), 0, stream,
                         mn.as<unsigned>(), mx.as<unsigned>(), h->d);
      hipLaunchKernelGGL(k_minmax_dims, grid1d(n * h->d), dim3(256), 0, stream,
                         resid.as<float>(), n, h->d, mn.as<unsigned>(),
                         mx.as<unsigned>());
      hipLaunchKernelGGL(k_minmax_decode, grid1d(h->d), dim3(256), 0, stream,
                         mn.as<unsigned>(), mx.as<unsigned>(), h->d,
                         h->sq_vmin.as<float>(), h->sq_vdiff.as<float>(),
                         h->sq_scale.as<float>());
    }
  }
  refresh_cent_bf16(h, stream);  // + term2 now that codebooks exist
  HIP_CHECK(hipStreamSynchronize(stream));
  h->trained = true;
}

// Merge [old CSR + pending appends] into a fresh CSR slab set.
// Order-preserving two-source merge: new list l = old rows of l (already
// in ascending-arrival order) followed by pending rows of l (arrival
// order via the counting-sort permutation), so CSR position order within
// a list stays == ascending arrival id (the scan tie-break contract).
// Old slabs are freed behind the write window: because new pos >= old
// pos for every old row, old slab s is fully consumed once the write
// cursor passes (s+1)*rps + pend_rows — peak memory stays
// ~codes + pending + one slab instead of the round-1 2x duplication.
static void rebuild_csr(dfann_index *h, hipStream_t stream) {
  if (!h->dirty) return;
  int64_t n_pend = (int64_t)h->h_pend_assign.size();
  int64_t n_old = h->csr_rows;
  int64_t n_new = n_old + n_pend;
  // pending counting sort (stable -> arrival order within each list)
  std::vector<int64_t> pend_off(h->nlist + 1, 0);
  for (int32_t a : h->h_pend_assign) pend_off[a + 1]++;
  for (int l = 0; l < h->nlist; ++l) pend_off[l + 1] += pend_off[l];
  std::vector<unsigned> psrc(std::max<int64_t>(n_pend, 1));
  {
    std::vector<int64_t> fill(pend_off.begin(), pend_off.end() - 1);
    for (int64_t i = 0; i < n_pend; ++i)
      psrc[fill[h->h_pend_assign[i]]++] = (unsigned)i;
  }
  std::vector<int64_t> old_off = h->h_off;
  if ((int64_t)old_off.size() != h->nlist + 1)
    old_off.assign(h->nlist + 1, 0);
  std::vector<int64_t> new_off(h->nlist + 1, 0);
  for (int l = 0; l < h->nlist; ++l)
    new_off[l + 1] = new_off[l] + (old_off[l + 1] - old_off[l]) +
                     (pend_off[l + 1] - pend_off[l]);
  // device tables
  DevBuf d_new_off, d_old_off, d_pend_off, d_psrc;
  d_new_off.ensure((size_t)(h->nlist + 1) * 8);
  d_old_off.ensure((size_t)(h->nlist + 1) * 8);
  d_pend_off.ensure((size_t)(h->nlist + 1) * 8);
  d_psrc.ensure((size_t)std::max<int64_t>(n_pend, 1) * 4);
  HIP_CHECK(hipMemcpyAsync(d_new_off.p, new_off.data(),
                           (size_t)(h->nlist + 1) * 8, hipMemcpyHostToDevice,
                           stream));
  HIP_CHECK(hipMemcpyAsync(d_old_off.p, old_off.data(),
                           (size_t)(h->nlist + 1) * 8, hipMemcpyHostToDevice,
                           stream));
  HIP_CHECK(hipMemcpyAsync(d_pend_off.p, pend_off.data(),
                           (size_t)(h->nlist + 1) * 8, hipMemcpyHostToDevice,
                           stream));
  if (n_pend)
    HIP_CHECK(hipMemcpyAsync(d_psrc.p, psrc.data(), (size_t)n_pend * 4,
                             hipMemcpyHostToDevice, stream));
  h->cr_ids_new.ensure((size_t)std::max<int64_t>(n_new, 1) * 8);
  h->id2pos.ensure((size_t)std::max<int64_t>(n_new, 1) * 4);
  SlabArena next;
  next.init(h->stride, h->csr_arena.slab_bytes());
  const int64_t rps = next.rps();
  int64_t id_base = n_old;  // pending row p arrived as id n_old + p
  size_t old_free_cursor = 0;
  const uint8_t *const *old_tab = h->csr_arena.dev_table(stream);
  const uint8_t *const *pend_tab = h->pend_arena.dev_table(stream);
  for (int64_t j0 = 0; j0 < n_new; j0 += rps) {
    int64_t j1 = std::min(n_new, j0 + rps);
    next.ensure_rows(j1);
    hipLaunchKernelGGL(k_rebuild_gather, grid1d(j1 - j0), dim3(256), 0, stream,
                       j0, j1, d_new_off.as<int64_t>(), d_old_off.as<int64_t>(),
                       d_pend_off.as<int64_t>(), d_psrc.as<unsigned>(),
                       old_tab, pend_tab, next.dev_table_mut(stream),
                       h->cr_ids.as<int64_t>(), id_base,
                       h->cr_ids_new.as<int64_t>(), h->id2pos.as<unsigned>(),
                       h->nlist, next.rlog, h->stride);
    HIP_CHECK(hipGetLastError());
    // recycle old slabs fully behind the window: every old row s maps to
    // new pos >= its old pos, so old pos < j1 - n_pend are consumed
    if (h->csr_arena.rlog == next.rlog) {
      int64_t consumed = j1 - n_pend;
      bool freed_any = false;
      while ((int64_t)((old_free_cursor + 1) << next.rlog) <= consumed &&
             old_free_cursor < h->csr_arena.slabs.size()) {
        if (!freed_any) {
          HIP_CHECK(hipStreamSynchronize(stream));
          freed_any = true;
        }
        h->csr_arena.free_slab(old_free_cursor++);
      }
    }
  }
  HIP_CHECK(hipStreamSynchronize(stream));
  h->csr_arena.reset();
  std::swap(h->csr_arena.stride, next.stride);
  std::swap(h->csr_arena.rlog, next.rlog);
  std::swap(h->csr_arena.slabs, next.slabs);
  h->csr_arena.rows = n_new;
  h->csr_arena.table_dirty = true;
  next.rows = 0;
  h->pend_arena.reset();
  h->pend_arena.init(h->stride, h->csr_arena.slab_bytes());
  h->h_pend_assign.clear();
  h->csr_rows = n_new;
  // swap id arrays (DevBuf members)
  std::swap(h->cr_ids.p, h->cr_ids_new.p);
  std::swap(h->cr_ids.cap, h->cr_ids_new.cap);
  h->h_off = new_off;
  h->cr_off.ensure((size_t)(h->nlist + 1) * 8);
  HIP_CHECK(hipMemcpy(h->cr_off.p, h->h_off.data(), (size_t)(h->nlist + 1) * 8,
                      hipMemcpyHostToDevice));
  h->dirty = false;
}

// encode a chunk of rows into the pending arena at rows [row0, row0+n)
static void encode_chunk(dfann_index *h, int64_t n, const float *x,
                         const int *asg, int64_t row0, hipStream_t stream) {
  h->pend_arena.ensure_rows(row0 + n);
  uint8_t *const *dst = h->pend_arena.dev_table_mut(stream);
  int rlog = h->pend_arena.rlog;
  if (h->type == T_IVFFLAT) {
    hipLaunchKernelGGL(k_pack_rows, grid1d(n * h->d), dim3(256), 0, stream, x,
                       n, h->d, h->stride, rlog, row0, dst);
    return;
  }
  DevBuf resid;
  resid.ensure((size_t)n * h->d * 4);
  hipLaunchKernelGGL(k_residual, grid1d(n * h->d), dim3(256), 0, stream, x,
                     h->centroids.as<float>(), asg, n, h->d,
                     resid.as<float>());
  if (h->type == T_IVFPQ) {
    // per-subspace encode as a distance GEMM + running argmin (~20x a
    // naive per-thread 256-way argmin kernel)
    DevBuf sub, best, bestv, cbn;
    sub.ensure((size_t)n * h->dsub * 4);
    best.ensure((size_t)n * 4);
    bestv.ensure((size_t)n * 4);
    cbn.ensure(256 * 4);
    int64_t chunk = std::max<int64_t>(1, ((int64_t)h->ws_mb << 20) / (256 * 4));
    chunk = std::min<int64_t>(chunk, n);
    h->ws1.ensure((size_t)chunk * 256 * 4);
    for (int j = 0; j < h->m; ++j) {
      hipLaunchKernelGGL(k_subspace_slice, grid1d(n * h->dsub), dim3(256), 0,
                         stream, resid.as<float>(), n, h->d, j * h->dsub,
                         h->dsub, sub.as<float>());
      const float *cbj = h->codebooks.as<float>() + (size_t)j * 256 * h->dsub;
      rownorms(cbj, 256, h->dsub, cbn.as<float>(), stream);
      hipLaunchKernelGGL(k_assign_init, grid1d(n), dim3(256), 0, stream,
                         bestv.as<float>(), best.as<int>(), n);
      for (int64_t s0 = 0; s0 < n; s0 += chunk) {
        int64_t c = std::min(chunk, n - s0);
        gemm_keys(nullptr, sub.as<float>() + s0 * h->dsub, c, cbj, 256,
                  h->dsub, cbn.as<float>(), nullptr, 1, h->ws1.as<float>(),
                  stream);
        hipLaunchKernelGGL(k_assign_rowblock, dim3((unsigned)c), dim3(256), 0,
                           stream, h->ws1.as<float>(), c, (long long)256,
                           (long long)256, 0, bestv.as<float>() + s0,
                           best.as<int>() + s0);
      }
      hipLaunchKernelGGL(k_codes_from_best, grid1d(n), dim3(256), 0, stream,
                         best.as<int>(), n, j, h->stride, rlog, row0, dst);
    }
  } else {
    hipLaunchKernelGGL(k_sq_encode, grid1d(n * h->d), dim3(256), 0, stream,
                       resid.as<float>(),
                       h->sq8 ? h->sq_vmin.as<float>() : nullptr,
                       h->sq8 ? h->sq_vdiff.as<float>() : nullptr, n, h->d,
                       h->stride, h->sq8 ? 0 : 1, rlog, row0, dst);
  }
  HIP_CHECK(hipGetLastError());
}

static void hnsw_add(dfann_index *h, int64_t n, const float *x,
                     hipStream_t stream);

static void add_impl(dfann_index *h, int64_t n, const float *x,
                     hipStream_t stream) {
  if (!h->trained) throw std::runtime_error("add on untrained index");
  if (n == 0) return;
  if (h->type == T_HNSW) { hnsw_add(h, n, x, stream); return; }
  if (h->type == T_FLAT) {
    h->flat.grow_keep(((size_t)h->ntotal + n) * h->d * 4,
                      (size_t)h->ntotal * h->d * 4);
    HIP_CHECK(hipMemcpyAsync(h->flat.as<float>() + (size_t)h->ntotal * h->d, x,
                             (size_t)n * h->d * 4, hipMemcpyDeviceToDevice,
                             stream));
    HIP_CHECK(hipStreamSynchronize(stream));
    h->ntotal += n;
    return;
  }
  // internal chunking: bounds the residual/assign transients for huge
  // ingests (a 125M x 128 add would otherwise stage a 64 GB fp32
  // residual buffer)
  const int64_t CH = std::max<int64_t>(
      65536, ((int64_t)h->ws_mb << 20) / std::max(4 * h->d, 1));
  DevBuf asg;
  for (int64_t s = 0; s < n; s += CH) {
    int64_t c = std::min(CH, n - s);
    asg.ensure((size_t)c * 4);
    assign_rows(h, x + s * h->d, c, asg.as<int>(), stream);
    int64_t row0 = (int64_t)h->h_pend_assign.size();
    encode_chunk(h, c, x + s * h->d, asg.as<int>(), row0, stream);
    size_t old = h->h_pend_assign.size();
    h->h_pend_assign.resize(old + c);
    HIP_CHECK(hipMemcpyAsync(h->h_pend_assign.data() + old, asg.p,
                             (size_t)c * 4, hipMemcpyDeviceToHost, stream));
    HIP_CHECK(hipStreamSynchronize(stream));
    h->ntotal += c;
    h->dirty = true;
    // incremental merge: bounds the pending arena (and so the rebuild
    // transient) to ~merge_mb
    if ((int64_t)h->h_pend_assign.size() * h->stride >=
        ((int64_t)h->merge_mb << 20))
      rebuild_csr(h, stream);
  }
}

static void finalize_csr(dfann_index *h, hipStream_t stream) {
  rebuild_csr(h, stream);
}


// ---------------------------------------------------------------------------
// HNSW host orchestration (type "hnswsq")
// ---------------------------------------------------------------------------

// deterministic level draw (splitmix64; clamped). The oracle does not
// replicate this (graph parity is via the engine's own dump).
static int hnsw_draw_level(uint64_t seed, int64_t id, int M) {
  SplitMix64 rng(seed ^ (uint64_t)(id * 0x9E3779B97F4A7C15ULL + 0x51ED2701ULL));
  double u = ((double)(rng.next() >> 11) + 1.0) * (1.0 / 9007199254740992.0);
  double mL = 1.0 / log((double)M);
  int l = (int)floor(-log(u) * mL);
  if (l < 0) l = 0;
  if (l > HNSW_MAXL) l = HNSW_MAXL;
  return l;
}

// drain + apply the reverse-link requests of one insertion wave in
// (dst, level, src) sorted order (deterministic)
static void hnsw_apply_reqs(dfann_index *h, hipStream_t stream,
                            size_t lds_apply, int64_t req_cap) {
  const int M = h->m, deg0 = 2 * h->m;
  int rcnt = 0;
  HIP_CHECK(hipMemcpyAsync(&rcnt, h->hn_reqcnt.p, 4, hipMemcpyDeviceToHost,
                           stream));
  HIP_CHECK(hipStreamSynchronize(stream));
  if (rcnt > req_cap)
    throw std::runtime_error("hnsw: reverse-link buffer overflow");
  if (rcnt <= 0) return;
  std::vector<int> hreq((size_t)rcnt * 4);
  HIP_CHECK(hipMemcpy(hreq.data(), h->hn_req.p, (size_t)rcnt * 16,
                      hipMemcpyDeviceToHost));
  std::vector<int64_t> order(rcnt);
  for (int i = 0; i < rcnt; ++i) order[i] = i;
  std::sort(order.begin(), order.end(), [&](int64_t a, int64_t b) {
    const int *ra = &hreq[(size_t)a * 4], *rb = &hreq[(size_t)b * 4];
    if (ra[0] != rb[0]) return ra[0] < rb[0];
    if (ra[2] != rb[2]) return ra[2] < rb[2];
    return ra[1] < rb[1];
  });
  std::vector<int> sorted((size_t)rcnt * 4);
  std::vector<int> runoff;
  for (int i = 0; i < rcnt; ++i) {
    const int *r = &hreq[(size_t)order[i] * 4];
    if (i == 0 || r[0] != sorted[(size_t)(i - 1) * 4] ||
        r[2] != sorted[(size_t)(i - 1) * 4 + 2])
      runoff.push_back(i);
    memcpy(&sorted[(size_t)i * 4], r, 16);
  }
  runoff.push_back(rcnt);
  int n_runs = (int)runoff.size() - 1;
  HIP_CHECK(hipMemcpy(h->hn_req.p, sorted.data(), (size_t)rcnt * 16,
                      hipMemcpyHostToDevice));
  h->hn_runoff.ensure(runoff.size() * 4);
  HIP_CHECK(hipMemcpy(h->hn_runoff.p, runoff.data(), runoff.size() * 4,
                      hipMemcpyHostToDevice));
  hipLaunchKernelGGL(k_hnsw_apply, dim3((unsigned)n_runs), dim3(256),
                     lds_apply, stream, h->sq_scale.as<float>(),
                     h->csr_arena.dev_table(stream), h->csr_arena.rlog,
                     h->stride, h->d, h->hn_nbr0.as<int>(),
                     h->hn_cnt0.as<int>(), h->hn_upslot.as<int>(),
                     h->hn_nbrU.as<int>(), h->hn_cntU.as<int>(), deg0, M,
                     h->hn_req.as<int>(), h->hn_runoff.as<int>(), n_runs);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipStreamSynchronize(stream));
}

static void hnsw_add(dfann_index *h, int64_t n, const float *x,
                     hipStream_t stream) {
  const int M = h->m, deg0 = 2 * h->m;
  const int64_t n0 = h->ntotal, ntot = n0 + n;
  // 1) encode rows [n0, ntot) into the code arena (arrival order; the
  //    HNSW arena is never rebuilt — ids ARE row positions)
  h->csr_arena.ensure_rows(ntot);
  {
    const int64_t CH = std::max<int64_t>(
        65536, ((int64_t)h->ws_mb << 20) / std::max(4 * h->d, 1));
    for (int64_t s0 = 0; s0 < n; s0 += CH) {
      int64_t c = std::min(CH, n - s0);
      hipLaunchKernelGGL(k_sq_encode, grid1d(c * h->d), dim3(256), 0, stream,
                         x + s0 * h->d, h->sq_vmin.as<float>(),
                         h->sq_vdiff.as<float>(), c, h->d, h->stride, 0,
                         h->csr_arena.rlog, n0 + s0,
                         h->csr_arena.dev_table_mut(stream));
    }
    HIP_CHECK(hipGetLastError());
  }
  // 2) levels + upper slots (host, deterministic)
  h->h_levels.resize(ntot);
  h->h_upslot.resize(ntot);
  int old_nslots = h->hnsw_nslots;
  for (int64_t i = n0; i < ntot; ++i) {
    int lv = hnsw_draw_level(h->seed, i, M);
    h->h_levels[i] = lv;
    h->h_upslot[i] = lv > 0 ? h->hnsw_nslots++ : -1;
  }
  // 3) grow device graph arrays (grow_keep preserves; new ranges zeroed
  //    or uploaded). Graph memory is deg0*4 B/node — the dominant HNSW
  //    cost, same as faiss's links storage.
  h->hn_levels.grow_keep((size_t)ntot * 4, (size_t)n0 * 4);
  HIP_CHECK(hipMemcpyAsync(h->hn_levels.as<int>() + n0,
                           h->h_levels.data() + n0, (size_t)n * 4,
                           hipMemcpyHostToDevice, stream));
  h->hn_upslot.grow_keep((size_t)ntot * 4, (size_t)n0 * 4);
  HIP_CHECK(hipMemcpyAsync(h->hn_upslot.as<int>() + n0,
                           h->h_upslot.data() + n0, (size_t)n * 4,
                           hipMemcpyHostToDevice, stream));
  h->hn_nbr0.grow_keep((size_t)ntot * deg0 * 4, (size_t)n0 * deg0 * 4);
  // -1-fill fresh adjacency: slots beyond each cnt stay deterministic
  // (dump comparisons, persistence) instead of hipMalloc garbage
  HIP_CHECK(hipMemsetAsync(h->hn_nbr0.as<int>() + (size_t)n0 * deg0, 0xFF,
                           (size_t)n * deg0 * 4, stream));
  h->hn_cnt0.grow_keep((size_t)ntot * 4, (size_t)n0 * 4);
  HIP_CHECK(hipMemsetAsync(h->hn_cnt0.as<int>() + n0, 0, (size_t)n * 4,
                           stream));
  if (h->hnsw_nslots > old_nslots) {
    h->hn_nbrU.grow_keep((size_t)h->hnsw_nslots * HNSW_MAXL * M * 4,
                         (size_t)old_nslots * HNSW_MAXL * M * 4);
    HIP_CHECK(hipMemsetAsync(
        h->hn_nbrU.as<int>() + (size_t)old_nslots * HNSW_MAXL * M, 0xFF,
        (size_t)(h->hnsw_nslots - old_nslots) * HNSW_MAXL * M * 4, stream));
    h->hn_cntU.grow_keep((size_t)h->hnsw_nslots * HNSW_MAXL * 4,
                         (size_t)old_nslots * HNSW_MAXL * 4);
    HIP_CHECK(hipMemsetAsync(
        h->hn_cntU.as<int>() + (size_t)old_nslots * HNSW_MAXL, 0,
        (size_t)(h->hnsw_nslots - old_nslots) * HNSW_MAXL * 4, stream));
  }
  // 4) wave insertion over frozen snapshots
  const int64_t WMAX = 4096;
  // x2: the refine pass queues both directions per kept link
  const int64_t req_cap = 2 * WMAX * M * (HNSW_MAXL + 1);
  h->hn_req.ensure((size_t)req_cap * 16);
  h->hn_reqcnt.ensure(4);
  h->hn_u.ensure((size_t)WMAX * h->d * 4);
  size_t lds_ins = HNSW_LDS_INS(h->d, 256);
  size_t lds_apply = HNSW_LDS_APPLY(h->d);
  std::vector<int> hreq;
  std::vector<int64_t> order;
  std::vector<int> runoff;
  int64_t w0 = n0;
  while (w0 < ntot) {
    if (h->hnsw_entry < 0) {  // very first point
      h->hnsw_entry = 0;
      h->hnsw_maxlevel = h->h_levels[0];
    }
    int64_t W = w0 == 0
                    ? 1
                    : std::min<int64_t>(
                          std::min(WMAX, std::max<int64_t>(256, w0 / 8)),
                          ntot - w0);
    W = std::min(W, ntot - w0);
    hipLaunchKernelGGL(k_hnsw_prep, grid1d(W * h->d), dim3(256), 0, stream,
                       x + (w0 - n0) * h->d, h->sq_vmin.as<float>(),
                       h->sq_scale.as<float>(), W, h->d, h->hn_u.as<float>());
    HIP_CHECK(hipMemsetAsync(h->hn_reqcnt.p, 0, 4, stream));
    hipLaunchKernelGGL(k_hnsw_insert, dim3((unsigned)W), dim3(256), lds_ins,
                       stream, h->hn_u.as<float>(), h->sq_scale.as<float>(),
                       h->csr_arena.dev_table(stream), h->csr_arena.rlog,
                       h->stride, h->d, h->hn_levels.as<int>(),
                       h->hn_nbr0.as<int>(), h->hn_cnt0.as<int>(),
                       h->hn_upslot.as<int>(), h->hn_nbrU.as<int>(),
                       h->hn_cntU.as<int>(), deg0, M, w0, h->hnsw_entry,
                       h->hnsw_maxlevel, w0, (int)W, h->hnsw_efc,
                       h->hn_req.as<int>(), h->hn_reqcnt.as<int>(),
                       (int)req_cap, 0, (const int *)nullptr,
                       (const int *)nullptr, (const int *)nullptr,
                       (const int *)nullptr);
    HIP_CHECK(hipGetLastError());
    hnsw_apply_reqs(h, stream, lds_apply, req_cap);
    // entry update: highest new level (lowest id on ties) beats the old
    for (int64_t i = w0; i < w0 + W; ++i)
      if (h->h_levels[i] > h->hnsw_maxlevel) {
        h->hnsw_maxlevel = h->h_levels[i];
        h->hnsw_entry = (int)i;
      }
    w0 += W;
  }
  h->csr_arena.rows = ntot;
  h->ntotal = ntot;
  // REFINEMENT pass: re-link every new point over the
  // FINAL graph — the initial waves insert over stale snapshots (a
  // point cannot link to its own wave, and early points saw a tiny
  // graph). One extra pass raises mean degree and recall substantially
  // (measured in scripts/hnsw_diag.py); deterministic: fixed wave order
  // over the evolving-but-deterministic graph. spec "hnsw_refine" = 0
  // disables.
  if (json_int(h->spec_json, "hnsw_refine", 1) != 0) {
    // freeze the pre-refine graph: refine blocks traverse the copy
    // while own-link writes and reverse-link merges go to the live
    // arrays — race-free and deterministic
    DevBuf fz_nbr0, fz_cnt0, fz_nbrU, fz_cntU;
    fz_nbr0.ensure((size_t)ntot * deg0 * 4);
    fz_cnt0.ensure((size_t)ntot * 4);
    HIP_CHECK(hipMemcpyAsync(fz_nbr0.p, h->hn_nbr0.p, (size_t)ntot * deg0 * 4,
                             hipMemcpyDeviceToDevice, stream));
    HIP_CHECK(hipMemcpyAsync(fz_cnt0.p, h->hn_cnt0.p, (size_t)ntot * 4,
                             hipMemcpyDeviceToDevice, stream));
    if (h->hnsw_nslots) {
      fz_nbrU.ensure((size_t)h->hnsw_nslots * HNSW_MAXL * M * 4);
      fz_cntU.ensure((size_t)h->hnsw_nslots * HNSW_MAXL * 4);
      HIP_CHECK(hipMemcpyAsync(fz_nbrU.p, h->hn_nbrU.p,
                               (size_t)h->hnsw_nslots * HNSW_MAXL * M * 4,
                               hipMemcpyDeviceToDevice, stream));
      HIP_CHECK(hipMemcpyAsync(fz_cntU.p, h->hn_cntU.p,
                               (size_t)h->hnsw_nslots * HNSW_MAXL * 4,
                               hipMemcpyDeviceToDevice, stream));
    } else {
      fz_nbrU.ensure(16);
      fz_cntU.ensure(16);
    }
    int64_t r0 = n0;
    while (r0 < ntot) {
      int64_t W = std::min<int64_t>(WMAX, ntot - r0);
      hipLaunchKernelGGL(k_hnsw_prep, grid1d(W * h->d), dim3(256), 0, stream,
                         x + (r0 - n0) * h->d, h->sq_vmin.as<float>(),
                         h->sq_scale.as<float>(), W, h->d,
                         h->hn_u.as<float>());
      HIP_CHECK(hipMemsetAsync(h->hn_reqcnt.p, 0, 4, stream));
      hipLaunchKernelGGL(k_hnsw_insert, dim3((unsigned)W), dim3(256), lds_ins,
                         stream, h->hn_u.as<float>(), h->sq_scale.as<float>(),
                         h->csr_arena.dev_table(stream), h->csr_arena.rlog,
                         h->stride, h->d, h->hn_levels.as<int>(),
                         h->hn_nbr0.as<int>(), h->hn_cnt0.as<int>(),
                         h->hn_upslot.as<int>(), h->hn_nbrU.as<int>(),
                         h->hn_cntU.as<int>(), deg0, M, ntot, h->hnsw_entry,
                         h->hnsw_maxlevel, r0, (int)W, h->hnsw_efc,
                         h->hn_req.as<int>(), h->hn_reqcnt.as<int>(),
                         (int)req_cap, 1, fz_nbr0.as<const int>(),
                         fz_cnt0.as<const int>(), fz_nbrU.as<const int>(),
                         fz_cntU.as<const int>());
      HIP_CHECK(hipGetLastError());
      hnsw_apply_reqs(h, stream, lds_apply, req_cap);
      r0 += W;
    }
  }
}

static void hnsw_search(dfann_index *h, int64_t nq, const float *q, int k,
                        float *D, int64_t *I, hipStream_t stream) {
  int ef = std::max(h->nprobe, k);
  if (ef > 512) ef = 512;
  h->hn_u.ensure((size_t)nq * h->d * 4);
  hipLaunchKernelGGL(k_hnsw_prep, grid1d(nq * h->d), dim3(256), 0, stream, q,
                     h->sq_vmin.as<float>(), h->sq_scale.as<float>(), nq, h->d,
                     h->hn_u.as<float>());
  TimingEv e;
  if (h->timing) e = h->ev_begin(stream);
  hipLaunchKernelGGL(k_hnsw_search, dim3((unsigned)nq), dim3(256),
                     HNSW_LDS(h->d, 256), stream, h->hn_u.as<float>(),
                     h->sq_scale.as<float>(), h->csr_arena.dev_table(stream),
                     h->csr_arena.rlog, h->stride, h->d,
                     h->hn_levels.as<int>(), h->hn_nbr0.as<int>(),
                     h->hn_cnt0.as<int>(), h->hn_upslot.as<int>(),
                     h->hn_nbrU.as<int>(), h->hn_cntU.as<int>(), 2 * h->m,
                     h->m, h->ntotal, h->hnsw_entry, h->hnsw_maxlevel, nq, ef,
                     k, D, I);
  if (h->timing) h->ev_end(e, stream, h->ev_scan);
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// search
// ---------------------------------------------------------------------------

static void pad_fill(dfann_index *h, int64_t nq, int k, float *D, int64_t *I,
                     hipStream_t stream) {
  // empty index: all pads — run the merge kernel on zero candidates
  hipLaunchKernelGGL(k_merge_cand, dim3((unsigned)nq), dim3(256), SEL_LDS_BYTES,
                     stream, (const float *)nullptr, (const unsigned *)nullptr,
                     nq, 0, k, (const int64_t *)nullptr,
                     h->metric == M_IP ? 1 : 0, D, I);
  HIP_CHECK(hipGetLastError());
}

static void coarse_impl(dfann_index *h, int64_t nq, const float *q, int nprobe,
                        int32_t *probes, float *keys, hipStream_t stream) {
  int nlist = h->nlist;
  int64_t chunk = std::max<int64_t>(1, ((int64_t)h->ws_mb << 20) / ((int64_t)nlist * 4));
  chunk = std::min<int64_t>(chunk, nq);
  h->ws1.ensure((size_t)chunk * nlist * 4);
  float *sc = h->ws1.as<float>();
  for (int64_t s = 0; s < nq; s += chunk) {
    int64_t c = std::min(chunk, nq - s);
    gemm_keys_b(h, q + s * h->d, c, h->centroids.as<float>(),
                h->cent_bf16.as<unsigned short>(), nlist, h->d,
                h->cnorm.as<float>(), nullptr, h->metric == M_IP ? 0 : 1, sc,
                stream);
    if (use_regsel(nprobe)) {
      // 256 measured best (128 is -3%: this kernel is one streaming
      // phase — no phase diversity to recover, unlike the scans)
      unsigned tk_bs = 256;
      if (const char *e = getenv("DFANN_TOPK_BS"))  // experiment override
        if (int v = atoi(e)) tk_bs = (unsigned)((v / 64) * 64);
      hipLaunchKernelGGL(k_topk_rows_rk, dim3((unsigned)c), dim3(tk_bs),
                         REGSEL_LDS_BYTES, stream, sc, c, (long long)nlist,
                         (long long)nlist, nprobe, 0u, (long long)nprobe,
                         keys + s * nprobe,
                         reinterpret_cast<unsigned *>(probes) + s * nprobe);
    } else {
      hipLaunchKernelGGL(k_topk_rows, dim3((unsigned)c), dim3(256),
                         SEL_LDS_BYTES, stream, sc, c, (long long)nlist,
                         (long long)nlist, nprobe, 0u, (long long)nprobe,
                         keys + s * nprobe,
                         reinterpret_cast<unsigned *>(probes) + s * nprobe);
    }
  }
  HIP_CHECK(hipGetLastError());
}

static void scan_and_merge(dfann_index *h, int64_t nq, const float *q,
                           int nprobe, const int32_t *probes,
                           const float *keys, int k, float *D, int64_t *I,
                           hipStream_t stream) {
  // candidate arrays (sized after `fan` is chosen below)
  float *cand_d;
  unsigned *cand_p;
  int fam_floats;
  void (*kern)(const float *, const float *, const float *, const float *,
               const float *, const int *, const float *,
               const uint8_t *const *, const int64_t *, int, int, int, int,
               int, int, int, int, float *, unsigned *, int, int) = nullptr;
  bool ip = h->metric == M_IP;
  bool rk = use_regsel(k);
  switch (h->type) {
    case T_IVFPQ:
      fam_floats = h->m * 256 + h->d;
      kern = rk ? (ip ? k_scan_pq_ip_rk : k_scan_pq_l2_rk)
                : (ip ? k_scan_pq_ip : k_scan_pq_l2);
      break;
    case T_IVFFLAT:
      fam_floats = h->d;
      kern = rk ? (ip ? k_scan_ivfflat_ip_rk : k_scan_ivfflat_l2_rk)
                : (ip ? k_scan_ivfflat_ip : k_scan_ivfflat_l2);
      break;
    case T_IVFSQ:
      if (h->sq8) {
        fam_floats = 2 * h->d;
        kern = rk ? (ip ? k_scan_sq8_ip_rk : k_scan_sq8_l2_rk)
                  : (ip ? k_scan_sq8_ip : k_scan_sq8_l2);
        // DFANN_SCAN_NT=1: non-temporal code-row loads (A/B experiment)
        if (rk)
          if (const char *e = getenv("DFANN_SCAN_NT"))
            if (atoi(e) == 1)
              kern = ip ? k_scan_sq8_ip_rk_nt : k_scan_sq8_l2_rk_nt;
        // DFANN_SCAN_U8=1: 8 rows in flight per wave group (A/B)
        if (rk)
          if (const char *e = getenv("DFANN_SCAN_U8"))
            if (atoi(e) == 1)
              kern = ip ? k_scan_sq8_ip_rk_u8 : k_scan_sq8_l2_rk_u8;
      } else {
        fam_floats = h->d;
        kern = rk ? (ip ? k_scan_sqf_ip_rk : k_scan_sqf_l2_rk)
                  : (ip ? k_scan_sqf_ip : k_scan_sqf_l2);
      }
      break;
    default:
      throw std::runtime_error("scan on flat index");
  }
  bool use_pre = h->type == T_IVFPQ && h->metric == M_L2 && h->pq_pre &&
                 h->term2.p;
  if (use_pre) fam_floats = h->m * 256;  // LUT only, no rbuf
  // GLUT path: LUTs built once to HBM by k_pq_lut, scan stages them as
  // coalesced float4 slabs. Auto at m >= 32: the in-kernel build
  // re-reads the full m*256*dsub codebook from L2 per (query, probe)
  // block and dominates the launch (measured 8.1 ms/launch at the
  // configs[3] shape, ~125 GB/s of algorithmic code bytes).
  int glut_knob = h->pq_lut_global;
  if (const char *e = getenv("DFANN_PQ_LUT_GLOBAL"))  // experiment override
    glut_knob = atoi(e);
  bool lut_f16 = h->pq_lut_f16;
  if (const char *e = getenv("DFANN_PQ_LUT_F16"))  // experiment override
    lut_f16 = atoi(e) != 0;
  // f32 GLUT measured a wash vs the in-kernel build (BASELINE.md ladder),
  // so auto engages only for the fp16-LUT approximation path where the
  // halved round-trip wins; =1 still forces the f32 variant for A/B.
  bool use_glut = h->type == T_IVFPQ && !use_pre && h->dsub <= 64 &&
                  (glut_knob == 1 || (glut_knob != 0 && lut_f16));
  lut_f16 = lut_f16 && use_glut;
  if (use_glut)  // LUT only, no rbuf; fam_floats is in FLOAT units
    fam_floats = lut_f16 ? h->m * 128 : h->m * 256;
  // segment fan (spec "scan_fan"): kept as an experiment knob — measured
  // NEGATIVE at the 10M SQ8 shape (per-block staging/extraction overhead
  // outweighs tail imbalance: 1936 -> 1339 GB/s at fan 8), so default 1.
  int fan = h->type != T_IVFPQ ? h->scan_fan : 1;
  // REGSEL extract scratch aliases the fam region (used only after the
  // scan, behind a barrier) -> max, not sum (kernels.hip ivf_scan_body)
  size_t lds = rk ? std::max((size_t)fam_floats * 4, (size_t)REGSEL_LDS_BYTES)
                  : (size_t)fam_floats * 4 + SEL_LDS_BYTES;
  if (lds > 160 * 1024)
    throw std::runtime_error("scan LDS over budget (m too large)");
  // big-LDS blocks (m=64 LUTs: ~69 KB -> 2 blocks/CU) run 512 threads so
  // the CU still holds 4 waves/SIMD (register path only; the LDS-buffer
  // selection path is capacity-sized for 256)
  // Register-topk scans: SMALL blocks win (measured sweeps, BASELINE.md
  // ladder) — more independent blocks per CU de-correlates the
  // per-block phases (staging burst -> rows -> extract) and keeps the
  // memory system busy (the SQ wait decomposition showed 53% of scan
  // wave cycles parked): 1M f16 GLUT (8 KB LDS) 128 threads =
  // 16 blocks/CU (+27% step); configs[3] f16 (32 KB) 256 = 5 blocks
  // (+8% over 384); SQ8 10M 128 threads +8%. The LDS-buffer selection
  // path keeps 256 (Sel capacity is sized for it).
  unsigned scan_bs = 256;
  // round-2 addition: SQ scans drop to 64-thread blocks — measured +4%
  // on the config-5 125M SQ8 scan (2075 vs 1991 GB/s, gpurun_out/r2j_*)
  // and +6% at 10M; even more independent per-CU phases than the r1
  // 128-thread finding. IVF-Flat measured −5% at 64 (r2z_ivfflat), so
  // the drop is SQ-only.
  if (rk) scan_bs = lds <= 16 * 1024 ? 128 : (lds <= 32 * 1024 ? 256 : 512);
  if (rk && h->type == T_IVFSQ && lds <= 4 * 1024) scan_bs = 64;
  if (const char *e = getenv("DFANN_SCAN_BS"))  // experiment override
    if (int v = atoi(e)) scan_bs = (unsigned)((v / 64) * 64);
  h->ws3.ensure((size_t)nq * nprobe * fan * k * 4);
  h->ws4.ensure((size_t)nq * nprobe * fan * k * 4);
  cand_d = h->ws3.as<float>();
  cand_p = h->ws4.as<unsigned>();
  TimingEv e;
  if (use_pre) {
    // per-batch tables: q norms + term3 (counted as LUT-build time)
    TimingEv el;
    if (h->timing) el = h->ev_begin(stream);
    h->qn_ws.ensure((size_t)nq * 4);
    h->term3_ws.ensure((size_t)nq * h->m * 256 * 4);
    rownorms(q, nq, h->d, h->qn_ws.as<float>(), stream);
    hipLaunchKernelGGL(k_pq_term3, grid1d(nq * (int64_t)h->m * 256), dim3(256),
                       0, stream, q, h->codebooks.as<float>(), nq, h->m,
                       h->dsub, h->term3_ws.as<float>());
    if (h->timing) h->ev_end(el, stream, h->ev_lut);
    if (h->timing) e = h->ev_begin(stream);
    auto pk = rk ? k_scan_pq_l2_pre_rk : k_scan_pq_l2_pre;
    hipLaunchKernelGGL(pk, dim3((unsigned)(nq * nprobe)), dim3(scan_bs), lds,
                       stream, q, h->centroids.as<float>(),
                       h->codebooks.as<float>(), h->sq_vmin.as<float>(),
                       h->sq_scale.as<float>(), probes, keys,
                       h->csr_arena.dev_table(stream),
                       h->cr_off.as<int64_t>(),
                       (int)nq, nprobe, h->d, h->m, h->dsub, k, h->stride,
                       h->csr_arena.rlog,
                       cand_d, cand_p, fam_floats, h->term2.as<float>(),
                       h->term3_ws.as<float>(), h->qn_ws.as<float>());
    if (h->timing) h->ev_end(e, stream, h->ev_scan);
  } else if (use_glut) {
    auto gk = lut_f16 ? (rk ? (ip ? k_scan_pq_ip_gh_rk : k_scan_pq_l2_gh_rk)
                            : (ip ? k_scan_pq_ip_gh : k_scan_pq_l2_gh))
                      : (rk ? (ip ? k_scan_pq_ip_g_rk : k_scan_pq_l2_g_rk)
                            : (ip ? k_scan_pq_ip_g : k_scan_pq_l2_g));
    // chunk queries so the LUT buffer stays inside the budget (measured:
    // bigger chunks win — concurrency beats LLC residency)
    size_t row_b =
        (size_t)nprobe * h->m * 256 * (lut_f16 ? 2 : 4);  // bytes/query
    int lut_mb = h->pq_lut_mb;
    if (const char *e = getenv("DFANN_PQ_LUT_MB"))  // experiment override
      if (int v = atoi(e)) lut_mb = v;
    int budget_mb = std::min(h->ws_mb, lut_mb);
    int64_t qch =
        std::max<int64_t>(1, (int64_t)(((size_t)budget_mb << 20) / row_b));
    if (qch > nq) qch = nq;
    h->pq_lut_ws.ensure((size_t)qch * row_b);
    float *lutg = h->pq_lut_ws.as<float>();
    int pad = h->dsub | 1;
    size_t lut_lds = (size_t)(256 + PQ_LUT_QPT) * pad * 4;
    for (int64_t q0 = 0; q0 < nq; q0 += qch) {
      int64_t nqc = std::min<int64_t>(qch, nq - q0);
      long long qpn = (long long)nqc * nprobe;
      dim3 lg((unsigned)((qpn + PQ_LUT_QPT - 1) / PQ_LUT_QPT),
              (unsigned)h->m);
      TimingEv el;
      if (h->timing) el = h->ev_begin(stream);
      if (lut_f16) {
        hipLaunchKernelGGL(k_pq_lut_f16, lg, dim3(256), lut_lds, stream,
                           q + q0 * h->d, h->centroids.as<float>(),
                           h->codebooks.as<float>(), probes + q0 * nprobe,
                           (int)nqc, nprobe, h->d, h->m, h->dsub, ip ? 1 : 0,
                           h->pq_lut_ws.as<__half>());
      } else {
        hipLaunchKernelGGL(k_pq_lut, lg, dim3(256), lut_lds, stream,
                           q + q0 * h->d, h->centroids.as<float>(),
                           h->codebooks.as<float>(), probes + q0 * nprobe,
                           (int)nqc, nprobe, h->d, h->m, h->dsub, ip ? 1 : 0,
                           lutg);
      }
      if (h->timing) h->ev_end(el, stream, h->ev_lut);
      if (h->timing) e = h->ev_begin(stream);
      // persistent multi-pair variant (fp16 LUT + register top-k): a
      // fixed grid strides over the (query, probe) pairs and prefetches
      // the next pair's LUT during the current scan. MEASURED NEGATIVE
      // at the headline shape (gpurun_out/r2s_pers*.json: scan 5.7 ->
      // 6.9 ms at 2048 blocks — the hardware's cross-block overlap of
      // 80k one-pair blocks beats the in-block pipeline), so OFF by
      // default; DFANN_SCAN_PERS=1 re-enables for experiments.
      // Bit-identical results either way (r2s_pers_check.log).
      int pfn = 0;
      long long qpn_c = (long long)nqc * nprobe;
      const char *pers_env = getenv("DFANN_SCAN_PERS");
      if (lut_f16 && rk && pers_env && atoi(pers_env) == 1) {
        int lut_u4 = h->m * 32;  // m*256 halves / 8 per uint4
        if (lut_u4 % (int)scan_bs == 0) {
          int cand = lut_u4 / (int)scan_bs;
          if (cand == 2 || cand == 4 || cand == 8 || cand == 16) pfn = cand;
        }
      }
      if (pfn) {
        auto pk = ip ? (pfn == 2 ? k_scan_pq_ip_ghp2
                        : pfn == 4 ? k_scan_pq_ip_ghp4
                        : pfn == 8 ? k_scan_pq_ip_ghp8 : k_scan_pq_ip_ghp16)
                     : (pfn == 2 ? k_scan_pq_l2_ghp2
                        : pfn == 4 ? k_scan_pq_l2_ghp4
                        : pfn == 8 ? k_scan_pq_l2_ghp8 : k_scan_pq_l2_ghp16);
        long long pb = 2048;
        if (const char *e2 = getenv("DFANN_SCAN_PERS_BLOCKS"))
          if (atoll(e2) > 0) pb = atoll(e2);
        if (pb > qpn_c) pb = qpn_c;
        hipLaunchKernelGGL(pk, dim3((unsigned)pb), dim3(scan_bs), lds, stream,
                           q + q0 * h->d, h->centroids.as<float>(),
                           h->codebooks.as<float>(), h->sq_vmin.as<float>(),
                           h->sq_scale.as<float>(), probes + q0 * nprobe,
                           keys + q0 * nprobe, h->csr_arena.dev_table(stream),
                           h->cr_off.as<int64_t>(), (int)nqc, nprobe, h->d,
                           h->m, h->dsub, k, h->stride, h->csr_arena.rlog,
                           cand_d + q0 * nprobe * k, cand_p + q0 * nprobe * k,
                           fam_floats, lutg);
      } else {
        hipLaunchKernelGGL(gk, dim3((unsigned)(nqc * nprobe)), dim3(scan_bs),
                           lds, stream, q + q0 * h->d, h->centroids.as<float>(),
                           h->codebooks.as<float>(), h->sq_vmin.as<float>(),
                           h->sq_scale.as<float>(), probes + q0 * nprobe,
                           keys + q0 * nprobe, h->csr_arena.dev_table(stream),
                           h->cr_off.as<int64_t>(), (int)nqc, nprobe, h->d,
                           h->m, h->dsub, k, h->stride, h->csr_arena.rlog,
                           cand_d + q0 * nprobe * k, cand_p + q0 * nprobe * k,
                           fam_floats, lutg);
      }
      if (h->timing) h->ev_end(e, stream, h->ev_scan);
    }
  } else {
    if (h->timing) e = h->ev_begin(stream);
    hipLaunchKernelGGL(kern, dim3((unsigned)(nq * nprobe * fan)),
                       dim3(scan_bs), lds, stream, q, h->centroids.as<float>(),
                       h->codebooks.as<float>(), h->sq_vmin.as<float>(),
                       h->sq_scale.as<float>(), probes, keys,
                       h->csr_arena.dev_table(stream), h->cr_off.as<int64_t>(),
                       (int)nq, nprobe, h->d, h->m, h->dsub, k, h->stride,
                       h->csr_arena.rlog, cand_d, cand_p, fam_floats, fan);
    if (h->timing) h->ev_end(e, stream, h->ev_scan);
  }
  if (h->timing) {
    // algorithmic units: sum of probed list lengths
    std::vector<int32_t> hp((size_t)nq * nprobe);
    HIP_CHECK(hipMemcpy(hp.data(), probes, hp.size() * 4, hipMemcpyDeviceToHost));
    int64_t rows = 0;
    for (int32_t L : hp) rows += h->h_off[L + 1] - h->h_off[L];
    h->scan_rows += rows;
    h->scan_bytes += rows * h->stride;
  }
  TimingEv em;
  if (h->timing) em = h->ev_begin(stream);
  if (use_regsel(k)) {
    hipLaunchKernelGGL(k_merge_cand_rk, dim3((unsigned)nq), dim3(256),
                       REGSEL_LDS_BYTES + 128, stream, cand_d, cand_p, nq,
                       nprobe * fan * k, k, h->cr_ids.as<int64_t>(), ip ? 1 : 0,
                       D, I);
  } else {
    hipLaunchKernelGGL(k_merge_cand, dim3((unsigned)nq), dim3(256),
                       SEL_LDS_BYTES, stream, cand_d, cand_p, nq,
                       nprobe * fan * k, k, h->cr_ids.as<int64_t>(), ip ? 1 : 0,
                       D, I);
  }
  if (h->timing) h->ev_end(em, stream, h->ev_merge);
  HIP_CHECK(hipGetLastError());
}

static void flat_search_impl(dfann_index *h, int64_t nq, const float *q, int k,
                             float *D, int64_t *I, hipStream_t stream) {
  if (h->ntotal == 0) { pad_fill(h, nq, k, D, I, stream); return; }
  bool ip = h->metric == M_IP;
  const int64_t CH = std::min<int64_t>(65536, std::max<int64_t>(
      4096, ((int64_t)h->ws_mb << 20) / (64 * 4)));
  int64_t nch = (h->ntotal + CH - 1) / CH;
  // chunk queries so scores fit
  int64_t qch = std::max<int64_t>(1, ((int64_t)h->ws_mb << 20) / (CH * 4));
  qch = std::min(qch, nq);
  h->ws1.ensure((size_t)qch * CH * 4);
  h->ws2.ensure((size_t)std::max<int64_t>(CH, nq) * 4);            // bnorm / qnorm
  h->ws3.ensure((size_t)nq * nch * k * 4);                         // chunk winners d
  h->ws4.ensure((size_t)nq * nch * k * 4);                         // chunk winners p
  h->ws5.ensure((size_t)nq * 4);                                   // qnorm
  float *sc = h->ws1.as<float>();
  float *bn = h->ws2.as<float>();
  float *qn = h->ws5.as<float>();
  float *wd = h->ws3.as<float>();
  unsigned *wp = h->ws4.as<unsigned>();
  if (!ip) rownorms(q, nq, h->d, qn, stream);
  for (int64_t ci = 0; ci < nch; ++ci) {
    int64_t b0 = ci * CH;
    int64_t bn_rows = std::min(CH, h->ntotal - b0);
    const float *base = h->flat.as<float>() + (size_t)b0 * h->d;
    if (!ip) rownorms(base, bn_rows, h->d, bn, stream);
    for (int64_t s = 0; s < nq; s += qch) {
      int64_t c = std::min(qch, nq - s);
      gemm_keys(h, q + s * h->d, c, base, bn_rows, h->d, bn, qn + s,
                ip ? 0 : 2, sc, stream);
      // winners layout: (nq, nch, k) — row stride nch*k
      if (use_regsel(k)) {
        hipLaunchKernelGGL(k_topk_rows_rk, dim3((unsigned)c), dim3(256),
                           REGSEL_LDS_BYTES, stream, sc, c, (long long)bn_rows,
                           (long long)bn_rows, k, (unsigned)b0,
                           (long long)(nch * k), wd + (s * nch + ci) * k,
                           wp + (s * nch + ci) * k);
      } else {
        hipLaunchKernelGGL(k_topk_rows, dim3((unsigned)c), dim3(256),
                           SEL_LDS_BYTES, stream, sc, c, (long long)bn_rows,
                           (long long)bn_rows, k, (unsigned)b0,
                           (long long)(nch * k), wd + (s * nch + ci) * k,
                           wp + (s * nch + ci) * k);
      }
    }
  }
  // merge chunk winners
  if (use_regsel(k)) {
    hipLaunchKernelGGL(k_merge_cand_rk, dim3((unsigned)nq), dim3(256),
                       REGSEL_LDS_BYTES + 128, stream, wd, wp, nq,
                       (int)(nch * k), k, (const int64_t *)nullptr,
                       ip ? 1 : 0, D, I);
  } else {
    hipLaunchKernelGGL(k_merge_cand, dim3((unsigned)nq), dim3(256),
                       SEL_LDS_BYTES, stream, wd, wp, nq, (int)(nch * k), k,
                       (const int64_t *)nullptr, ip ? 1 : 0, D, I);
  }
  HIP_CHECK(hipGetLastError());
}

static void search_impl(dfann_index *h, int64_t nq, const float *q, int k,
                        float *D, int64_t *I, hipStream_t stream) {
  if (!h->trained) throw std::runtime_error("search on untrained index");
  if (k > 512) throw std::runtime_error("k > 512 unsupported");
  if (nq == 0) return;
  if (h->type == T_FLAT) { flat_search_impl(h, nq, q, k, D, I, stream); return; }
  if (h->ntotal == 0) { pad_fill(h, nq, k, D, I, stream); return; }
  if (h->type == T_HNSW) { hnsw_search(h, nq, q, k, D, I, stream); return; }
  finalize_csr(h, stream);
  int nprobe = std::min(h->nprobe, h->nlist);
  if (nprobe > 512) {
    // documented clamp (include/dfann.h Limits): warn once per process
    static bool warned = false;
    if (!warned) {
      warned = true;
      fprintf(stderr,
              "[dfann] warning: nprobe=%d exceeds the engine cap; clamped to "
              "512 (the faiss-backed reference would probe more lists)\n",
              nprobe);
    }
    nprobe = 512;
  }
  // probes + keys
  DevBuf &pb = h->ws2;
  pb.ensure((size_t)nq * nprobe * 8);
  int32_t *probes = pb.as<int32_t>();
  float *keys = reinterpret_cast<float *>(pb.as<uint8_t>() + (size_t)nq * nprobe * 4);
  coarse_impl(h, nq, q, nprobe, probes, keys, stream);
  scan_and_merge(h, nq, q, nprobe, probes, keys, k, D, I, stream);
}

// ---------------------------------------------------------------------------
// C-ABI
// ---------------------------------------------------------------------------

#define API_BEGIN try {
#define API_END                                                                \
  return 0;                                                                    \
  }                                                                            \
  catch (const std::exception &e) { g_err = e.what(); return 1; }

extern "C" int dfann_create(const char *spec_json, dfann_index **out) {
  API_BEGIN
  *out = create_from_spec(spec_json);
  API_END
}

extern "C" int dfann_destroy(dfann_index *h) {
  API_BEGIN
  delete h;
  API_END
}

extern "C" int dfann_train(dfann_index *h, int64_t n, const float *x_dev,
                           dfann_stream stream) {
  API_BEGIN
  train_impl(h, n, x_dev, (hipStream_t)stream);
  API_END
}

extern "C" int dfann_add(dfann_index *h, int64_t n, const float *x_dev,
                         dfann_stream stream) {
  API_BEGIN
  add_impl(h, n, x_dev, (hipStream_t)stream);
  API_END
}

extern "C" int dfann_search(dfann_index *h, int64_t nq, const float *q_dev,
                            int k, float *D_dev, int64_t *I_dev,
                            dfann_stream stream) {
  API_BEGIN
  search_impl(h, nq, q_dev, k, D_dev, I_dev, (hipStream_t)stream);
  API_END
}

extern "C" int dfann_coarse(dfann_index *h, int64_t nq, const float *q_dev,
                            int nprobe, int32_t *probes_dev, float *keys_dev,
                            dfann_stream stream) {
  API_BEGIN
  if (!h->trained || h->type == T_FLAT)
    throw std::runtime_error("coarse needs a trained IVF index");
  nprobe = std::min(nprobe, h->nlist);
  coarse_impl(h, nq, q_dev, nprobe, probes_dev, keys_dev, (hipStream_t)stream);
  API_END
}

extern "C" int dfann_search_preassigned(dfann_index *h, int64_t nq,
                                        const float *q_dev, int nprobe,
                                        const int32_t *probes_dev,
                                        const float *keys_dev, int k,
                                        float *D_dev, int64_t *I_dev,
                                        dfann_stream stream) {
  API_BEGIN
  if (!h->trained || h->type == T_FLAT)
    throw std::runtime_error("search_preassigned needs a trained IVF index");
  if (h->ntotal == 0) {
    pad_fill(h, nq, k, D_dev, I_dev, (hipStream_t)stream);
    return 0;
  }
  finalize_csr(h, (hipStream_t)stream);
  scan_and_merge(h, nq, q_dev, nprobe, probes_dev, keys_dev, k, D_dev, I_dev,
                 (hipStream_t)stream);
  API_END
}

extern "C" int dfann_search_reconstruct(dfann_index *h, int64_t nq,
                                        const float *q_dev, int k,
                                        float *D_dev, int64_t *I_dev,
                                        float *R_dev, dfann_stream stream) {
  API_BEGIN
  search_impl(h, nq, q_dev, k, D_dev, I_dev, (hipStream_t)stream);
  int rtype;
  const float *flat_src = nullptr;
  if (h->type == T_FLAT) { rtype = 0; flat_src = h->flat.as<float>(); }
  else if (h->type == T_IVFFLAT) rtype = 0;
  else if (h->type == T_IVFPQ) rtype = 2;
  else if (h->type == T_HNSW) rtype = 5;
  else rtype = h->sq8 ? 3 : 4;
  hipLaunchKernelGGL(k_reconstruct, dim3((unsigned)(nq * k)), dim3(64), 0,
                     (hipStream_t)stream, I_dev, nq, k, rtype, h->d, h->m,
                     h->dsub, h->stride, h->csr_arena.rlog, flat_src,
                     h->csr_arena.dev_table((hipStream_t)stream),
                     h->id2pos.as<unsigned>(), h->cr_off.as<int64_t>(),
                     h->nlist, h->centroids.as<float>(),
                     h->codebooks.as<float>(), h->sq_vmin.as<float>(),
                     h->sq_scale.as<float>(), R_dev);
  HIP_CHECK(hipGetLastError());
  API_END
}

extern "C" int dfann_set_nprobe(dfann_index *h, int nprobe) {
  API_BEGIN
  h->nprobe = nprobe;
  API_END
}

extern "C" int64_t dfann_ntotal(dfann_index *h) { return h->ntotal; }
extern "C" int dfann_nlist(dfann_index *h) { return h->nlist; }
extern "C" int dfann_is_trained(dfann_index *h) { return h->trained ? 1 : 0; }
extern "C" int dfann_dim(dfann_index *h) { return h->d; }
extern "C" const char *dfann_spec_json(dfann_index *h) {
  return h->spec_json.c_str();
}

extern "C" int dfann_get_centroids(dfann_index *h, float *out_host) {
  API_BEGIN
  if (h->type == T_FLAT)
    throw std::runtime_error("'flat' index has no quantizer");
  if (h->type == T_HNSW)
    throw std::runtime_error(
        "hnswsq index has no quantizer (reference AttributeError)");
  if (!h->trained) throw std::runtime_error("index not trained");
  HIP_CHECK(hipMemcpy(out_host, h->centroids.p, (size_t)h->nlist * h->d * 4,
                      hipMemcpyDeviceToHost));
  API_END
}

extern "C" int dfann_get_codebooks(dfann_index *h, float *out_host) {
  API_BEGIN
  if (h->type != T_IVFPQ) throw std::runtime_error("not an ivfpq index");
  HIP_CHECK(hipMemcpy(out_host, h->codebooks.p,
                      (size_t)h->m * 256 * h->dsub * 4, hipMemcpyDeviceToHost));
  API_END
}

extern "C" int dfann_get_lists(dfann_index *h, int64_t *off_host,
                               int64_t *ids_host, uint8_t *codes_host) {
  API_BEGIN
  if (h->type == T_FLAT) throw std::runtime_error("flat index has no lists");
  if (h->type == T_HNSW)
    throw std::runtime_error("hnswsq index has no inverted lists");
  finalize_csr(h, 0);
  memcpy(off_host, h->h_off.data(), (size_t)(h->nlist + 1) * 8);
  if (h->ntotal) {
    HIP_CHECK(hipMemcpy(ids_host, h->cr_ids.p, (size_t)h->ntotal * 8,
                        hipMemcpyDeviceToHost));
    h->csr_arena.copy_rows_to_host(codes_host, 0, h->ntotal);
  }
  API_END
}

extern "C" int dfann_code_stride(dfann_index *h) { return h->stride; }

extern "C" int dfann_get_sq_params(dfann_index *h, float *vmin_host,
                                   float *vdiff_host) {
  API_BEGIN
  if (!((h->type == T_IVFSQ || h->type == T_HNSW) && h->sq8))
    throw std::runtime_error("not an 8-bit ivfsq index");
  HIP_CHECK(hipMemcpy(vmin_host, h->sq_vmin.p, (size_t)h->d * 4,
                      hipMemcpyDeviceToHost));
  HIP_CHECK(hipMemcpy(vdiff_host, h->sq_vdiff.p, (size_t)h->d * 4,
                      hipMemcpyDeviceToHost));
  API_END
}

extern "C" int dfann_set_trained(dfann_index *h, const float *centroids_host,
                                 const float *codebooks_host,
                                 const float *vmin_host,
                                 const float *vdiff_host) {
  API_BEGIN
  if (h->type == T_FLAT) { h->trained = true; return 0; }
  if (!centroids_host) throw std::runtime_error("centroids required");
  h->centroids.ensure((size_t)h->nlist * h->d * 4);
  h->cnorm.ensure((size_t)h->nlist * 4);
  HIP_CHECK(hipMemcpy(h->centroids.p, centroids_host,
                      (size_t)h->nlist * h->d * 4, hipMemcpyHostToDevice));
  rownorms(h->centroids.as<float>(), h->nlist, h->d, h->cnorm.as<float>(), 0);
  if (h->type == T_IVFPQ) {
    if (!codebooks_host) throw std::runtime_error("codebooks required");
    h->codebooks.ensure((size_t)h->m * 256 * h->dsub * 4);
    HIP_CHECK(hipMemcpy(h->codebooks.p, codebooks_host,
                        (size_t)h->m * 256 * h->dsub * 4,
                        hipMemcpyHostToDevice));
  }
  if (h->type == T_IVFSQ && h->sq8) {
    if (!vmin_host || !vdiff_host) throw std::runtime_error("sq params required");
    h->sq_vmin.ensure((size_t)h->d * 4);
    h->sq_vdiff.ensure((size_t)h->d * 4);
    h->sq_scale.ensure((size_t)h->d * 4);
    HIP_CHECK(hipMemcpy(h->sq_vmin.p, vmin_host, (size_t)h->d * 4,
                        hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(h->sq_vdiff.p, vdiff_host, (size_t)h->d * 4,
                        hipMemcpyHostToDevice));
    std::vector<float> sc(h->d);
    std::vector<float> vd(h->d);
    memcpy(vd.data(), vdiff_host, (size_t)h->d * 4);
    for (int t = 0; t < h->d; ++t) sc[t] = vd[t] / 255.0f;
    HIP_CHECK(hipMemcpy(h->sq_scale.p, sc.data(), (size_t)h->d * 4,
                        hipMemcpyHostToDevice));
  }
  refresh_cent_bf16(h, 0);
  HIP_CHECK(hipDeviceSynchronize());
  h->trained = true;
  API_END
}

// ---------------------------------------------------------------------------
// shard merge
// ---------------------------------------------------------------------------

extern "C" int dfann_merge_topk(int64_t nq, int S, int k, const float *D_dev,
                                const int64_t *I_dev, int maximize,
                                float *Dout_dev, int64_t *Iout_dev,
                                dfann_stream stream) {
  API_BEGIN
  (void)I_dev;  // slots map to metadata host-side (ref client.py:297-298)
  if (k > 512) throw std::runtime_error("k > 512 unsupported");
  if (use_regsel(k)) {
    hipLaunchKernelGGL(k_merge_shards_rk, dim3((unsigned)nq), dim3(256),
                       REGSEL_LDS_BYTES + 128, (hipStream_t)stream, D_dev, nq,
                       S, k, maximize, Dout_dev, Iout_dev);
  } else {
    hipLaunchKernelGGL(k_merge_shards, dim3((unsigned)nq), dim3(256),
                       SEL_LDS_BYTES, (hipStream_t)stream, D_dev, nq, S, k,
                       maximize, Dout_dev, Iout_dev);
  }
  HIP_CHECK(hipGetLastError());
  API_END
}

// ---------------------------------------------------------------------------
// persistence (our own format, version 1)
// ---------------------------------------------------------------------------

static void fwrite_chk(const void *p, size_t n, FILE *f) {
  if (fwrite(p, 1, n, f) != n) throw std::runtime_error("short write");
}
static void fread_chk(void *p, size_t n, FILE *f) {
  if (fread(p, 1, n, f) != n) throw std::runtime_error("short read");
}

static void dump_dev(FILE *f, DevBuf &b, size_t bytes) {
  std::vector<char> tmp(bytes);
  if (bytes) HIP_CHECK(hipMemcpy(tmp.data(), b.p, bytes, hipMemcpyDeviceToHost));
  fwrite_chk(tmp.data(), bytes, f);
}
static void load_dev(FILE *f, DevBuf &b, size_t bytes) {
  std::vector<char> tmp(bytes);
  fread_chk(tmp.data(), bytes, f);
  b.ensure(std::max(bytes, (size_t)16));
  if (bytes) HIP_CHECK(hipMemcpy(b.p, tmp.data(), bytes, hipMemcpyHostToDevice));
}

extern "C" int dfann_save(dfann_index *h, const char *path) {
  API_BEGIN
  finalize_csr(h, 0);
  FILE *f = fopen(path, "wb");
  if (!f) throw std::runtime_error(std::string("cannot open ") + path);
  try {
    const uint64_t magic = 0x31304E4E414644ULL;  // "DFANN01"
    fwrite_chk(&magic, 8, f);
    uint64_t jlen = h->spec_json.size();
    fwrite_chk(&jlen, 8, f);
    fwrite_chk(h->spec_json.data(), jlen, f);
    int32_t hdr[8] = {h->type, h->metric, h->d, h->nlist, h->m,
                      h->nprobe, h->sq8 ? 1 : 0, h->stride};
    fwrite_chk(hdr, sizeof(hdr), f);
    uint8_t tr = h->trained ? 1 : 0;
    fwrite_chk(&tr, 1, f);
    fwrite_chk(&h->ntotal, 8, f);
    if (h->trained && h->type != T_FLAT) {
      if (h->type != T_HNSW)
        dump_dev(f, h->centroids, (size_t)h->nlist * h->d * 4);
      if (h->type == T_IVFPQ)
        dump_dev(f, h->codebooks, (size_t)h->m * 256 * h->dsub * 4);
      if ((h->type == T_IVFSQ || h->type == T_HNSW) && h->sq8) {
        dump_dev(f, h->sq_vmin, (size_t)h->d * 4);
        dump_dev(f, h->sq_vdiff, (size_t)h->d * 4);
      }
    }
    if (h->type == T_HNSW) {
      // codes (arrival order) + the graph
      const int64_t CHROWS =
          std::max<int64_t>(1, ((int64_t)64 << 20) / h->stride);
      std::vector<char> tmp((size_t)CHROWS * h->stride);
      for (int64_t r = 0; r < h->ntotal; r += CHROWS) {
        int64_t c = std::min(CHROWS, h->ntotal - r);
        h->csr_arena.copy_rows_to_host(tmp.data(), r, c);
        fwrite_chk(tmp.data(), (size_t)c * h->stride, f);
      }
      int32_t meta[4] = {h->hnsw_entry, h->hnsw_maxlevel, h->hnsw_nslots,
                         h->hnsw_efc};
      fwrite_chk(meta, sizeof(meta), f);
      fwrite_chk(h->h_levels.data(), (size_t)h->ntotal * 4, f);
      fwrite_chk(h->h_upslot.data(), (size_t)h->ntotal * 4, f);
      dump_dev(f, h->hn_cnt0, (size_t)h->ntotal * 4);
      dump_dev(f, h->hn_nbr0, (size_t)h->ntotal * 2 * h->m * 4);
      dump_dev(f, h->hn_cntU, (size_t)h->hnsw_nslots * HNSW_MAXL * 4);
      dump_dev(f, h->hn_nbrU, (size_t)h->hnsw_nslots * HNSW_MAXL * h->m * 4);
    } else if (h->type == T_FLAT) {
      dump_dev(f, h->flat, (size_t)h->ntotal * h->d * 4);
    } else if (h->ntotal) {
      fwrite_chk(h->h_off.data(), (size_t)(h->nlist + 1) * 8, f);
      dump_dev(f, h->cr_ids, (size_t)h->ntotal * 8);
      // codes: slab arena -> file in CSR row order (same byte layout as
      // the round-1 contiguous image)
      const int64_t CHROWS = std::max<int64_t>(1, ((int64_t)64 << 20) / h->stride);
      std::vector<char> tmp((size_t)CHROWS * h->stride);
      for (int64_t r = 0; r < h->ntotal; r += CHROWS) {
        int64_t c = std::min(CHROWS, h->ntotal - r);
        h->csr_arena.copy_rows_to_host(tmp.data(), r, c);
        fwrite_chk(tmp.data(), (size_t)c * h->stride, f);
      }
    }
  } catch (...) {
    fclose(f);
    throw;
  }
  fclose(f);
  API_END
}

extern "C" int dfann_load(const char *path, dfann_index **out) {
  API_BEGIN
  FILE *f = fopen(path, "rb");
  if (!f) throw std::runtime_error(std::string("cannot open ") + path);
  dfann_index *h = nullptr;
  try {
    uint64_t magic;
    fread_chk(&magic, 8, f);
    if (magic != 0x31304E4E414644ULL)
      throw std::runtime_error("bad dfann file magic");
    uint64_t jlen;
    fread_chk(&jlen, 8, f);
    std::string js(jlen, 0);
    fread_chk(&js[0], jlen, f);
    h = create_from_spec(js);
    int32_t hdr[8];
    fread_chk(hdr, sizeof(hdr), f);
    h->nprobe = hdr[5];
    uint8_t tr;
    fread_chk(&tr, 1, f);
    fread_chk(&h->ntotal, 8, f);
    h->trained = tr != 0;
    if (h->trained && h->type == T_HNSW) {
      load_dev(f, h->sq_vmin, (size_t)h->d * 4);
      load_dev(f, h->sq_vdiff, (size_t)h->d * 4);
      std::vector<float> vd(h->d), sc(h->d);
      HIP_CHECK(hipMemcpy(vd.data(), h->sq_vdiff.p, (size_t)h->d * 4,
                          hipMemcpyDeviceToHost));
      for (int t = 0; t < h->d; ++t) sc[t] = vd[t] / 255.0f;
      h->sq_scale.ensure((size_t)h->d * 4);
      HIP_CHECK(hipMemcpy(h->sq_scale.p, sc.data(), (size_t)h->d * 4,
                          hipMemcpyHostToDevice));
    } else if (h->trained && h->type != T_FLAT) {
      load_dev(f, h->centroids, (size_t)h->nlist * h->d * 4);
      h->cnorm.ensure((size_t)h->nlist * 4);
      rownorms(h->centroids.as<float>(), h->nlist, h->d, h->cnorm.as<float>(), 0);
      if (h->type == T_IVFPQ)
        load_dev(f, h->codebooks, (size_t)h->m * 256 * h->dsub * 4);
      refresh_cent_bf16(h, 0);
      if (h->type == T_IVFSQ && h->sq8) {
        load_dev(f, h->sq_vmin, (size_t)h->d * 4);
        load_dev(f, h->sq_vdiff, (size_t)h->d * 4);
        std::vector<float> vd(h->d), sc(h->d);
        HIP_CHECK(hipMemcpy(vd.data(), h->sq_vdiff.p, (size_t)h->d * 4,
                            hipMemcpyDeviceToHost));
        for (int t = 0; t < h->d; ++t) sc[t] = vd[t] / 255.0f;
        h->sq_scale.ensure((size_t)h->d * 4);
        HIP_CHECK(hipMemcpy(h->sq_scale.p, sc.data(), (size_t)h->d * 4,
                            hipMemcpyHostToDevice));
      }
    }
    if (h->type == T_HNSW) {
      if (h->ntotal) {
        const int64_t CHROWS =
            std::max<int64_t>(1, ((int64_t)64 << 20) / h->stride);
        std::vector<char> tmp((size_t)CHROWS * h->stride);
        for (int64_t r = 0; r < h->ntotal; r += CHROWS) {
          int64_t c = std::min(CHROWS, h->ntotal - r);
          fread_chk(tmp.data(), (size_t)c * h->stride, f);
          h->csr_arena.copy_rows_from_host(tmp.data(), r, c);
        }
        h->csr_arena.rows = h->ntotal;
        int32_t meta[4];
        fread_chk(meta, sizeof(meta), f);
        h->hnsw_entry = meta[0];
        h->hnsw_maxlevel = meta[1];
        h->hnsw_nslots = meta[2];
        h->hnsw_efc = meta[3];
        h->h_levels.resize(h->ntotal);
        h->h_upslot.resize(h->ntotal);
        fread_chk(h->h_levels.data(), (size_t)h->ntotal * 4, f);
        fread_chk(h->h_upslot.data(), (size_t)h->ntotal * 4, f);
        h->hn_levels.ensure((size_t)h->ntotal * 4);
        HIP_CHECK(hipMemcpy(h->hn_levels.p, h->h_levels.data(),
                            (size_t)h->ntotal * 4, hipMemcpyHostToDevice));
        h->hn_upslot.ensure((size_t)h->ntotal * 4);
        HIP_CHECK(hipMemcpy(h->hn_upslot.p, h->h_upslot.data(),
                            (size_t)h->ntotal * 4, hipMemcpyHostToDevice));
        load_dev(f, h->hn_cnt0, (size_t)h->ntotal * 4);
        load_dev(f, h->hn_nbr0, (size_t)h->ntotal * 2 * h->m * 4);
        load_dev(f, h->hn_cntU, (size_t)h->hnsw_nslots * HNSW_MAXL * 4);
        load_dev(f, h->hn_nbrU,
                 (size_t)h->hnsw_nslots * HNSW_MAXL * h->m * 4);
        HIP_CHECK(hipDeviceSynchronize());
      }
    } else if (h->type == T_FLAT) {
      load_dev(f, h->flat, (size_t)h->ntotal * h->d * 4);
    } else if (h->ntotal) {
      h->h_off.resize(h->nlist + 1);
      fread_chk(h->h_off.data(), (size_t)(h->nlist + 1) * 8, f);
      load_dev(f, h->cr_ids, (size_t)h->ntotal * 8);
      {
        // codes: file (CSR row order) -> slab arena
        const int64_t CHROWS =
            std::max<int64_t>(1, ((int64_t)64 << 20) / h->stride);
        std::vector<char> tmp((size_t)CHROWS * h->stride);
        for (int64_t r = 0; r < h->ntotal; r += CHROWS) {
          int64_t c = std::min(CHROWS, h->ntotal - r);
          fread_chk(tmp.data(), (size_t)c * h->stride, f);
          h->csr_arena.copy_rows_from_host(tmp.data(), r, c);
        }
        h->csr_arena.rows = h->ntotal;
        h->csr_rows = h->ntotal;
      }
      h->cr_off.ensure((size_t)(h->nlist + 1) * 8);
      HIP_CHECK(hipMemcpy(h->cr_off.p, h->h_off.data(),
                          (size_t)(h->nlist + 1) * 8, hipMemcpyHostToDevice));
      h->id2pos.ensure((size_t)h->ntotal * 4);
      std::vector<int64_t> ids(h->ntotal);
      HIP_CHECK(hipMemcpy(ids.data(), h->cr_ids.p, (size_t)h->ntotal * 8,
                          hipMemcpyDeviceToHost));
      std::vector<unsigned> i2p(h->ntotal);
      for (int l = 0; l < h->nlist; ++l)
        for (int64_t j = h->h_off[l]; j < h->h_off[l + 1]; ++j)
          i2p[ids[j]] = (unsigned)j;
      HIP_CHECK(hipMemcpy(h->id2pos.p, i2p.data(), (size_t)h->ntotal * 4,
                          hipMemcpyHostToDevice));
      HIP_CHECK(hipDeviceSynchronize());
    }
  } catch (...) {
    fclose(f);
    delete h;
    throw;
  }
  fclose(f);
  *out = h;
  API_END
}

// ---------------------------------------------------------------------------
// HNSW introspection (test plumbing: graph determinism + oracle-shared
// search parity — DESIGN.md §hnsw)
// ---------------------------------------------------------------------------

extern "C" int dfann_hnsw_info(dfann_index *h, int64_t out[6]) {
  API_BEGIN
  if (h->type != T_HNSW) throw std::runtime_error("not an hnswsq index");
  out[0] = h->m;
  out[1] = 2 * h->m;
  out[2] = h->hnsw_nslots;
  out[3] = h->hnsw_entry;
  out[4] = h->hnsw_maxlevel;
  out[5] = h->hnsw_efc;
  API_END
}

extern "C" int dfann_hnsw_dump(dfann_index *h, int32_t *levels_host,
                               int32_t *cnt0_host, int32_t *nbr0_host,
                               int32_t *upslot_host, int32_t *cntU_host,
                               int32_t *nbrU_host) {
  API_BEGIN
  if (h->type != T_HNSW) throw std::runtime_error("not an hnswsq index");
  int64_t n = h->ntotal;
  memcpy(levels_host, h->h_levels.data(), (size_t)n * 4);
  memcpy(upslot_host, h->h_upslot.data(), (size_t)n * 4);
  HIP_CHECK(hipMemcpy(cnt0_host, h->hn_cnt0.p, (size_t)n * 4,
                      hipMemcpyDeviceToHost));
  HIP_CHECK(hipMemcpy(nbr0_host, h->hn_nbr0.p, (size_t)n * 2 * h->m * 4,
                      hipMemcpyDeviceToHost));
  if (h->hnsw_nslots) {
    HIP_CHECK(hipMemcpy(cntU_host, h->hn_cntU.p,
                        (size_t)h->hnsw_nslots * HNSW_MAXL * 4,
                        hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(nbrU_host, h->hn_nbrU.p,
                        (size_t)h->hnsw_nslots * HNSW_MAXL * h->m * 4,
                        hipMemcpyDeviceToHost));
  }
  API_END
}

// ---------------------------------------------------------------------------
// timing
// ---------------------------------------------------------------------------

extern "C" int dfann_set_timing(dfann_index *h, int enabled) {
  API_BEGIN
  h->timing = enabled != 0;
  API_END
}

static double sum_events(std::vector<TimingEv> &v) {
  double ms = 0;
  for (auto &e : v) {
    HIP_CHECK(hipEventSynchronize(e.b));
    float el = 0;
    HIP_CHECK(hipEventElapsedTime(&el, e.a, e.b));
    ms += el;
    (void)hipEventDestroy(e.a);
    (void)hipEventDestroy(e.b);
  }
  v.clear();
  return ms;
}

extern "C" int dfann_get_timing(dfann_index *h, dfann_timing *out) {
  API_BEGIN
  out->scan_launches = (int64_t)h->ev_scan.size();
  out->merge_launches = (int64_t)h->ev_merge.size();
  out->lut_launches = (int64_t)h->ev_lut.size();
  out->scan_ms = sum_events(h->ev_scan);
  out->gemm_ms = sum_events(h->ev_gemm);
  out->merge_ms = sum_events(h->ev_merge);
  out->lut_ms = sum_events(h->ev_lut);
  out->scan_rows = h->scan_rows;
  out->scan_bytes = h->scan_bytes;
  out->gemm_flops = h->gemm_flops;
  h->scan_rows = h->scan_bytes = h->gemm_flops = 0;
  API_END
}
