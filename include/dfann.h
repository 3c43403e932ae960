/* dfann — MI355X-native ANN engine, C-ABI boundary.
 *
 * This library replaces the faiss Python-module surface consumed by the
 * reference's distributed_faiss/index.py and client.py.
 * Each entry point cites the reference interface it replaces. Host
 * bindings: ctypes (distributed_faiss_amd/hip_engine.py); see
 * INTEGRATION.md for the binding a maintainer would add.
 *
 * Conventions:
 *  - all functions return 0 on success, nonzero on error;
 *    dfann_last_error() gives the message (thread-local).
 *  - pointers suffixed _dev are DEVICE (HIP) pointers — tensors already
 *    resident in HBM; pointers suffixed _host are host memory.
 *  - `stream` is a hipStream_t (pass the caller's current stream; 0 ok).
 *  - distances follow faiss conventions: L2 = squared distance
 *    (minimize), IP = dot product (maximize); unfilled result slots are
 *    I = -1 with D = +FLT_MAX (L2) / -FLT_MAX (IP).
 *  - ids are implicit arrival positions per index (reference quirk,
 *    SURVEY.md §2 item 9).
 *
 * Limits (engine caps; the faiss-backed reference has none of these):
 *  - k (top-k) <= 512 everywhere (dfann_search*, dfann_merge_topk):
 *    selection buffers are sized SEL_CAP=1024 with k<=512 headroom.
 *    Larger k returns an error ("k > 512 unsupported").
 *  - nprobe is clamped to min(nlist, 512). A request above 512 is
 *    honored as 512 and a one-time warning is printed to stderr (the
 *    reference would probe more lists; recall at nprobe=512 is the cap).
 *  - per-shard ntotal < 2^32: ids round-trip through u32 inside the
 *    scan/merge kernels (CSR positions and merge slots are u32), so one
 *    shard holds at most 4.29e9 vectors. The C-ABI keeps i64 ids; the
 *    1B-vector headline config is 8 shards x 125M, well inside the cap.
 */
#ifndef DFANN_H
#define DFANN_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct dfann_index dfann_index; /* opaque */
typedef void *dfann_stream;             /* hipStream_t */

/* --- lifecycle -------------------------------------------------------- */

/* Build an index from a JSON spec {"type": "flat"|"ivf_flat"|"ivfpq"|
 * "ivfsq", "dim": int, "metric": 0|1, "nlist": int, "m": int, "nbits": 8,
 * "sq_type": "fp16"|"8bit", "nprobe": int, "seed": int}.
 * Replaces: faiss.IndexFlatIP/IndexFlatL2 (ref index.py:28-30,94),
 * faiss.IndexIVFFlat (ref index.py:38), faiss.IndexIVFPQ (ref
 * index.py:46), faiss.IndexIVFScalarQuantizer + QT_fp16/QT_8bit (ref
 * index.py:55,64-66), faiss.index_factory (ref index.py:396). */
int dfann_create(const char *spec_json, dfann_index **out);
int dfann_destroy(dfann_index *h);

/* --- build path ------------------------------------------------------- */

/* Replaces faiss `Index.train` at ref index.py:217 (k-means coarse
 * quantizer; + per-subspace PQ codebooks / SQ ranges on residuals). */
int dfann_train(dfann_index *h, int64_t n, const float *x_dev,
                dfann_stream stream);

/* Replaces faiss `Index.add` at ref index.py:425 (coarse assign + encode
 * + inverted-list append; ids = arrival order). */
int dfann_add(dfann_index *h, int64_t n, const float *x_dev,
              dfann_stream stream);

/* --- search path ------------------------------------------------------ */

/* Replaces faiss `Index.search` at ref index.py:257. D_dev: (nq,k) f32,
 * I_dev: (nq,k) i64. k <= 512 (see Limits above). */
int dfann_search(dfann_index *h, int64_t nq, const float *q_dev, int k,
                 float *D_dev, int64_t *I_dev, dfann_stream stream);

/* Replaces faiss `Index.search_and_reconstruct` at ref index.py:255.
 * R_dev: (nq,k,d) f32 (decoded vectors; zeros at padded slots). */
int dfann_search_reconstruct(dfann_index *h, int64_t nq, const float *q_dev,
                             int k, float *D_dev, int64_t *I_dev,
                             float *R_dev, dfann_stream stream);

/* Coarse quantization only: top-nprobe list ids (i32, (nq,nprobe)) and
 * their minimize-keys (f32; L2: |c|^2-2q.c, IP: -q.c). Mirrors the
 * quantizer search half of faiss IndexIVF::search. */
int dfann_coarse(dfann_index *h, int64_t nq, const float *q_dev, int nprobe,
                 int32_t *probes_dev, float *keys_dev, dfann_stream stream);

/* Search with externally supplied probe lists (+keys for the IP bias).
 * Mirrors faiss IndexIVF::search_preassigned; also the parity-test hook
 * for probe-set-independent bit-exactness. */
int dfann_search_preassigned(dfann_index *h, int64_t nq, const float *q_dev,
                             int nprobe, const int32_t *probes_dev,
                             const float *keys_dev, int k, float *D_dev,
                             int64_t *I_dev, dfann_stream stream);

/* --- knobs / introspection -------------------------------------------- */

/* ref index.py:352-356. Effective nprobe is min(nprobe, nlist, 512) at
 * search time (see Limits above); values above 512 warn once. */
int dfann_set_nprobe(dfann_index *h, int nprobe);
int64_t dfann_ntotal(dfann_index *h);             /* faiss .ntotal */
int dfann_nlist(dfann_index *h);                  /* faiss .nlist */
int dfann_is_trained(dfann_index *h);
int dfann_dim(dfann_index *h);
/* spec json the index was created with (valid until destroy) */
const char *dfann_spec_json(dfann_index *h);

/* Coarse centroids to host, (nlist,d) f32. Replaces
 * quantizer.reconstruct_n(0, nlist) at ref index.py:350. Errors on flat
 * (no quantizer — the reference would AttributeError). */
int dfann_get_centroids(dfann_index *h, float *out_host);

/* --- persistence (replaces faiss read_index/write_index at ref
 *     index.py:297,460; our own file format) --- */
int dfann_save(dfann_index *h, const char *path);
int dfann_load(const char *path, dfann_index **out);

/* --- trained-artifact exchange (parity-test plumbing: train once, load
 *     the same artifacts into oracle and engine — SURVEY.md §8c) ------- */
int dfann_set_trained(dfann_index *h, const float *centroids_host,
                      const float *codebooks_host, const float *vmin_host,
                      const float *vdiff_host);
int dfann_get_codebooks(dfann_index *h, float *out_host); /* (m,256,dsub) */
/* Dump the finalized inverted lists to host: off (nlist+1 i64), ids
 * (ntotal i64, arrival ids in CSR order), codes (ntotal * stride u8).
 * stride = code_bytes rounded up to 16. Used by the CPU-baseline leg of
 * bench.py to scan the SAME index content the GPU holds. */
int dfann_get_lists(dfann_index *h, int64_t *off_host, int64_t *ids_host,
                    uint8_t *codes_host);
int dfann_code_stride(dfann_index *h);
int dfann_get_sq_params(dfann_index *h, float *vmin_host,
                        float *vdiff_host); /* (d,), (d,) */

/* --- shard-merge (replaces ResultHeap + _aggregate_results' arithmetic,
 *     ref client.py:29-54,265-310; fed by the RCCL all-gather of
 *     per-shard top-k in the multi-GPU path) -------------------------- */
/* D_dev: (S,nq,k) f32 shard distances (faiss sign conventions);
 * I_dev: (S,nq,k) i64 shard ids (-1 pads). maximize: 1 for dot.
 * Output ids are GLOBAL slots s*nq*k + q*k + j into the gathered input
 * (the caller maps slots to (shard, local id) / metadata, mirroring ref
 * client.py:290,297-298). Dout is negated for maximize (ref quirk 2).
 * k <= 512 (see Limits above). */
int dfann_merge_topk(int64_t nq, int S, int k, const float *D_dev,
                     const int64_t *I_dev, int maximize, float *Dout_dev,
                     int64_t *Iout_dev, dfann_stream stream);

/* --- hnswsq (replaces faiss.IndexHNSWSQ at ref index.py:51-60) --------
 * spec: {"type": "hnswsq", "dim", "metric": 1 (L2 only — the reference
 * asserts), "m": store_n (link cap M; level-0 cap 2M), "ef_construction",
 * "nprobe" (= efSearch; honored dynamically — deviation: the
 * reference's set_nprobe is a silent no-op on HNSW), "seed"}.
 * Build is a BATCHED wave insertion over frozen snapshots with reverse
 * links applied in sorted order — deterministic given (data, seed, wave
 * schedule), but a DIFFERENT graph than faiss's sequential insertion
 * (quality gated by recall tests). Search parity is
 * pinned by the oracle restatement over the dumped graph. ------------- */

/* info: out = {M, deg0, nslots, entry, maxlevel, ef_construction} */
int dfann_hnsw_info(dfann_index *h, int64_t out[6]);
/* dump the graph to host: levels (n i32), cnt0 (n), nbr0 (n x 2M),
 * upslot (n), cntU (nslots x 8), nbrU (nslots x 8 x M) */
int dfann_hnsw_dump(dfann_index *h, int32_t *levels_host, int32_t *cnt0_host,
                    int32_t *nbr0_host, int32_t *upslot_host,
                    int32_t *cntU_host, int32_t *nbrU_host);

/* --- kernel timing for the roofline harness (bench.py) ---------------- */

typedef struct dfann_timing {
  double scan_ms;      /* ivf list-scan kernel ALONE, summed over launches */
  int64_t scan_launches;
  int64_t scan_rows;   /* codes scanned (algorithmic units) */
  int64_t scan_bytes;  /* scan_rows * packed code stride */
  double gemm_ms;      /* coarse/flat distance GEMM */
  int64_t gemm_flops;  /* 2*M*N*K summed */
  double merge_ms;
  int64_t merge_launches;
  double lut_ms;       /* ADC-table build (k_pq_lut / term3), GLUT/PRE paths */
  int64_t lut_launches;
} dfann_timing;

int dfann_set_timing(dfann_index *h, int enabled);
/* Synchronizes the recorded events, fills out, and resets accumulators. */
int dfann_get_timing(dfann_index *h, dfann_timing *out);

const char *dfann_last_error(void);

#ifdef __cplusplus
}
#endif
#endif /* DFANN_H */
