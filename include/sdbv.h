/* sdbv.h — C-ABI of the MI355X-native SurrealDB vector-KNN hot path.
 *
 * This is the drop-in boundary: the Rust host (surrealdb) would bind these
 * entry points via hip-sys/bindgen (see INTEGRATION.md for the binding stub).
 * Each function cites the reference interface it replaces:
 *
 *  - sdbv_knn_bruteforce / sdbv_knn_batch replace the distance+top-K core of
 *    the brute-force KnnTopK operator
 *    (reference: surrealdb/core/src/exec/operators/knn_topk.rs:166-267 and the
 *    legacy KnnPriorityList path surrealdb/core/src/idx/planner/knn.rs:11-106).
 *    The host keeps: record streaming, Value->vector extraction
 *    (knn_topk.rs:274-288), WHERE filtering, and record materialisation.
 *  - sdbv_stage_corpus replaces the per-search scattered vector access of
 *    the reference (VectorCache LRU + He KV keys,
 *    surrealdb/core/src/idx/trees/hnsw/cache.rs,
 *    surrealdb/core/src/idx/trees/hnsw/elements.rs:94-140) with a one-time
 *    HBM-resident staging of a table's vectors.
 *  - sdbv_hnsw_* replace the layer-0 best-first search loop
 *    (surrealdb/core/src/idx/trees/hnsw/layer.rs:184-223) behind
 *    HnswIndex::knn_search (surrealdb/core/src/idx/trees/hnsw/index.rs:270-335),
 *    plus the graph build (sequential = parity; parallel/snapshot = bench
 *    mode) and element removal with neighbour repair.
 *  - sdbv_index_* replace the whole HnswIndex operator surface: the Hp
 *    pendings queue (index.rs:138-257), VecDocs/Ids64 doc expansion
 *    (docs.rs, knn.rs:163-326), doc-id allocation with recycling, the
 *    pendings-merged knn_search (index.rs:270-335) and the cond_filter
 *    flow (hnsw/filter.rs) behind a host truthiness callback. The host
 *    keeps only what touches the KV transaction: value->vector
 *    extraction, the RecordIdKey<->handle map, WHERE evaluation, record
 *    materialisation.
 *  - sdbv_kvload_* / sdbv_*_dump_kv speak the reference's on-disk formats
 *    (key/index/{he,hn,hs,hv,hp}.rs) for cold-start bulk loads and dumps.
 *
 * Plain pointers and sizes only; no torch types. All calls are synchronous
 * w.r.t. results and thread-safe (internal per-context mutex). The caller
 * owns all host buffers; the context owns all device memory.
 */
#ifndef SDBV_H
#define SDBV_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Error codes (negative) — 0 == OK. */
enum {
	SDBV_OK = 0,
	SDBV_ERR_HIP = -1,          /* HIP runtime error; see sdbv_last_error */
	SDBV_ERR_NO_TABLE = -2,     /* table id not staged */
	SDBV_ERR_BAD_ARG = -3,      /* dimension mismatch, k out of range, ... */
	SDBV_ERR_OOM = -4,          /* device allocation failed */
	SDBV_ERR_UNSUPPORTED = -5,  /* metric not GPU-supported (host fallback is the
	                               caller's oracle-free CPU path, not ours) */
	SDBV_ERR_IDS_UNSORTED = -6, /* ids must be strictly increasing (tie-break contract) */
};

/* Distance metric — mirrors catalog::Distance
 * (surrealdb/core/src/catalog/schema/index.rs:250-284), all 8 variants on
 * the GPU scan. COSINE/EUCLIDEAN run the LDS-tiled k_scan (the headline
 * path); the others run an all-distances kernel + exact top-k selection
 * (each restating the reference's F32 chain, vector.rs:206-451 —
 * including jaccard's bit-pattern sets :317-340 and the F32 |I|/|U|
 * asymmetry, and pearson's similarity-as-distance :413-440). MINKOWSKI
 * takes its order from sdbv_table_set_order (default 0 — degenerate, like
 * the reference with order 0). */
enum {
	SDBV_METRIC_COSINE = 0,
	SDBV_METRIC_EUCLIDEAN = 1,
	SDBV_METRIC_MANHATTAN = 2,
	SDBV_METRIC_CHEBYSHEV = 3,
	SDBV_METRIC_HAMMING = 4,
	SDBV_METRIC_JACCARD = 5,
	SDBV_METRIC_MINKOWSKI = 6,
	SDBV_METRIC_PEARSON = 7,
};

typedef struct sdbv_ctx sdbv_ctx; /* owns HIP stream, device pools, staged tables */

typedef struct sdbv_stats {
	double last_scan_kernel_ms;  /* HIP-event time of the last distance-scan kernel */
	double last_merge_kernel_ms; /* HIP-event time of the last top-K merge kernel */
	double last_total_ms;        /* host wall of the last search call (incl. D2H of K results) */
	uint64_t bytes_staged;       /* device bytes held by staged tables */
	uint64_t last_rows_scanned;  /* rows the last scan covered */
} sdbv_stats;

/* Create a context on HIP device `device` (-1 = current device). */
int sdbv_init(int device, sdbv_ctx **out);
void sdbv_shutdown(sdbv_ctx *);

const char *sdbv_last_error(sdbv_ctx *);
int sdbv_get_stats(sdbv_ctx *, sdbv_stats *out);

/* Stage a table's vectors into HBM as a feature-major f32 store
 * (column-major [d][n_pad]) plus per-row f64 norms precomputed with the
 * reference's norm semantics (vector.rs:244-249).
 * `rows` is row-major n x d host memory. `ids` maps row -> record/doc id and
 * must be strictly increasing (NULL => 0..n-1); result tie-break is
 * (distance asc, id asc), which equals the reference's orderings for
 * monotone ids (knn.rs:363 BTreeSet<(FloatKey, VectorId)>, knn_topk.rs:61-73
 * insertion-order seq). Restages (replaces) if `table` already staged. */
int sdbv_stage_corpus(sdbv_ctx *, uint64_t table, const float *rows,
                      const uint64_t *ids, uint64_t n, uint32_t d,
                      uint8_t metric);

/* Stage a deterministic synthetic corpus directly on-device (no PCIe):
 * element (row_offset+i, j) = splitmix64-derived U[-20,20) f32 — the committed
 * restatement of the BASELINE synthetic-data contract (matches
 * RandomItemGenerator::Float(-20,20), knn.rs:633-643, by distribution;
 * exact bit-stream defined by oracle/orc_gen_f32). ids = id_base + i. */
int sdbv_stage_synthetic(sdbv_ctx *, uint64_t table, uint64_t n, uint32_t d,
                         uint8_t metric, uint64_t seed, uint64_t row_offset,
                         uint64_t id_base);

uint64_t sdbv_table_rows(sdbv_ctx *, uint64_t table);
int sdbv_drop_table(sdbv_ctx *, uint64_t table);

/* Minkowski order for a staged table (Distance::Minkowski(Number),
 * index.rs:258): call after staging; ignored by other metrics. */
int sdbv_table_set_order(sdbv_ctx *, uint64_t table, double order);

/* Host-side generator of the committed synthetic-data contract (bench/test
 * input prep; bit-identical to the device staging generator). */
void sdbv_gen_f32(uint64_t seed, uint64_t row0, uint64_t nrows, uint32_t d,
                  float *out);

/* Brute-force exact K-nearest-neighbour scan of a staged table.
 * Results sorted ascending by (distance f64-total_cmp, id).
 * Distances follow Distance::calculate F32 semantics (vector.rs:244-249 /
 * 282-283): f32 accumulation per the restated ndarray contract, f64 finish.
 * k <= 64 runs fully on-device (block-local exact top-K); larger k
 * (the reference accepts any) takes one all-distances launch + exact host
 * selection — same ordering contract, costing an n-f64 transfer.
 * *out_n = min(k, n). */
int sdbv_knn_bruteforce(sdbv_ctx *, uint64_t table, const float *q, uint32_t d,
                        uint32_t k, uint64_t *out_ids, double *out_dists,
                        uint32_t *out_n);

/* Batched-query brute-force: Q is b x d row-major; out_ids/out_dists are
 * b x k row-major. The dense Q x corpus^T product runs on MFMA. */
int sdbv_knn_batch(sdbv_ctx *, uint64_t table, const float *Q, uint32_t b,
                   uint32_t d, uint32_t k, uint64_t *out_ids,
                   double *out_dists);

/* ---- HNSW (graph search; graph topology stays host-side) ---- */

/* Batched gather+distance for HNSW layer-0 neighbour expansion
 * (replaces the per-visit get_vector + Distance::calculate of
 * layer.rs:184-223): for frontier rows[i] (row indices into the staged
 * table), out_dists[i] = distance(q, row). */
int sdbv_gather_distance(sdbv_ctx *, uint64_t table, const uint32_t *rows,
                         uint32_t nrows, const float *q, uint32_t d,
                         double *out_dists);

/* HNSW index — the Hnsw/HnswIndex replacement (hnsw/mod.rs, hnsw/index.rs,
 * layer.rs, heuristic.rs). Graph topology and upper layers live host-side
 * (as in the reference engine); layer-0 neighbour expansion runs as the
 * batched gather+distance kernel over the staged table; doc-id expansion,
 * pendings merge and cond filtering stay with the caller.
 *
 * Element ids are insertion ordinals (the caller maps them to records, as
 * HnswDocs does, docs.rs:39-145).
 *
 * Defaults mirror DEFINE INDEX ... HNSW (syn/parser/stmt/define.rs:1102-1180):
 * m=12, m0=2m, efc=150, ml=1/ln(m), extend=keep=0, metric EUCLIDEAN, F32. */
typedef struct sdbv_hnsw sdbv_hnsw;
int sdbv_hnsw_create(sdbv_ctx *, uint32_t d, uint8_t metric, uint32_t m,
                     uint32_t m0, uint32_t efc, int extend_candidates,
                     int keep_pruned, uint64_t rng_seed, double ml,
                     sdbv_hnsw **out);
/* Insert one vector (Hnsw::insert, hnsw/mod.rs:389-394; level drawn from the
 * committed RNG restatement of :263-266). Sequential build — deterministic,
 * used for parity. */
int sdbv_hnsw_insert(sdbv_hnsw *, const float *pt);
/* Parallel bulk build (bench mode): levels are drawn deterministically per
 * ordinal, insert order is nondeterministic across threads (striped node
 * locks) — graph quality validated by the reference's recall bars, like the
 * reference's own lock-free enqueue + batched apply (index.rs:138-211). */
int sdbv_hnsw_insert_batch(sdbv_hnsw *, const float *pts, uint64_t n,
                           int nthreads);
/* Chunked SNAPSHOT bulk build (bench mode, one step beyond the parallel
 * build's relaxed ordering): per chunk, every level-0 element's efc-search
 * runs against the graph as of the chunk start (read-only, parallel, and —
 * round 2 — batched onto the persistent device kernel), then the apply
 * half runs under the striped node locks. Quality pinned by the same
 * recall bars as the parallel build. */
int sdbv_hnsw_insert_batch_snapshot(sdbv_hnsw *, const float *pts,
                                    uint64_t n, uint32_t chunk,
                                    int nthreads);
/* Snapshot build with the BATCHED apply schedule at layers 1 and 0 (one
 * fixed serialization of the parallel apply per layer: all selects, then
 * all appends in element order, then one prune pass). Levels >= 2 — the
 * tiny skeleton layers — insert classically (progressive), and a
 * 64-element classic warm-up bootstraps an empty graph (both measured
 * requirements). Same algorithm, quality pinned by the same
 * recall bars; the bit-exact host reference for the _gpu build below. */
int sdbv_hnsw_insert_batch_snapshot2(sdbv_hnsw *, const float *pts,
                                     uint64_t n, uint32_t chunk,
                                     int nthreads);
/* GPU-accelerated chunked snapshot build (hnsw/mod.rs:230-394 build hot
 * loop): bit-identical to
 * sdbv_hnsw_insert_batch_snapshot2 — each chunk's layer-1 and layer-0
 * efc-searches run as multi-ep persistent-kernel launches against
 * delta-updated padded device adjacencies (ef=1 launches carry the
 * level-0 elements' layer-1 descent hop), and the select/prune
 * pair-distance work (the RAM-bound part of the apply) runs on device
 * with the exact heuristic (k_pair_mats/k_heur_select). Requires a
 * device context; efc <= 512; extend-candidates falls back to the host
 * twin. */
int sdbv_hnsw_insert_batch_snapshot_gpu(sdbv_hnsw *, const float *pts,
                                        uint64_t n, uint32_t chunk,
                                        int nthreads);
/* Stage vectors (feature-major + norms) into the device table slot `table`;
 * required before sdbv_hnsw_knn. */
int sdbv_hnsw_finalize(sdbv_hnsw *, uint64_t table);
/* knn_search (hnsw/index.rs:270-335 minus host-kept parts): upper-layer
 * greedy descent host-side, layer-0 ef-search with GPU batched neighbour
 * expansion. Results ascending (dist total_cmp, id), truncated to k. */
int sdbv_hnsw_knn(sdbv_hnsw *, const float *q, uint32_t k, uint32_t ef,
                  uint64_t *out_ids, double *out_dists, uint32_t *out_n);
/* Host-side knn_search (no device): the build path's search algorithm over
 * the host graph — same exact result contract. For CPU-side tests, quality
 * audits and boxes without a staged device table. */
int sdbv_hnsw_knn_host(sdbv_hnsw *, const float *q, uint32_t k, uint32_t ef,
                       uint64_t *out_ids, double *out_dists, uint32_t *out_n);
/* Batched ef-search on the persistent kernel (one query per workgroup, the
 * whole layer-0 best-first loop in-kernel — same exact result contract as
 * sdbv_hnsw_knn). Q is b x d; outputs are b x k (+ out_ns per query).
 * ef <= 512. */
int sdbv_hnsw_knn_batch(sdbv_hnsw *, const float *Q, uint32_t b, uint32_t k,
                        uint32_t ef, uint64_t *out_ids, double *out_dists,
                        uint32_t *out_ns);
/* Full per-layer graph export (CSR + membership + enter point + zero-copy
 * vector view): lets a host-side searcher (the bench's oracle cpu_baseline
 * leg) run on the exact graph the device searches. */
uint64_t sdbv_hnsw_layer_edge_count(sdbv_hnsw *, uint32_t layer);
void sdbv_hnsw_layer_export(sdbv_hnsw *, uint32_t layer, uint32_t *offsets,
                            uint32_t *edges, uint8_t *in_layer);
int64_t sdbv_hnsw_enter_point(sdbv_hnsw *);
const float *sdbv_hnsw_vecs_ptr(sdbv_hnsw *);
void sdbv_hnsw_destroy(sdbv_hnsw *);
/* Remove one graph element with neighbour repair (Hnsw::remove,
 * hnsw/mod.rs:398-455 + layer.rs:408-460). Pre-finalize host graphs only;
 * a finalized index mutates through sdbv_index_* (which re-finalizes).
 * Returns 1 if removed, 0 if the element did not exist. */
int sdbv_hnsw_remove(sdbv_hnsw *, uint64_t e_id);
/* Introspection (parity tests): */
uint64_t sdbv_hnsw_n(sdbv_hnsw *);
uint32_t sdbv_hnsw_layers(sdbv_hnsw *);
uint64_t sdbv_hnsw_l0_edge_count(sdbv_hnsw *);
void sdbv_hnsw_l0_export(sdbv_hnsw *, uint32_t *offsets /* n+1 */,
                         uint32_t *edges);

/* ------------------------------------------------------------------------
 * Index layer — the HnswIndex operator surface (hnsw/index.rs) with the
 * KV-backed parts (Hp pendings queue, Hv vector->docs entries, hi/hd
 * record-key<->doc-id maps) held in host memory. The host binding passes
 * record keys as opaque u64 handles ordered like its RecordIdKey ordering
 * (INTEGRATION.md "record-key handles"); value->vector extraction
 * (content_to_vectors, index.rs:118-129) stays on the host side of this
 * boundary. ctx may be NULL for a host-only (CPU-testable) index; with a
 * ctx, searches run the GPU per-hop path, re-finalizing into device table
 * slot `table` after writes. */
typedef struct sdbv_index sdbv_index;
int sdbv_index_create(sdbv_ctx * /* nullable */, uint64_t table, uint32_t d,
                      uint8_t metric, uint32_t m, uint32_t m0, uint32_t efc,
                      int extend_candidates, int keep_pruned,
                      uint64_t rng_seed, double ml, sdbv_index **out);
void sdbv_index_destroy(sdbv_index *);
/* HnswIndex::index (index.rs:138-186): append one pending update (old/new
 * vectors, n*d f32 each; the id kind — DocId vs RecordKey — resolves via
 * the record-key map exactly as HnswDocs::get_doc_id does). */
int sdbv_index_enqueue(sdbv_index *, uint64_t record_key, const float *olds,
                       uint32_t n_old, const float *news, uint32_t n_new);
/* HnswIndex::index_pendings (index.rs:188-257): drain + apply the queue in
 * appending order through VecDocs (insert/remove, graph element removal
 * with repair, doc-id recycling). */
int sdbv_index_apply_pendings(sdbv_index *, uint64_t *out_count);
/* HnswIndex::knn_search (index.rs:270-335 minus record materialisation):
 * pendings overlay + graph search (pending docs excluded from the
 * expansion frontier, layer.rs:320-338) + Ids64 doc expansion through one
 * KnnResultBuilder. Out arrays sized k; entries ascending
 * (dist total_cmp, VectorId); kind 0 = DocId, 1 = RecordKey handle. */
int sdbv_index_knn(sdbv_index *, const float *q, uint32_t k, uint32_t ef,
                   uint8_t *out_kinds, uint64_t *out_ids, double *out_dists,
                   uint32_t *out_n);
/* Filtered KNN (cond_filter pushdown — hnsw/filter.rs + layer.rs:110-318):
 * the host evaluates the WHERE condition per VectorId (is_record_truthy,
 * filter.rs:111-138 — KV fetch + expression compute stay host-side); the
 * library keeps the FilterCache semantics (one evaluation per id while
 * cached; `expire` fires when the result builder evicts an id,
 * filter.rs:141-151) and all accept/expand gating (add_if_truthy,
 * layer.rs:278-306). `truthy` must be deterministic within one call;
 * `expire` may be NULL. */
typedef int (*sdbv_truthy_cb)(void *user, uint8_t kind, uint64_t id);
typedef void (*sdbv_expire_cb)(void *user, uint8_t kind, uint64_t id);
int sdbv_index_knn_filtered(sdbv_index *, const float *q, uint32_t k,
                            uint32_t ef, sdbv_truthy_cb truthy,
                            sdbv_expire_cb expire, void *user,
                            uint8_t *out_kinds, uint64_t *out_ids,
                            double *out_dists, uint32_t *out_n);
uint64_t sdbv_index_doc_count(sdbv_index *);
uint64_t sdbv_index_pending_count(sdbv_index *);
/* check_hnsw_properties (hnsw/mod.rs:561-570): 0 == OK. */
int sdbv_index_check_props(sdbv_index *, uint64_t expected_count);
/* Test access to the underlying graph (CSR parity vs the oracle). */
sdbv_hnsw *sdbv_index_hnsw(sdbv_index *);
/* Level-RNG state carry-over (extension): the reference's insert-level RNG
 * is entropy-seeded per process (hnsw/mod.rs:263-266) so any state is
 * conformant; persisting it keeps this framework's same-seed determinism
 * across cold starts. */
uint64_t sdbv_index_level_rng(sdbv_index *);
void sdbv_index_set_level_rng(sdbv_index *, uint64_t state);

/* ------------------------------------------------------------------------
 * KV codec + cold-start staging pipeline: bulk-load a
 * dumped surrealdb HNSW index — He element vectors (key/index/he.rs), Hn
 * per-node edge lists (hn.rs), Hs graph state (hs.rs), Hv vector->docs
 * entries (hv.rs) — straight into the host graph (no re-insertion), then
 * finalize stages it to the device SoA. Key/value byte formats are pinned
 * by the reference's own key tests (he.rs/hn.rs/hs.rs/hv.rs golden
 * bytes); values follow the `revision` 0.17.0 wire format. Post-Hl
 * (per-node Hn) storage only; legacy Hl chunks and Bits (roaring) doc
 * sets are rejected with SDBV_ERR_UNSUPPORTED. hd/hi/hp/hh pairs are
 * host-kept and skipped by the loader; the host re-binds record keys via
 * sdbv_index_bind_doc_key. */
typedef struct sdbv_kvload sdbv_kvload;
int sdbv_kvload_new(sdbv_ctx * /* nullable */, uint32_t d, uint8_t metric,
                    uint32_t m, uint32_t m0, uint32_t efc,
                    int extend_candidates, int keep_pruned, uint64_t rng_seed,
                    double ml, sdbv_kvload **out);
int sdbv_kvload_feed(sdbv_kvload *, const uint8_t *key, uint64_t klen,
                     const uint8_t *val, uint64_t vlen);
int sdbv_kvload_finish_hnsw(sdbv_kvload *, sdbv_hnsw **out);
int sdbv_kvload_finish_index(sdbv_kvload *, uint64_t table, sdbv_index **out);
void sdbv_kvload_abort(sdbv_kvload *);
int sdbv_index_bind_doc_key(sdbv_index *, uint64_t doc_id,
                            uint64_t record_key);
/* Export the doc-id <-> record-key map (the hi/hd state the host persists
 * for cold starts). Returns the total entry count; fills up to cap. */
uint64_t sdbv_index_doc_keys(sdbv_index *, uint64_t *out_docs,
                             uint64_t *out_keys, uint64_t cap);
/* Dump the graph / index back to reference-format KV pairs (round-trip +
 * migration tooling). The callback returns non-zero to abort. */
typedef int (*sdbv_kv_write_cb)(void *user, const uint8_t *key, uint64_t klen,
                                const uint8_t *val, uint64_t vlen);
int sdbv_hnsw_dump_kv(sdbv_hnsw *, uint32_t ns, uint32_t db, const char *tb,
                      uint32_t ix, sdbv_kv_write_cb, void *user);
int sdbv_index_dump_kv(sdbv_index *, uint32_t ns, uint32_t db, const char *tb,
                       uint32_t ix, sdbv_kv_write_cb, void *user);

#ifdef __cplusplus
}
#endif

#endif /* SDBV_H */
