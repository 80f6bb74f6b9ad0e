"""surrealdb_amd — MI355X-native SurrealDB vector-KNN hot path.

PRODUCT PATH. Loads the in-tree HIP C-ABI library (libsdbv.so, gfx950) and
mirrors the reference's Index/Knn operator surface (see host.py). On a GPU
machine this package FAILS LOUDLY if the HIP extension is missing or a GPU
call fails — there is no CPU fallback here (the parity oracle under oracle/
is test infrastructure and is never imported by this package).
"""
import ctypes
import os

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libsdbv.so")

METRIC_COSINE = 0
METRIC_EUCLIDEAN = 1
# catalog::Distance (index.rs:250-284) — all 8 on the GPU scan
METRICS = {
    "cosine": 0,
    "euclidean": 1,
    "manhattan": 2,
    "chebyshev": 3,
    "hamming": 4,
    "jaccard": 5,
    "minkowski": 6,
    "pearson": 7,
}

TRUTHY_CB = None  # ctypes callback types, set when lib() loads
EXPIRE_CB = None
KV_WRITE_CB = None

_ERRS = {
    0: "OK",
    -1: "HIP runtime error",
    -2: "table not staged",
    -3: "bad argument",
    -4: "device OOM",
    -5: "unsupported",
    -6: "ids not strictly increasing",
}


class SdbvError(RuntimeError):
    pass


class _Stats(ctypes.Structure):
    _fields_ = [
        ("last_scan_kernel_ms", ctypes.c_double),
        ("last_merge_kernel_ms", ctypes.c_double),
        ("last_total_ms", ctypes.c_double),
        ("bytes_staged", ctypes.c_uint64),
        ("last_rows_scanned", ctypes.c_uint64),
    ]


_lib = None


def lib():
    """Load libsdbv.so. Raises if absent — no silent fallback."""
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_SO):
        raise SdbvError(
            f"HIP extension not built: {_SO} missing. Run "
            "python surrealdb_amd/build.py (hipcc --offload-arch=gfx950)."
        )
    L = ctypes.CDLL(_SO)
    u64, u32, u8 = ctypes.c_uint64, ctypes.c_uint32, ctypes.c_uint8
    f32p = ctypes.POINTER(ctypes.c_float)
    f64p = ctypes.POINTER(ctypes.c_double)
    u64p = ctypes.POINTER(ctypes.c_uint64)
    u32p = ctypes.POINTER(ctypes.c_uint32)
    vp = ctypes.c_void_p
    L.sdbv_init.argtypes = [ctypes.c_int, ctypes.POINTER(vp)]
    L.sdbv_shutdown.argtypes = [vp]
    L.sdbv_last_error.restype = ctypes.c_char_p
    L.sdbv_last_error.argtypes = [vp]
    L.sdbv_get_stats.argtypes = [vp, ctypes.POINTER(_Stats)]
    L.sdbv_stage_corpus.argtypes = [vp, u64, f32p, u64p, u64, u32, u8]
    L.sdbv_stage_synthetic.argtypes = [vp, u64, u64, u32, u8, u64, u64, u64]
    L.sdbv_table_rows.restype = u64
    L.sdbv_table_rows.argtypes = [vp, u64]
    L.sdbv_drop_table.argtypes = [vp, u64]
    L.sdbv_knn_bruteforce.argtypes = [vp, u64, f32p, u32, u32, u64p, f64p, u32p]
    L.sdbv_all_distances.argtypes = [vp, u64, f32p, u32, f64p]
    L.sdbv_gather_distance.argtypes = [vp, u64, u32p, u32, f32p, u32, f64p]
    L.sdbv_knn_batch.argtypes = [vp, u64, f32p, u32, u32, u32, u64p, f64p]
    L.sdbv_table_set_order.argtypes = [vp, u64, ctypes.c_double]
    L.sdbv_hnsw_create.argtypes = [vp, u32, u8, u32, u32, u32, ctypes.c_int,
                                   ctypes.c_int, u64, ctypes.c_double,
                                   ctypes.POINTER(vp)]
    L.sdbv_hnsw_insert.argtypes = [vp, f32p]
    L.sdbv_hnsw_insert_batch.argtypes = [vp, f32p, u64, ctypes.c_int]
    L.sdbv_hnsw_insert_batch_snapshot.argtypes = [vp, f32p, u64, u32,
                                                  ctypes.c_int]
    L.sdbv_hnsw_insert_batch_snapshot2.argtypes = [vp, f32p, u64, u32,
                                                   ctypes.c_int]
    L.sdbv_hnsw_insert_batch_snapshot_gpu.argtypes = [vp, f32p, u64, u32,
                                                      ctypes.c_int]
    L.sdbv_hnsw_finalize.argtypes = [vp, u64]
    L.sdbv_hnsw_knn.argtypes = [vp, f32p, u32, u32, u64p, f64p, u32p]
    L.sdbv_hnsw_knn_batch.argtypes = [vp, f32p, u32, u32, u32, u64p, f64p, u32p]
    L.sdbv_hnsw_destroy.argtypes = [vp]
    L.sdbv_hnsw_n.restype = u64
    L.sdbv_hnsw_n.argtypes = [vp]
    L.sdbv_hnsw_layers.restype = u32
    L.sdbv_hnsw_layers.argtypes = [vp]
    L.sdbv_hnsw_l0_edge_count.restype = u64
    L.sdbv_hnsw_l0_edge_count.argtypes = [vp]
    L.sdbv_hnsw_l0_export.argtypes = [vp, u32p, u32p]
    L.sdbv_hnsw_layer_edge_count.restype = u64
    L.sdbv_hnsw_layer_edge_count.argtypes = [vp, u32]
    L.sdbv_hnsw_layer_export.argtypes = [vp, u32, u32p, u32p,
                                         ctypes.POINTER(u8)]
    L.sdbv_hnsw_enter_point.restype = ctypes.c_int64
    L.sdbv_hnsw_enter_point.argtypes = [vp]
    L.sdbv_hnsw_vecs_ptr.restype = ctypes.POINTER(ctypes.c_float)
    L.sdbv_hnsw_vecs_ptr.argtypes = [vp]
    L.sdbv_hnsw_remove.restype = ctypes.c_int
    L.sdbv_hnsw_remove.argtypes = [vp, u64]
    L.sdbv_hnsw_knn_host.argtypes = [vp, f32p, u32, u32, u64p, f64p, u32p]
    u8p = ctypes.POINTER(u8)
    L.sdbv_index_create.argtypes = [vp, u64, u32, u8, u32, u32, u32,
                                    ctypes.c_int, ctypes.c_int, u64,
                                    ctypes.c_double, ctypes.POINTER(vp)]
    L.sdbv_index_destroy.argtypes = [vp]
    L.sdbv_index_enqueue.argtypes = [vp, u64, f32p, u32, f32p, u32]
    L.sdbv_index_apply_pendings.argtypes = [vp, u64p]
    L.sdbv_index_knn.argtypes = [vp, f32p, u32, u32, u8p, u64p, f64p, u32p]
    global TRUTHY_CB, EXPIRE_CB
    TRUTHY_CB = ctypes.CFUNCTYPE(ctypes.c_int, vp, u8, u64)
    EXPIRE_CB = ctypes.CFUNCTYPE(None, vp, u8, u64)
    L.sdbv_index_knn_filtered.argtypes = [vp, f32p, u32, u32, TRUTHY_CB,
                                          EXPIRE_CB, vp, u8p, u64p, f64p,
                                          u32p]
    L.sdbv_index_doc_count.restype = u64
    L.sdbv_index_doc_count.argtypes = [vp]
    L.sdbv_index_pending_count.restype = u64
    L.sdbv_index_pending_count.argtypes = [vp]
    L.sdbv_index_check_props.restype = ctypes.c_int
    L.sdbv_index_check_props.argtypes = [vp, u64]
    L.sdbv_index_hnsw.restype = vp
    L.sdbv_index_hnsw.argtypes = [vp]
    u8pp = ctypes.POINTER(u8)
    L.sdbv_kvload_new.argtypes = [vp, u32, u8, u32, u32, u32, ctypes.c_int,
                                  ctypes.c_int, u64, ctypes.c_double,
                                  ctypes.POINTER(vp)]
    L.sdbv_kvload_feed.argtypes = [vp, u8pp, u64, u8pp, u64]
    L.sdbv_kvload_finish_hnsw.argtypes = [vp, ctypes.POINTER(vp)]
    L.sdbv_kvload_finish_index.argtypes = [vp, u64, ctypes.POINTER(vp)]
    L.sdbv_kvload_abort.argtypes = [vp]
    L.sdbv_index_bind_doc_key.argtypes = [vp, u64, u64]
    L.sdbv_index_doc_keys.restype = u64
    L.sdbv_index_doc_keys.argtypes = [vp, u64p, u64p, u64]
    L.sdbv_index_level_rng.restype = u64
    L.sdbv_index_level_rng.argtypes = [vp]
    L.sdbv_index_set_level_rng.argtypes = [vp, u64]
    global KV_WRITE_CB
    KV_WRITE_CB = ctypes.CFUNCTYPE(ctypes.c_int, vp, u8pp, u64, u8pp, u64)
    L.sdbv_hnsw_dump_kv.argtypes = [vp, u32, u32, ctypes.c_char_p, u32,
                                    KV_WRITE_CB, vp]
    L.sdbv_index_dump_kv.argtypes = [vp, u32, u32, ctypes.c_char_p, u32,
                                     KV_WRITE_CB, vp]
    _lib = L
    return L


def _check(ctx, rc, what):
    if rc != 0:
        msg = _ERRS.get(rc, str(rc))
        detail = ""
        if ctx is not None:
            try:
                detail = lib().sdbv_last_error(ctx).decode()
            except Exception:
                pass
        raise SdbvError(f"{what}: {msg} {detail}")


class Context:
    """Owns the device context (HIP stream + staged tables)."""

    def __init__(self, device=-1):
        import numpy as np  # noqa: F401  (numpy required by callers)
        self._ptr = ctypes.c_void_p()
        _check(None, lib().sdbv_init(device, ctypes.byref(self._ptr)), "sdbv_init")

    def close(self):
        if self._ptr:
            lib().sdbv_shutdown(self._ptr)
            self._ptr = ctypes.c_void_p()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def stats(self):
        s = _Stats()
        _check(self._ptr, lib().sdbv_get_stats(self._ptr, ctypes.byref(s)),
               "sdbv_get_stats")
        return {f: getattr(s, f) for f, _ in s._fields_}

    def stage_corpus(self, table, rows, ids=None, metric="cosine",
                     order=None):
        import numpy as np
        rows = np.ascontiguousarray(rows, dtype=np.float32)
        n, d = rows.shape
        idp = None
        if ids is not None:
            ids = np.ascontiguousarray(ids, dtype=np.uint64)
            idp = ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64))
        _check(self._ptr, lib().sdbv_stage_corpus(
            self._ptr, table, rows.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            idp, n, d, METRICS[metric]), "sdbv_stage_corpus")
        if order is not None:
            _check(self._ptr,
                   lib().sdbv_table_set_order(self._ptr, table, order),
                   "sdbv_table_set_order")

    def stage_synthetic(self, table, n, d, metric="cosine", seed=0x5DB1,
                        row_offset=0, id_base=0):
        _check(self._ptr, lib().sdbv_stage_synthetic(
            self._ptr, table, n, d, METRICS[metric], seed, row_offset, id_base),
            "sdbv_stage_synthetic")

    def table_rows(self, table):
        return lib().sdbv_table_rows(self._ptr, table)

    def drop_table(self, table):
        _check(self._ptr, lib().sdbv_drop_table(self._ptr, table), "drop_table")

    def knn_bruteforce(self, table, q, k):
        import numpy as np
        q = np.ascontiguousarray(q, dtype=np.float32)
        ids = np.empty(k, dtype=np.uint64)
        dists = np.empty(k, dtype=np.float64)
        out_n = ctypes.c_uint32(0)
        _check(self._ptr, lib().sdbv_knn_bruteforce(
            self._ptr, table, q.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            q.size, k,
            ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            ctypes.byref(out_n)), "sdbv_knn_bruteforce")
        m = out_n.value
        return ids[:m], dists[:m]

    def all_distances(self, table, q):
        import numpy as np
        q = np.ascontiguousarray(q, dtype=np.float32)
        n = self.table_rows(table)
        out = np.empty(n, dtype=np.float64)
        _check(self._ptr, lib().sdbv_all_distances(
            self._ptr, table, q.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            q.size, out.ctypes.data_as(ctypes.POINTER(ctypes.c_double))),
            "sdbv_all_distances")
        return out

    def gather_distance(self, table, rows, q):
        import numpy as np
        rows = np.ascontiguousarray(rows, dtype=np.uint32)
        q = np.ascontiguousarray(q, dtype=np.float32)
        out = np.empty(rows.size, dtype=np.float64)
        _check(self._ptr, lib().sdbv_gather_distance(
            self._ptr, table,
            rows.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)), rows.size,
            q.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), q.size,
            out.ctypes.data_as(ctypes.POINTER(ctypes.c_double))),
            "sdbv_gather_distance")
        return out

    def hnsw_create(self, d, metric="euclidean", m=12, m0=None, efc=150,
                    extend=False, keep=False, seed=0x5DB1, ml=None):
        import math
        if m0 is None:
            m0 = 2 * m
        if ml is None:
            ml = 1.0 / math.log(m)
        out = ctypes.c_void_p()
        _check(self._ptr, lib().sdbv_hnsw_create(
            self._ptr, d, METRICS[metric], m, m0, efc, int(extend), int(keep),
            seed, ml, ctypes.byref(out)), "sdbv_hnsw_create")
        return Hnsw(self, out, d)

    def knn_batch(self, table, Q, k):
        import numpy as np
        Q = np.ascontiguousarray(Q, dtype=np.float32)
        b, d = Q.shape
        ids = np.empty((b, k), dtype=np.uint64)
        dists = np.empty((b, k), dtype=np.float64)
        _check(self._ptr, lib().sdbv_knn_batch(
            self._ptr, table, Q.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            b, d, k,
            ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double))),
            "sdbv_knn_batch")
        return ids, dists


def hnsw_create_host(d, metric="euclidean", m=12, m0=None, efc=150,
                     extend=False, keep=False, seed=0x5DB1, ml=None):
    """Host-only HNSW index (ctx = NULL): graph build + CSR export work on
    any machine; finalize()/knn_search() fail loudly (no GPU context).
    Lets the pure-CPU graph-build path be tested without a GPU."""
    import math
    if m0 is None:
        m0 = 2 * m
    if ml is None:
        ml = 1.0 / math.log(m)
    out = ctypes.c_void_p()
    _check(None, lib().sdbv_hnsw_create(
        None, d, METRICS[metric], m, m0, efc, int(extend), int(keep),
        seed, ml, ctypes.byref(out)), "sdbv_hnsw_create")
    return Hnsw(None, out, d)


class _NullCtx:
    _ptr = None


class Hnsw:
    """Product HNSW index (mirrors HnswIndex, hnsw/index.rs): host graph +
    GPU layer-0 expansion. Element ids are insertion ordinals."""

    def __init__(self, ctx, ptr, d):
        self._ctx = ctx if ctx is not None else _NullCtx()
        self._ptr = ptr
        self.d = d

    def insert(self, pt):
        import numpy as np
        pt = np.ascontiguousarray(pt, dtype=np.float32)
        _check(self._ctx._ptr, lib().sdbv_hnsw_insert(
            self._ptr, pt.ctypes.data_as(ctypes.POINTER(ctypes.c_float))),
            "sdbv_hnsw_insert")

    def insert_batch(self, pts, nthreads=0):
        import numpy as np
        pts = np.ascontiguousarray(pts, dtype=np.float32)
        _check(self._ctx._ptr, lib().sdbv_hnsw_insert_batch(
            self._ptr, pts.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            pts.shape[0], nthreads), "sdbv_hnsw_insert_batch")

    def insert_batch_snapshot(self, pts, chunk, nthreads=0):
        """Chunked snapshot bulk build (bench mode; §8f rank 3 structure —
        per-chunk searches against the chunk-start graph, GPU-batchable)."""
        import numpy as np
        pts = np.ascontiguousarray(pts, dtype=np.float32)
        _check(self._ctx._ptr if self._ctx else None,
               lib().sdbv_hnsw_insert_batch_snapshot(
                   self._ptr,
                   pts.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
                   pts.shape[0], chunk, nthreads),
               "sdbv_hnsw_insert_batch_snapshot")

    def insert_batch_snapshot2(self, pts, chunk, nthreads=0):
        """Snapshot build, batched-apply schedule (host twin of the GPU
        build: all selects, then appends in element order, then one prune
        pass; upper elements' layer-0 half joins the batch)."""
        import numpy as np
        pts = np.ascontiguousarray(pts, dtype=np.float32)
        _check(self._ctx._ptr if self._ctx else None,
               lib().sdbv_hnsw_insert_batch_snapshot2(
                   self._ptr,
                   pts.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
                   pts.shape[0], chunk, nthreads),
               "sdbv_hnsw_insert_batch_snapshot2")

    def insert_batch_snapshot_gpu(self, pts, chunk, nthreads=0):
        """GPU-accelerated chunked snapshot build: per-chunk level-0
        efc-searches as one persistent-kernel launch; same resulting graph
        as insert_batch_snapshot (device context required)."""
        import numpy as np
        pts = np.ascontiguousarray(pts, dtype=np.float32)
        _check(self._ctx._ptr,
               lib().sdbv_hnsw_insert_batch_snapshot_gpu(
                   self._ptr,
                   pts.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
                   pts.shape[0], chunk, nthreads),
               "sdbv_hnsw_insert_batch_snapshot_gpu")

    def finalize(self, table):
        _check(self._ctx._ptr, lib().sdbv_hnsw_finalize(self._ptr, table),
               "sdbv_hnsw_finalize")

    def knn_search_host(self, q, k, ef):
        """Host-side search (no device): the build path's algorithm over
        the host graph — for CPU tests and quality audits."""
        import numpy as np
        q = np.ascontiguousarray(q, dtype=np.float32)
        ids = np.empty(k, dtype=np.uint64)
        dists = np.empty(k, dtype=np.float64)
        out_n = ctypes.c_uint32(0)
        _check(None, lib().sdbv_hnsw_knn_host(
            self._ptr, q.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), k,
            ef, ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            ctypes.byref(out_n)), "sdbv_hnsw_knn_host")
        n = out_n.value
        return ids[:n], dists[:n]

    def knn_search(self, q, k, ef):
        import numpy as np
        q = np.ascontiguousarray(q, dtype=np.float32)
        cap = max(k, ef)
        ids = np.empty(cap, dtype=np.uint64)
        dists = np.empty(cap, dtype=np.float64)
        out_n = ctypes.c_uint32(0)
        _check(self._ctx._ptr, lib().sdbv_hnsw_knn(
            self._ptr, q.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), k, ef,
            ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            ctypes.byref(out_n)), "sdbv_hnsw_knn")
        n = out_n.value
        return ids[:n], dists[:n]

    def knn_search_batch(self, Q, k, ef):
        """Batched ef-search on the persistent kernel (one query per
        workgroup); same exact result contract as knn_search."""
        import numpy as np
        Q = np.ascontiguousarray(Q, dtype=np.float32)
        b = Q.shape[0]
        ids = np.empty((b, k), dtype=np.uint64)
        dists = np.empty((b, k), dtype=np.float64)
        ns = np.empty(b, dtype=np.uint32)
        _check(self._ctx._ptr, lib().sdbv_hnsw_knn_batch(
            self._ptr, Q.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), b, k,
            ef, ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            ns.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32))),
            "sdbv_hnsw_knn_batch")
        return ids, dists, ns

    def n(self):
        return lib().sdbv_hnsw_n(self._ptr)

    def num_layers(self):
        return lib().sdbv_hnsw_layers(self._ptr)

    def l0_csr(self):
        import numpy as np
        n = self.n()
        ec = lib().sdbv_hnsw_l0_edge_count(self._ptr)
        offsets = np.empty(n + 1, dtype=np.uint32)
        edges = np.empty(max(ec, 1), dtype=np.uint32)
        lib().sdbv_hnsw_l0_export(
            self._ptr,
            offsets.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
            edges.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)))
        return offsets, edges[:ec]

    def dump_kv(self, ns=1, db=2, tb="testtb", ix=3):
        """Dump the graph as reference-format (key, value) KV pairs
        (He/Hn/Hs; key/index/{he,hn,hs}.rs byte layouts)."""
        return _dump_kv(lib().sdbv_hnsw_dump_kv, self._ptr, ns, db, tb, ix)

    def layer_csr(self, l):
        """(offsets, edges, in_layer) of layer l."""
        import numpy as np
        n = self.n()
        ec = lib().sdbv_hnsw_layer_edge_count(self._ptr, l)
        offsets = np.empty(n + 1, dtype=np.uint32)
        edges = np.empty(max(ec, 1), dtype=np.uint32)
        in_layer = np.empty(n, dtype=np.uint8)
        u32p = ctypes.POINTER(ctypes.c_uint32)
        lib().sdbv_hnsw_layer_export(
            self._ptr, l, offsets.ctypes.data_as(u32p),
            edges.ctypes.data_as(u32p),
            in_layer.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)))
        return offsets, edges[:ec], in_layer

    def enter_point(self):
        return lib().sdbv_hnsw_enter_point(self._ptr)

    def vecs_view(self):
        """Zero-copy numpy view of the host n x d f32 row store."""
        import numpy as np
        n = self.n()
        ptr = lib().sdbv_hnsw_vecs_ptr(self._ptr)
        return np.ctypeslib.as_array(ptr, shape=(n, self.d))

    def remove(self, e_id):
        """Hnsw::remove (hnsw/mod.rs:398-455). True if removed. Host graphs
        only (pre-finalize); finalized indexes mutate through Index."""
        rc = lib().sdbv_hnsw_remove(self._ptr, e_id)
        if rc < 0:
            raise SdbvError(f"sdbv_hnsw_remove: {_ERRS.get(rc, rc)}")
        return bool(rc)

    def destroy(self):
        if not getattr(self, "_owned", True):
            self._ptr = None  # view over an Index-owned graph
            return
        if self._ptr:
            lib().sdbv_hnsw_destroy(self._ptr)
            self._ptr = None


def index_create_host(d, metric="euclidean", m=12, m0=None, efc=150,
                      extend=False, keep=False, seed=0x5DB1, ml=None):
    """Host-only Index (ctx = NULL): the full write path + pendings-merged
    search on the host graph — the CPU-testable configuration."""
    return Index(None, 0, d, metric, m, m0, efc, extend, keep, seed, ml)


def _kvload_new(ctx, d, metric, m, m0, efc, extend, keep, seed, ml):
    import math
    if m0 is None:
        m0 = 2 * m
    if ml is None:
        ml = 1.0 / math.log(m)
    out = ctypes.c_void_p()
    cptr = ctx._ptr if ctx is not None else None
    _check(cptr, lib().sdbv_kvload_new(
        cptr, d, METRICS[metric], m, m0, efc, int(extend), int(keep), seed,
        ml, ctypes.byref(out)), "sdbv_kvload_new")
    return out


def _kvload_feed_all(loader, pairs):
    u8p = ctypes.POINTER(ctypes.c_uint8)
    for key, val in pairs:
        k = (ctypes.c_uint8 * len(key)).from_buffer_copy(key)
        v = (ctypes.c_uint8 * max(len(val), 1)).from_buffer_copy(
            val if val else b"\0")
        rc = lib().sdbv_kvload_feed(loader, ctypes.cast(k, u8p), len(key),
                                    ctypes.cast(v, u8p), len(val))
        if rc != 0:
            lib().sdbv_kvload_abort(loader)
            raise SdbvError(f"sdbv_kvload_feed: {_ERRS.get(rc, rc)}")


def load_kv_hnsw(pairs, d, metric="euclidean", m=12, m0=None, efc=150,
                 extend=False, keep=False, seed=0x5DB1, ml=None, ctx=None):
    """Cold-start bulk load of a dumped reference HNSW graph (He/Hn/Hs KV
    pairs) into a host graph — no re-insertion (key/index/{he,hn,hs}.rs +
    HnswLayer::load, layer.rs:504-563)."""
    loader = _kvload_new(ctx, d, metric, m, m0, efc, extend, keep, seed, ml)
    _kvload_feed_all(loader, pairs)
    out = ctypes.c_void_p()
    _check(None, lib().sdbv_kvload_finish_hnsw(loader, ctypes.byref(out)),
           "sdbv_kvload_finish_hnsw")
    return Hnsw(ctx, out, d)


def load_kv_index(pairs, table, d, metric="euclidean", m=12, m0=None,
                  efc=150, extend=False, keep=False, seed=0x5DB1, ml=None,
                  ctx=None, doc_keys=None):
    """Cold-start bulk load of a dumped reference HNSW *index* (He/Hn/Hs/Hv
    pairs; hd/hi pairs are host-kept — pass their mapping as
    doc_keys={doc_id: record_key_handle})."""
    loader = _kvload_new(ctx, d, metric, m, m0, efc, extend, keep, seed, ml)
    _kvload_feed_all(loader, pairs)
    out = ctypes.c_void_p()
    _check(None, lib().sdbv_kvload_finish_index(loader, table,
                                                ctypes.byref(out)),
           "sdbv_kvload_finish_index")
    ix = Index.__new__(Index)
    ix._ctx = ctx
    ix._ptr = out
    ix.d = d
    for doc_id, key in (doc_keys or {}).items():
        _check(None, lib().sdbv_index_bind_doc_key(out, doc_id, key),
               "sdbv_index_bind_doc_key")
    return ix


def _dump_kv(fn, ptr, ns, db, tb, ix_id):
    pairs = []

    def w(user, key, klen, val, vlen):
        pairs.append((bytes(bytearray(key[i] for i in range(klen))),
                      bytes(bytearray(val[i] for i in range(vlen)))))
        return 0
    cb = KV_WRITE_CB(w)
    _check(None, fn(ptr, ns, db, tb.encode(), ix_id, cb, None), "dump_kv")
    return pairs


class Index:
    """Product HnswIndex (hnsw/index.rs operator surface): pendings queue,
    VecDocs/Ids64 doc expansion, doc-id allocation, pendings-merged
    knn_search. Record keys are opaque u64 handles (host's RecordIdKey
    mapping — INTEGRATION.md). With a Context, graph searches run the GPU
    per-hop path, auto-re-finalizing into `table` after writes."""

    def __init__(self, ctx, table, d, metric="euclidean", m=12, m0=None,
                 efc=150, extend=False, keep=False, seed=0x5DB1, ml=None):
        import math
        if m0 is None:
            m0 = 2 * m
        if ml is None:
            ml = 1.0 / math.log(m)
        out = ctypes.c_void_p()
        cptr = ctx._ptr if ctx is not None else None
        _check(cptr, lib().sdbv_index_create(
            cptr, table, d, METRICS[metric], m, m0, efc, int(extend),
            int(keep), seed, ml, ctypes.byref(out)), "sdbv_index_create")
        self._ctx = ctx
        self._ptr = out
        self.d = d

    def enqueue(self, record_key, old_vectors=None, new_vectors=None):
        """HnswIndex::index (index.rs:138-186). Vectors: (n, d) f32 arrays
        (None == no values of that kind)."""
        import numpy as np

        def flat(a):
            if a is None:
                return np.empty((0, self.d), dtype=np.float32)
            return np.ascontiguousarray(a, dtype=np.float32).reshape(
                -1, self.d)
        o, nw = flat(old_vectors), flat(new_vectors)
        _check(None, lib().sdbv_index_enqueue(
            self._ptr, record_key,
            o.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), o.shape[0],
            nw.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), nw.shape[0]),
            "sdbv_index_enqueue")

    def apply_pendings(self):
        n = ctypes.c_uint64(0)
        _check(None, lib().sdbv_index_apply_pendings(
            self._ptr, ctypes.byref(n)), "sdbv_index_apply_pendings")
        return n.value

    def knn_search(self, q, k, ef):
        """index.rs:270-335 minus record materialisation: (kinds u8
        [0=DocId, 1=RecordKey], ids u64, dists f64), ascending."""
        import numpy as np
        q = np.ascontiguousarray(q, dtype=np.float32)
        kinds = np.empty(k, dtype=np.uint8)
        ids = np.empty(k, dtype=np.uint64)
        dists = np.empty(k, dtype=np.float64)
        out_n = ctypes.c_uint32(0)
        _check(self._ctx._ptr if self._ctx else None, lib().sdbv_index_knn(
            self._ptr, q.ctypes.data_as(ctypes.POINTER(ctypes.c_float)),
            k, ef, kinds.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
            ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            ctypes.byref(out_n)), "sdbv_index_knn")
        n = out_n.value
        return kinds[:n], ids[:n], dists[:n]

    def knn_search_filtered(self, q, k, ef, truthy, expire=None):
        """Filtered knn (cond_filter pushdown): `truthy(kind, id) -> bool`
        is the host-side WHERE evaluation (filter.rs is_record_truthy);
        `expire(kind, id)` mirrors the filter-cache eviction signal."""
        import numpy as np
        q = np.ascontiguousarray(q, dtype=np.float32)
        kinds = np.empty(k, dtype=np.uint8)
        ids = np.empty(k, dtype=np.uint64)
        dists = np.empty(k, dtype=np.float64)
        out_n = ctypes.c_uint32(0)
        cb = TRUTHY_CB(lambda u, kind, i: 1 if truthy(kind, i) else 0)
        ex = EXPIRE_CB((lambda u, kind, i: expire(kind, i)) if expire
                       else (lambda u, kind, i: None))
        _check(self._ctx._ptr if self._ctx else None,
               lib().sdbv_index_knn_filtered(
                   self._ptr,
                   q.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), k, ef,
                   cb, ex, None,
                   kinds.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
                   ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
                   dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
                   ctypes.byref(out_n)), "sdbv_index_knn_filtered")
        n = out_n.value
        return kinds[:n], ids[:n], dists[:n]

    def doc_count(self):
        return lib().sdbv_index_doc_count(self._ptr)

    def level_rng(self):
        """Level-RNG state (extension; persist + restore across cold
        starts for same-seed graph determinism — any state is
        reference-conformant, the reference reseeds from entropy)."""
        return lib().sdbv_index_level_rng(self._ptr)

    def set_level_rng(self, state):
        lib().sdbv_index_set_level_rng(self._ptr, state)

    def doc_keys(self):
        """The doc-id -> record-key map (the hi/hd state a host persists
        for cold starts; pairs sdbv_index_bind_doc_key on reload)."""
        import numpy as np
        n = lib().sdbv_index_doc_keys(self._ptr, None, None, 0)
        docs = np.empty(max(n, 1), dtype=np.uint64)
        keys = np.empty(max(n, 1), dtype=np.uint64)
        lib().sdbv_index_doc_keys(
            self._ptr, docs.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            keys.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)), n)
        return {int(d): int(k) for d, k in zip(docs[:n], keys[:n])}

    def pending_count(self):
        return lib().sdbv_index_pending_count(self._ptr)

    def check_props(self, expected_count):
        return lib().sdbv_index_check_props(self._ptr, expected_count)

    def dump_kv(self, ns=1, db=2, tb="testtb", ix=3):
        """Dump graph + Hv vector->docs entries as reference-format KV
        pairs (He/Hn/Hs/Hv)."""
        return _dump_kv(lib().sdbv_index_dump_kv, self._ptr, ns, db, tb, ix)

    def hnsw(self):
        """Non-owning view of the underlying graph (parity introspection)."""
        h = Hnsw(self._ctx, ctypes.c_void_p(lib().sdbv_index_hnsw(self._ptr)),
                 self.d)
        h._owned = False
        return h

    def destroy(self):
        if self._ptr:
            lib().sdbv_index_destroy(self._ptr)
            self._ptr = None
