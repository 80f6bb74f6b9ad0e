# oracle — TEST INFRASTRUCTURE ONLY.
# ctypes wrapper around liborcl.so (the CPU restatement of the reference's
# vector-KNN semantics). Only tests/, __graft_entry__.smoke() and bench.py's
# cpu_baseline leg may import this package. The product path (surrealdb_amd)
# never touches it.
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liborcl.so")

TRUTHY_CB = None  # ctypes callback types, set when lib() loads
EXPIRE_CB = None

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


def build():
    subprocess.run(["make", "-C", _DIR, "-s"], check=True)


def _load():
    if not os.path.exists(_SO):
        build()
    lib = ctypes.CDLL(_SO)
    u64, u32, u8, f32p, f64p = (
        ctypes.c_uint64,
        ctypes.c_uint32,
        ctypes.c_uint8,
        ctypes.POINTER(ctypes.c_float),
        ctypes.POINTER(ctypes.c_double),
    )
    u64p = ctypes.POINTER(ctypes.c_uint64)
    u32p = ctypes.POINTER(ctypes.c_uint32)
    lib.orc_gen_elem.restype = ctypes.c_float
    lib.orc_gen_elem.argtypes = [u64, u64]
    lib.orc_gen_f32.argtypes = [u64, u64, u64, u32, f32p]
    lib.orc_dot_f32.restype = ctypes.c_float
    lib.orc_dot_f32.argtypes = [f32p, f32p, u64]
    lib.orc_dot_f64.restype = ctypes.c_double
    lib.orc_dot_f64.argtypes = [f64p, f64p, u64]
    lib.orc_sumsq_f32.restype = ctypes.c_float
    lib.orc_sumsq_f32.argtypes = [f32p, u64]
    lib.orc_sumsq_f64.restype = ctypes.c_double
    lib.orc_sumsq_f64.argtypes = [f64p, u64]
    for name in ("orc_dist_f32",):
        fn = getattr(lib, name)
        fn.restype = ctypes.c_double
        fn.argtypes = [u8, ctypes.c_double, f32p, f32p, u64]
    for name in ("orc_dist_f64", "orc_dist_number"):
        fn = getattr(lib, name)
        fn.restype = ctypes.c_double
        fn.argtypes = [u8, ctypes.c_double, f64p, f64p, u64]
    lib.orc_total_key.restype = u64
    lib.orc_total_key.argtypes = [ctypes.c_double]
    lib.orc_topk_f32.argtypes = [u8, ctypes.c_double, f32p, u64, u32, f32p, u32, u64p, f64p, u32p]
    lib.orc_topk_number.argtypes = [u8, ctypes.c_double, f64p, u64, u32, f64p, u32, u64p, f64p, u32p]
    lib.orc_topk_f32_mt.restype = ctypes.c_int
    lib.orc_topk_f32_mt.argtypes = [
        u8, ctypes.c_double, f32p, u64, u32, f32p, u32, u64p, f64p, u32p, ctypes.c_int,
    ]
    lib.orc_hnsw_new.restype = ctypes.c_void_p
    lib.orc_hnsw_new.argtypes = [u32, u8, ctypes.c_double, u32, u32, u32,
                                 ctypes.c_int, ctypes.c_int, u64, ctypes.c_double]
    lib.orc_hnsw_free.argtypes = [ctypes.c_void_p]
    lib.orc_hnsw_insert.argtypes = [ctypes.c_void_p, f32p]
    lib.orc_hnsw_insert_level.argtypes = [ctypes.c_void_p, f32p, u32]
    lib.orc_hnsw_search.restype = u32
    lib.orc_hnsw_search.argtypes = [ctypes.c_void_p, f32p, u32, u32, u64p, f64p]
    lib.orc_hnsw_check_props.restype = ctypes.c_int
    lib.orc_hnsw_check_props.argtypes = [ctypes.c_void_p]
    lib.orc_hnsw_num_layers.restype = u32
    lib.orc_hnsw_num_layers.argtypes = [ctypes.c_void_p]
    lib.orc_hnsw_num_elements.restype = u64
    lib.orc_hnsw_num_elements.argtypes = [ctypes.c_void_p]
    lib.orc_hnsw_l0_edge_count.restype = u64
    lib.orc_hnsw_l0_edge_count.argtypes = [ctypes.c_void_p]
    lib.orc_hnsw_l0_export.argtypes = [ctypes.c_void_p, u32p, u32p]
    lib.orc_hnsw_search_ep.argtypes = [ctypes.c_void_p, f32p, u64p, f64p]
    lib.orc_hnsw_entry_point.restype = ctypes.c_int64
    lib.orc_hnsw_entry_point.argtypes = [ctypes.c_void_p]
    lib.orc_hnsw_remove.restype = ctypes.c_int
    lib.orc_hnsw_remove.argtypes = [ctypes.c_void_p, u64]
    lib.orc_hnsw_import.restype = ctypes.c_void_p
    lib.orc_hnsw_import.argtypes = [u32, u8, ctypes.c_double, u32, u32, u32,
                                    u64, f32p, ctypes.c_int64, u32]
    lib.orc_hnsw_import_layer.restype = ctypes.c_int
    lib.orc_hnsw_import_layer.argtypes = [ctypes.c_void_p, u32, u32p, u32p,
                                          ctypes.POINTER(u8)]
    # index layer (hnsw/index.rs + docs.rs + knn.rs Ids64)
    u8p = ctypes.POINTER(u8)
    lib.orc_index_new.restype = ctypes.c_void_p
    lib.orc_index_new.argtypes = [u32, u8, ctypes.c_double, u32, u32, u32,
                                  ctypes.c_int, ctypes.c_int, u64,
                                  ctypes.c_double]
    lib.orc_index_free.argtypes = [ctypes.c_void_p]
    lib.orc_index_hnsw.restype = ctypes.c_void_p
    lib.orc_index_hnsw.argtypes = [ctypes.c_void_p]
    lib.orc_index_doc_count.restype = u64
    lib.orc_index_doc_count.argtypes = [ctypes.c_void_p]
    lib.orc_index_pending_count.restype = u64
    lib.orc_index_pending_count.argtypes = [ctypes.c_void_p]
    lib.orc_index_enqueue.restype = ctypes.c_int
    lib.orc_index_enqueue.argtypes = [ctypes.c_void_p, u64, f32p, u32, f32p,
                                      u32]
    lib.orc_index_apply.restype = u64
    lib.orc_index_apply.argtypes = [ctypes.c_void_p]
    lib.orc_index_knn.restype = u32
    lib.orc_index_knn.argtypes = [ctypes.c_void_p, f32p, u32, u32, u8p, u64p,
                                  f64p]
    global TRUTHY_CB, EXPIRE_CB
    TRUTHY_CB = ctypes.CFUNCTYPE(ctypes.c_int, ctypes.c_void_p, u8, u64)
    EXPIRE_CB = ctypes.CFUNCTYPE(None, ctypes.c_void_p, u8, u64)
    lib.orc_index_knn_filtered.restype = u32
    lib.orc_index_knn_filtered.argtypes = [ctypes.c_void_p, f32p, u32, u32,
                                           TRUTHY_CB, EXPIRE_CB,
                                           ctypes.c_void_p, u8p, u64p, f64p]
    lib.orc_index_check_props.restype = ctypes.c_int
    lib.orc_index_check_props.argtypes = [ctypes.c_void_p, u64]
    lib.orc_ids64_new.restype = ctypes.c_void_p
    lib.orc_ids64_free.argtypes = [ctypes.c_void_p]
    lib.orc_ids64_insert.restype = ctypes.c_int
    lib.orc_ids64_insert.argtypes = [ctypes.c_void_p, u64]
    lib.orc_ids64_remove.restype = ctypes.c_int
    lib.orc_ids64_remove.argtypes = [ctypes.c_void_p, u64]
    lib.orc_ids64_export.restype = u32
    lib.orc_ids64_export.argtypes = [ctypes.c_void_p, u64p,
                                     ctypes.POINTER(ctypes.c_int)]
    return lib


_lib = None


def lib():
    global _lib
    if _lib is None:
        _lib = _load()
    return _lib


def _f32p(a):
    assert a.dtype == np.float32 and a.flags.c_contiguous
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_float))


def _f64p(a):
    assert a.dtype == np.float64 and a.flags.c_contiguous
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_double))


def gen_f32(seed, row0, nrows, d):
    """The committed synthetic-data contract (see sdbv_oracle.cpp)."""
    out = np.empty((nrows, d), dtype=np.float32)
    lib().orc_gen_f32(seed, row0, nrows, d, _f32p(out))
    return out


def dist_f32(metric, a, b, order=0.0):
    return lib().orc_dist_f32(METRICS[metric], order, _f32p(a), _f32p(b), a.size)


def dist_f64(metric, a, b, order=0.0):
    return lib().orc_dist_f64(METRICS[metric], order, _f64p(a), _f64p(b), a.size)


def dist_number(metric, a, b, order=0.0):
    return lib().orc_dist_number(METRICS[metric], order, _f64p(a), _f64p(b), a.size)


def topk_f32(metric, corpus, q, k, order=0.0):
    n = corpus.shape[0]
    ids = np.empty(k, dtype=np.uint64)
    dists = np.empty(k, dtype=np.float64)
    out_n = ctypes.c_uint32(0)
    lib().orc_topk_f32(
        METRICS[metric], order, _f32p(corpus), n, corpus.shape[1], _f32p(q), k,
        ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        ctypes.byref(out_n))
    m = out_n.value
    return ids[:m], dists[:m]


def topk_number(metric, corpus, q, k, order=0.0):
    n = corpus.shape[0]
    ids = np.empty(k, dtype=np.uint64)
    dists = np.empty(k, dtype=np.float64)
    out_n = ctypes.c_uint32(0)
    lib().orc_topk_number(
        METRICS[metric], order, _f64p(corpus), n, corpus.shape[1], _f64p(q), k,
        ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        ctypes.byref(out_n))
    m = out_n.value
    return ids[:m], dists[:m]


def topk_f32_mt(metric, corpus, q, k, order=0.0, nthreads=0):
    n = corpus.shape[0]
    ids = np.empty(k, dtype=np.uint64)
    dists = np.empty(k, dtype=np.float64)
    out_n = ctypes.c_uint32(0)
    used = lib().orc_topk_f32_mt(
        METRICS[metric], order, _f32p(corpus), n, corpus.shape[1], _f32p(q), k,
        ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
        dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
        ctypes.byref(out_n), nthreads)
    m = out_n.value
    return ids[:m], dists[:m], used


class Hnsw:
    """Oracle HNSW (restates hnsw/mod.rs + layer.rs + heuristic.rs)."""

    def __init__(self, d, metric="euclidean", order=0.0, m=12, m0=None,
                 efc=150, extend=False, keep=False, seed=0x5DB1, ml=None):
        import math
        if m0 is None:
            m0 = 2 * m
        if ml is None:
            ml = 1.0 / math.log(m)
        self._h = lib().orc_hnsw_new(d, METRICS[metric], order, m, m0, efc,
                                     int(extend), int(keep), seed, ml)
        self.d = d

    def insert(self, pt, level=None):
        pt = np.ascontiguousarray(pt, dtype=np.float32)
        if level is None:
            lib().orc_hnsw_insert(self._h, _f32p(pt))
        else:
            lib().orc_hnsw_insert_level(self._h, _f32p(pt), level)

    def search(self, q, k, ef):
        q = np.ascontiguousarray(q, dtype=np.float32)
        cap = max(k, ef)
        ids = np.empty(cap, dtype=np.uint64)
        dists = np.empty(cap, dtype=np.float64)
        n = lib().orc_hnsw_search(
            self._h, _f32p(q), k, ef,
            ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double)))
        return ids[:n], dists[:n]

    def search_ep(self, q):
        q = np.ascontiguousarray(q, dtype=np.float32)
        ep = ctypes.c_uint64(0)
        dist = ctypes.c_double(0)
        lib().orc_hnsw_search_ep(self._h, _f32p(q), ctypes.byref(ep), ctypes.byref(dist))
        return ep.value, dist.value

    def check_props(self):
        return lib().orc_hnsw_check_props(self._h)

    def num_layers(self):
        return lib().orc_hnsw_num_layers(self._h)

    def l0_csr(self):
        n = lib().orc_hnsw_num_elements(self._h)
        ec = lib().orc_hnsw_l0_edge_count(self._h)
        offsets = np.empty(n + 1, dtype=np.uint32)
        edges = np.empty(max(ec, 1), dtype=np.uint32)
        lib().orc_hnsw_l0_export(
            self._h,
            offsets.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
            edges.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)))
        return offsets, edges[:ec]

    def entry_point(self):
        return lib().orc_hnsw_entry_point(self._h)

    def remove(self, e_id):
        """Hnsw::remove (hnsw/mod.rs:398-455). True if removed."""
        return bool(lib().orc_hnsw_remove(self._h, e_id))

    @classmethod
    def import_graph(cls, d, metric, m, m0, efc, vecs, enter_point, layers,
                     order=0.0):
        """Build an oracle searcher over a graph built elsewhere (the
        bench cpu_baseline leg searches the product's own graph). `vecs` is
        the n x d f32 row store; `layers` a list of (offsets u32[n+1],
        edges u32[], in_layer u8[n]) from layer 0 up."""
        vecs = np.ascontiguousarray(vecs, dtype=np.float32)
        n = vecs.shape[0]
        self = cls.__new__(cls)
        self.d = d
        self._h = lib().orc_hnsw_import(d, METRICS[metric], order, m, m0,
                                        efc, n, _f32p(vecs), enter_point,
                                        len(layers))
        u32p = ctypes.POINTER(ctypes.c_uint32)
        u8p = ctypes.POINTER(ctypes.c_uint8)
        for l, (offsets, edges, in_layer) in enumerate(layers):
            offsets = np.ascontiguousarray(offsets, dtype=np.uint32)
            edges = np.ascontiguousarray(edges, dtype=np.uint32)
            if edges.size == 0:
                edges = np.zeros(1, dtype=np.uint32)
            in_layer = np.ascontiguousarray(in_layer, dtype=np.uint8)
            rc = lib().orc_hnsw_import_layer(
                self._h, l, offsets.ctypes.data_as(u32p),
                edges.ctypes.data_as(u32p), in_layer.ctypes.data_as(u8p))
            assert rc == 0, rc
        return self

    def __del__(self):
        try:
            lib().orc_hnsw_free(self._h)
        except Exception:
            pass


class _HnswView(Hnsw):
    """Non-owning Hnsw view over an index's internal graph."""

    def __init__(self, ptr, d):
        self._h = ptr
        self.d = d

    def __del__(self):
        pass  # owned by the Index


class Ids64:
    """Test hook over the restated Ids64 (knn.rs:163-326). insert()/remove()
    return True when a new variant was produced (the reference's Some) —
    the contract VecDocs persists on."""

    def __init__(self):
        self._s = lib().orc_ids64_new()

    def insert(self, d):
        return bool(lib().orc_ids64_insert(self._s, d))

    def remove(self, d):
        return bool(lib().orc_ids64_remove(self._s, d))

    def export(self):
        out = np.empty(4096, dtype=np.uint64)
        is_bits = ctypes.c_int(0)
        n = lib().orc_ids64_export(
            self._s, out.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            ctypes.byref(is_bits))
        return out[:n].tolist(), bool(is_bits.value)

    def __del__(self):
        try:
            lib().orc_ids64_free(self._s)
        except Exception:
            pass


class Index:
    """Oracle HnswIndex (hnsw/index.rs): the pendings queue, VecDocs/Ids64
    doc expansion and the pendings-merged knn_search over the restated graph.
    Record keys are opaque u64 handles (the host's RecordIdKey mapping)."""

    def __init__(self, d, metric="euclidean", order=0.0, m=12, m0=None,
                 efc=150, extend=False, keep=False, seed=0x5DB1, ml=None):
        import math
        if m0 is None:
            m0 = 2 * m
        if ml is None:
            ml = 1.0 / math.log(m)
        self._ix = lib().orc_index_new(d, METRICS[metric], order, m, m0, efc,
                                       int(extend), int(keep), seed, ml)
        self.d = d

    def hnsw(self):
        return _HnswView(lib().orc_index_hnsw(self._ix), self.d)

    def enqueue(self, record_key, old_vectors=None, new_vectors=None):
        """HnswIndex::index (index.rs:138-186): one pending update. Vectors
        are (n, d) f32 arrays (None == no values of that kind)."""
        def flat(a):
            if a is None:
                return np.empty((0, self.d), dtype=np.float32)
            a = np.ascontiguousarray(a, dtype=np.float32).reshape(-1, self.d)
            return a
        o, nw = flat(old_vectors), flat(new_vectors)
        rc = lib().orc_index_enqueue(self._ix, record_key, _f32p(o),
                                     o.shape[0], _f32p(nw), nw.shape[0])
        assert rc == 0

    def apply_pendings(self):
        return lib().orc_index_apply(self._ix)

    def knn_search(self, q, k, ef):
        """index.rs:270-335 without the record materialisation: returns
        (kinds u8 [0=DocId, 1=RecordKey], ids u64, dists f64) ascending."""
        q = np.ascontiguousarray(q, dtype=np.float32)
        kinds = np.empty(k, dtype=np.uint8)
        ids = np.empty(k, dtype=np.uint64)
        dists = np.empty(k, dtype=np.float64)
        n = lib().orc_index_knn(
            self._ix, _f32p(q), k, ef,
            kinds.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
            ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double)))
        return kinds[:n], ids[:n], dists[:n]

    def knn_search_filtered(self, q, k, ef, truthy, expire=None):
        """Filtered knn (index.rs:270-335 with cond_filter): truthy(kind,
        id) -> bool is the host WHERE evaluation; expire mirrors the
        filter-cache eviction signal."""
        q = np.ascontiguousarray(q, dtype=np.float32)
        kinds = np.empty(k, dtype=np.uint8)
        ids = np.empty(k, dtype=np.uint64)
        dists = np.empty(k, dtype=np.float64)
        cb = TRUTHY_CB(lambda u, kind, i: 1 if truthy(kind, i) else 0)
        ex = EXPIRE_CB((lambda u, kind, i: expire(kind, i)) if expire
                       else (lambda u, kind, i: None))
        n = lib().orc_index_knn_filtered(
            self._ix, _f32p(q), k, ef, cb, ex, None,
            kinds.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)),
            ids.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            dists.ctypes.data_as(ctypes.POINTER(ctypes.c_double)))
        return kinds[:n], ids[:n], dists[:n]

    def doc_count(self):
        return lib().orc_index_doc_count(self._ix)

    def pending_count(self):
        return lib().orc_index_pending_count(self._ix)

    def check_props(self, expected_count):
        return lib().orc_index_check_props(self._ix, expected_count)

    def __del__(self):
        try:
            lib().orc_index_free(self._ix)
        except Exception:
            pass
