"""Host-side mirror of the reference's Knn operator surface.

The reference is compiled (Rust) code, so the real host side of this framework
is the C++ behind the C-ABI (csrc/sdbv.hip); this module is the
harness-facing mirror of the operator interface, with the same names,
argument meaning and result semantics as the reference seams it replaces:

 - KnnTopK                 <- surrealdb/core/src/exec/operators/knn_topk.rs
 - HnswIndex               <- surrealdb/core/src/idx/trees/hnsw/index.rs
   (index :138, index_pendings :188, knn_search :270, check_state :260)

PRODUCT PATH ONLY: everything here routes to the HIP extension; no CPU
fallback, no oracle imports.
"""
import numpy as np

from . import Context, Index, SdbvError


class KnnTopK:
    """Brute-force KNN operator (pipeline-breaking), GPU-native.

    Reference: KnnTopK::new(input, field, query_vector, k, distance)
    (knn_topk.rs:100-118). Here `input` is a staged table id in a Context
    (the vectors were extracted and staged once, replacing the per-record
    extract_vector + Distance::compute hot loop, knn_topk.rs:196-227).

    Results: ascending (distance, id) — identical ordering to the
    reference's (distance, insertion-seq) for monotone ids.
    """

    def __init__(self, ctx: Context, table: int, query_vector, k: int,
                 distance: str = "cosine"):
        from . import METRICS
        if distance not in METRICS:
            raise SdbvError(
                f"KnnTopK: unknown distance {distance!r} "
                f"(catalog::Distance variants: {sorted(METRICS)})")
        self.ctx = ctx
        self.table = table
        self.query_vector = np.ascontiguousarray(query_vector, dtype=np.float32)
        self.k = int(k)
        self.distance = distance

    def execute(self):
        """Consume the staged table; return (ids, dists) ascending."""
        return self.ctx.knn_bruteforce(self.table, self.query_vector, self.k)


class HnswIndex:
    """HNSW index operator surface (hnsw/index.rs::HnswIndex).

    Same call shapes as the reference seams; record ids are opaque u64
    handles (the Rust binding's RecordIdKey map — INTEGRATION.md
    "record-key handles"), and Value->vector extraction
    (content_to_vectors, index.rs:118-129) is the caller's job: vectors
    arrive as (n, d) f32 arrays.
    """

    def __init__(self, ctx, table, dimension, distance="euclidean", m=12,
                 m0=None, efc=150, extend_candidates=False,
                 keep_pruned_connections=False, seed=0x5DB1):
        self._ix = Index(ctx, table, dimension, metric=distance, m=m, m0=m0,
                         efc=efc, extend=extend_candidates,
                         keep=keep_pruned_connections, seed=seed)
        self.dim = dimension

    def index(self, rid, old_values=None, new_values=None):
        """Lock-free enqueue of one vector update (index.rs:138-186).
        `rid` is the record-key handle; values are (n, d) f32 arrays or
        None (the reference's Option<Vec<Value>>)."""
        if old_values is None and new_values is None:
            return  # index.rs:145-147
        self._ix.enqueue(rid, old_values, new_values)

    def index_pendings(self):
        """Drain + apply the pendings queue (index.rs:188-211); returns the
        number of updates applied."""
        return self._ix.apply_pendings()

    def knn_search(self, pt, k, ef, cond_filter=None):
        """Pendings-merged KNN search (index.rs:270-335). `cond_filter` is
        the host's WHERE evaluation `truthy(kind, id) -> bool`
        (HnswTruthyDocumentFilter's is_record_truthy seam). Returns
        (kinds, ids, dists) ascending — kind 0 ids are DocIds, kind 1 ids
        are record-key handles (VectorId, hnsw/mod.rs:107-115)."""
        if cond_filter is None:
            return self._ix.knn_search(pt, k, ef)
        return self._ix.knn_search_filtered(pt, k, ef, cond_filter)

    def check_state(self):
        """index.rs:260: ensure in-memory layers match persisted state —
        an in-memory index is always current; a cold start goes through
        surrealdb_amd.load_kv_index instead."""
        return None

    def destroy(self):
        self._ix.destroy()
