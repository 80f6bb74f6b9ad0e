"""Host-side mirror of the reference's Knn operator surface.

The reference is compiled (Rust) code, so the real host side of this framework
is the C++ behind the C-ABI (csrc/sdbv.hip); this module is the
harness-facing mirror of the operator interface, with the same names,
argument meaning and result semantics as the reference seams it replaces:

 - KnnTopK                 <- surrealdb/core/src/exec/operators/knn_topk.rs
 - knn (two-phase legacy)  <- surrealdb/core/src/idx/planner/knn.rs
 - HnswIndex.knn_search    <- surrealdb/core/src/idx/trees/hnsw/index.rs:270
   (HNSW search lands in the next milestone; see DESIGN.md round plan)

PRODUCT PATH ONLY: everything here routes to the HIP extension; no CPU
fallback, no oracle imports.
"""
import numpy as np

from . import Context, SdbvError


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
        if distance not in ("cosine", "euclidean"):
            raise SdbvError(
                f"KnnTopK GPU path supports cosine/euclidean; got {distance}")
        self.ctx = ctx
        self.table = table
        self.query_vector = np.ascontiguousarray(query_vector, dtype=np.float32)
        self.k = int(k)
        self.distance = distance

    def execute(self):
        """Consume the staged table; return (ids, dists) ascending."""
        return self.ctx.knn_bruteforce(self.table, self.query_vector, self.k)
