"""Generate committed HNSW oracle fixtures for the GPU parity tests.

The oracle build (oracle/sdbv_oracle.cpp, the restated reference algorithm)
is deterministic but slower than the product at 768-dim, so the GPU test suite
does NOT rebuild it on the GPU box: this script builds each test
configuration ONCE on a CPU box and commits the expected graph (layer-0 CSR,
entry point, layer count) and the builder-sorted search results. The GPU
tests then build only the product graph (C++ host build, the thing under
test) and compare graph + GPU search results against these fixtures.

Run from the repo root:  python tests/golden/make_hnsw_fixtures.py
"""
import math
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
import oracle  # noqa: E402

OUT_DIR = os.path.dirname(os.path.abspath(__file__))


def builder_sort(ids, dists):
    """KnnResultBuilder final ordering (dist f64 total_cmp, then id) applied
    to an oracle result (knn.rs:170-326 semantics; matches
    surrealdb_amd.shard.total_key without importing the product package)."""
    bits = dists.view(np.uint64).astype(np.int64)
    key = np.where(bits < 0, np.iinfo(np.int64).min - bits, bits)
    order = np.lexsort((ids, key))
    return ids[order], dists[order]


# Each fixture: (name, d, n, metric, m, m0, efc, seed, data_seed,
#                n_queries, query_seed, k, ef_list)
FIXTURES = [
    # persistent-kernel + per-hop cosine test (test_gpu_hnsw.py)
    ("seq768_cos", 768, 4000, "cosine", 16, 32, 150, 0x9, 0x5DB1,
     32, 0xBEEF, 10, (64,)),
    # per-hop euclidean test
    ("seq768_euc", 768, 3000, "euclidean", 12, 24, 150, 0x5DB1, 0x5DB1,
     20, 0xBEEF, 10, (64,)),
    # cosine d=128 test
    ("seq128_cos", 128, 5000, "cosine", 8, 16, 100, 0x11, 0x77,
     10, 0x88, 10, (40,)),
    # deeper-graph coverage (round-2 scale; ~5 min oracle build)
    ("seq768_cos_16k", 768, 16000, "cosine", 16, 32, 150, 0x10, 0x5DB1,
     48, 0xBEEF, 10, (64,)),
]


def main():
    for (name, d, n, metric, m, m0, efc, seed, dseed,
         nq, qseed, k, efs) in FIXTURES:
        t0 = time.perf_counter()
        rows = oracle.gen_f32(dseed, 0, n, d)
        o = oracle.Hnsw(d, metric=metric, m=m, m0=m0, efc=efc,
                        ml=1.0 / math.log(m), seed=seed)
        for r in rows:
            o.insert(r)
        offsets, edges = o.l0_csr()
        queries = oracle.gen_f32(qseed, 0, nq, d)
        out = {
            "d": d, "n": n, "metric": metric, "m": m, "m0": m0, "efc": efc,
            "seed": seed, "data_seed": dseed, "query_seed": qseed, "k": k,
            "num_layers": o.num_layers(), "entry_point": o.entry_point(),
            "l0_offsets": offsets, "l0_edges": edges,
        }
        for ef in efs:
            all_ids, all_dists, all_n = [], [], []
            for q in queries:
                ids, dists = builder_sort(*o.search(q, k, ef))
                nn = len(ids)
                all_n.append(nn)
                all_ids.append(np.pad(ids, (0, k - nn),
                                      constant_values=np.iinfo(np.uint64).max))
                all_dists.append(np.pad(dists, (0, k - nn),
                                        constant_values=np.inf))
            out[f"ids_ef{ef}"] = np.stack(all_ids)
            out[f"dists_ef{ef}"] = np.stack(all_dists)
            out[f"n_ef{ef}"] = np.array(all_n, dtype=np.uint32)
        path = os.path.join(OUT_DIR, f"hnsw_fix_{name}.npz")
        np.savez_compressed(path, **out)
        print(f"{name}: n={n} d={d} {metric} built in "
              f"{time.perf_counter()-t0:.1f}s -> {path} "
              f"({os.path.getsize(path)//1024} KiB)")


if __name__ == "__main__":
    main()
