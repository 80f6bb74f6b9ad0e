"""Multi-GPU row-sharding of a corpus and the per-shard top-K merge.

The reference never shards a single KNN search (no compute
parallelism exists in surrealdb); this is new MI355X design constrained only
by result parity: rows split contiguously across ranks, each rank runs the
local scan, then ONE collective — an all-gather of the per-shard top-K
(K*(8B dist + 8B id) per rank, latency-bound over xGMI) — and a final
merge with the reference tie-break order (dist total_cmp asc, id asc;
knn.rs:128-160, :363).

Pure-python/numpy so it is testable on CPU with the gloo backend
(tests/test_multigpu_gloo.py); bench.py uses it over RCCL.
"""
import numpy as np


def total_key(dists):
    """f64 total_cmp key (knn.rs:128-160) as monotone uint64, vectorised."""
    bits = np.ascontiguousarray(dists, dtype=np.float64).view(np.uint64)
    neg = bits >> np.uint64(63) != 0
    out = np.where(neg, ~bits, bits | np.uint64(0x8000000000000000))
    return out


def shard_range(n_total, rank, world):
    """Contiguous row range [begin, end) of `rank` out of `world` shards."""
    per = (n_total + world - 1) // world
    begin = min(rank * per, n_total)
    end = min(begin + per, n_total)
    return begin, end


def merge_topk(ids_list, dists_list, k):
    """Merge per-shard (ids, dists) into the global top-k.

    Order: ascending (total_cmp(dist), id) — the reference's
    BTreeSet<(FloatKey, VectorId)> ordering (knn.rs:363).
    """
    ids = np.concatenate([np.asarray(i, dtype=np.uint64) for i in ids_list])
    dists = np.concatenate([np.asarray(d, dtype=np.float64) for d in dists_list])
    keys = total_key(dists)
    order = np.lexsort((ids, keys))
    take = order[:k]
    return ids[take], dists[take]
