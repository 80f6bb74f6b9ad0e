"""Top-K semantics: the oracle's restated KnnTopK/KnnResultBuilder ordering.

Anchors:
 - knn_result_builder_test (knn.rs:645-668): tie-heavy fixture.
 - KnnTopK eviction rules (knn_topk.rs:216-227): strict-closer insert,
   insertion-order tie-break.
"""
import numpy as np

import oracle
from surrealdb_amd.shard import merge_topk, shard_range, total_key


def brute_sorted(metric, corpus, q, k):
    """Independent check: full argsort by (total_cmp(dist), row)."""
    dists = np.array([
        oracle.dist_f32(metric, q, corpus[i]) for i in range(corpus.shape[0])
    ])
    order = np.lexsort((np.arange(corpus.shape[0]), total_key(dists)))[:k]
    return order.astype(np.uint64), dists[order]


def test_result_builder_tie_fixture():
    """knn.rs:645-668 restated: k=7 over distances with heavy 0.2 ties ->
    ascending (dist, id): (0,5),(0.2,0),(0.2,1),(0.2,2),(0.2,3),(0.2,6),(0.2,8)."""
    # corpus of 9 1-d rows whose euclidean distance to q=0 is the fixture's
    d = {5: 0.0, 0: 0.2, 1: 0.2, 2: 0.2, 3: 0.2, 6: 0.2, 8: 0.2, 4: 0.9, 7: 0.9}
    corpus = np.zeros((9, 4), dtype=np.float32)
    for row, dist in d.items():
        corpus[row, 0] = dist
    q = np.zeros(4, dtype=np.float32)
    ids, dists = oracle.topk_f32("euclidean", corpus, q, 7)
    assert list(ids) == [5, 0, 1, 2, 3, 6, 8]
    assert dists[0] == 0.0
    assert np.allclose(dists[1:], 0.2, atol=1e-7)


def test_topk_matches_full_sort():
    rng = np.random.default_rng(11)
    corpus = rng.uniform(-20, 20, (500, 32)).astype(np.float32)
    q = rng.uniform(-20, 20, 32).astype(np.float32)
    for metric in ("cosine", "euclidean"):
        ids, dists = oracle.topk_f32(metric, corpus, q, 10)
        bids, bdists = brute_sorted(metric, corpus, q, 10)
        assert np.array_equal(ids, bids)
        assert np.array_equal(dists, bdists)


def test_topk_duplicates_tiebreak_by_id():
    rng = np.random.default_rng(13)
    base = rng.uniform(-20, 20, (50, 16)).astype(np.float32)
    corpus = np.concatenate([base, base[:5]])  # rows 50..54 duplicate 0..4
    q = base[2].copy()
    ids, dists = oracle.topk_f32("euclidean", corpus, q, 4)
    # row 2 and its duplicate row 52 tie at 0 -> smaller id first
    assert ids[0] == 2 and ids[1] == 52
    assert dists[0] == 0.0 and dists[1] == 0.0


def test_topk_n_smaller_than_k():
    rng = np.random.default_rng(17)
    corpus = rng.uniform(-20, 20, (3, 8)).astype(np.float32)
    q = rng.uniform(-20, 20, 8).astype(np.float32)
    ids, dists = oracle.topk_f32("cosine", corpus, q, 10)
    assert len(ids) == 3
    assert (np.diff(total_key(dists)) >= 0).all()


def test_number_path_vs_f32_rank_parity():
    """The product stages f32 (north_star contract); the reference brute-force
    computes on f64 Numbers. On the seeded synthetic corpus ranks must agree
    and scores must be within 1e-5 relative (the parity bar)."""
    corpus32 = oracle.gen_f32(0x5DB1, 0, 2000, 128)
    q32 = oracle.gen_f32(0xBEEF, 0, 1, 128)[0]
    ids32, d32 = oracle.topk_f32("cosine", corpus32, q32, 10)
    ids64, d64 = oracle.topk_number("cosine", corpus32.astype(np.float64),
                                    q32.astype(np.float64), 10)
    assert np.array_equal(ids32, ids64)
    assert np.allclose(d32, d64, rtol=1e-5)


def test_mt_equals_st():
    corpus = oracle.gen_f32(1, 0, 5000, 64)
    q = oracle.gen_f32(2, 0, 1, 64)[0]
    for metric in ("cosine", "euclidean"):
        i1, d1 = oracle.topk_f32(metric, corpus, q, 13)
        i2, d2, used = oracle.topk_f32_mt(metric, corpus, q, 13, nthreads=4)
        assert np.array_equal(i1, i2)
        assert np.array_equal(d1, d2)
        assert used == 4


def test_shard_merge_equals_global():
    """The multi-GPU merge contract on CPU: per-shard top-k + merge ==
    global top-k (ids offset per shard)."""
    corpus = oracle.gen_f32(0x5DB1, 0, 4000, 64)
    q = oracle.gen_f32(3, 0, 1, 64)[0]
    k = 10
    gids, gdists = oracle.topk_f32("cosine", corpus, q, k)
    ids_list, dists_list = [], []
    world = 4
    for rank in range(world):
        b, e = shard_range(4000, rank, world)
        ids, dists = oracle.topk_f32("cosine", corpus[b:e], q, k)
        ids_list.append(ids + b)
        dists_list.append(dists)
    mids, mdists = merge_topk(ids_list, dists_list, k)
    assert np.array_equal(mids, gids)
    assert np.array_equal(mdists, gdists)


def test_negative_distance_ordering():
    """Cosine distance can round slightly below 0 for near-duplicates; the
    total_cmp key must order negatives correctly."""
    d = np.array([-1e-16, 0.0, 1e-16, -0.0])
    k = total_key(d)
    order = np.argsort(k)
    assert list(order) == [0, 3, 1, 2]  # -1e-16 < -0.0 < 0.0 < 1e-16


def test_integer_vectors_number_path_exact():
    """a5: the Number path keeps Int x Int in i64 (val/number.rs:926-1009).
    For integer-valued vectors within f64-exact range the restated f64
    accumulation produces the identical values (every partial sum is an
    integer < 2^53, exactly representable), so the oracle's Number path is
    exact for Int corpora too — asserted against exact Python ints — and
    the f32-staged scan agrees in ranks for ints within f32-exact range."""
    rng = np.random.default_rng(11)
    corpus_i = rng.integers(-1000, 1000, size=(500, 64))
    q_i = rng.integers(-1000, 1000, size=64)
    ids, dists = oracle.topk_number("euclidean",
                                    corpus_i.astype(np.float64),
                                    q_i.astype(np.float64), 5)
    import math as _m
    exact = sorted(
        ( _m.sqrt(int(sum((int(a) - int(b)) ** 2
                          for a, b in zip(q_i, row)))), i)
        for i, row in enumerate(corpus_i))
    for rank, (ed, ei) in enumerate(exact[:5]):
        assert ids[rank] == ei
        assert dists[rank] == ed  # bit-exact: f64 sqrt of the exact i64 sum
    ids32, d32 = oracle.topk_f32("euclidean",
                                 corpus_i.astype(np.float32),
                                 q_i.astype(np.float32), 5)
    assert np.array_equal(ids, ids32)  # ints < 2^24: f32 staging exact
