"""GPU parity for the batched-query (MFMA GEMM) path vs the oracle.

The batch path selects with a monotone f32 key but re-computes survivors with
the exact restated chain, so final ids/ranks/distance-bits must equal the
single-query path and the oracle exactly."""
import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctx():
    import surrealdb_amd
    c = surrealdb_amd.Context()
    yield c
    c.close()


@pytest.mark.parametrize("metric", ["cosine", "euclidean"])
@pytest.mark.parametrize("n,d,b,k", [(50_000, 768, 32, 10), (4096, 128, 7, 5),
                                     (300_000, 768, 64, 48)])
def test_batch_matches_oracle(ctx, metric, n, d, b, k):
    corpus = oracle.gen_f32(0x5DB1, 0, n, d)
    Q = oracle.gen_f32(0xBEEF, 0, b, d)
    ctx.stage_corpus(11, corpus, metric=metric)
    ids, dists = ctx.knn_batch(11, Q, k)
    for j in range(b):
        oids, odists = oracle.topk_f32(metric, corpus, Q[j], k)
        assert np.array_equal(ids[j], oids), f"{metric} q{j}: ids differ"
        assert np.array_equal(dists[j], odists), f"{metric} q{j}: dist bits"
    ctx.drop_table(11)


def test_batch_equals_single_query_path(ctx):
    n, d, b, k = 100_000, 768, 16, 10
    ctx.stage_synthetic(12, n, d, metric="cosine", seed=0x5DB1)
    Q = oracle.gen_f32(0xBEEF, 0, b, d)
    bids, bdists = ctx.knn_batch(12, Q, k)
    for j in range(b):
        sids, sdists = ctx.knn_bruteforce(12, Q[j], k)
        assert np.array_equal(bids[j], sids)
        assert np.array_equal(bdists[j], sdists)
    ctx.drop_table(12)


def test_batch_n_smaller_than_k(ctx):
    corpus = oracle.gen_f32(0x44, 0, 6, 16)
    Q = oracle.gen_f32(0x55, 0, 3, 16)
    ctx.stage_corpus(13, corpus, metric="cosine")
    ids, dists = ctx.knn_batch(13, Q, 10)
    for j in range(3):
        oids, odists = oracle.topk_f32("cosine", corpus, Q[j], 10)
        assert np.array_equal(ids[j][:6], oids)
        assert (ids[j][6:] == np.iinfo(np.uint64).max).all()
        assert np.isinf(dists[j][6:]).all()
    ctx.drop_table(13)


def test_batch_duplicates(ctx):
    base = oracle.gen_f32(0x66, 0, 1000, 64)
    corpus = np.concatenate([base, base[:50]])
    Q = base[[3, 30, 49]].copy()
    ctx.stage_corpus(14, corpus, metric="euclidean")
    ids, dists = ctx.knn_batch(14, Q, 4)
    for j, orig in enumerate([3, 30, 49]):
        assert ids[j][0] == orig and ids[j][1] == 1000 + orig
        assert dists[j][0] == 0.0 and dists[j][1] == 0.0
    ctx.drop_table(14)


@pytest.mark.parametrize("metric", ["cosine", "euclidean"])
def test_fused_mfma_matches_oracle_and_legacy(ctx, metric):
    """The hand-written fused MFMA kernel (b % 128 == 0, n > 65536 engages
    it) must produce exactly the oracle's results AND exactly the legacy
    rocBLAS + k_batch_topk path's results."""
    import os
    n, d, b, k = 200_000, 768, 128, 10
    corpus = oracle.gen_f32(0x5DB1, 0, n, d)
    Q = oracle.gen_f32(0xBEEF, 0, b, d)
    ctx.stage_corpus(13, corpus, metric=metric)
    ids_f, dists_f = ctx.knn_batch(13, Q, k)       # fused (default)
    os.environ["SDBV_BATCH_LEGACY"] = "1"
    try:
        ids_l, dists_l = ctx.knn_batch(13, Q, k)   # legacy
    finally:
        del os.environ["SDBV_BATCH_LEGACY"]
    assert np.array_equal(ids_f, ids_l), f"{metric}: fused != legacy ids"
    assert np.array_equal(dists_f, dists_l), f"{metric}: fused != legacy"
    for j in range(0, b, 17):
        oids, odists = oracle.topk_f32(metric, corpus, Q[j], k)
        assert np.array_equal(ids_f[j], oids), f"{metric} q{j}: ids"
        assert np.array_equal(dists_f[j], odists), f"{metric} q{j}: bits"
    ctx.drop_table(13)


def test_fused_mfma_duplicate_ties(ctx):
    """Exact duplicates of the query inside the fused row range tie at
    distance 0 and must surface in ascending id order."""
    n, d, b, k = 140_000, 256, 128, 8
    corpus = oracle.gen_f32(0x99, 0, n, d)
    Q = oracle.gen_f32(0xAB, 0, b, d)
    corpus[70_000] = Q[3]  # inside the fused range (> 65536)
    corpus[100_001] = Q[3]
    corpus[12_345] = Q[3]  # inside the bootstrap range
    ctx.stage_corpus(14, corpus, metric="euclidean")
    ids, dists = ctx.knn_batch(14, Q, k)
    assert list(ids[3][:3]) == [12_345, 70_000, 100_001]
    assert dists[3][0] == 0.0 and dists[3][2] == 0.0
    oids, odists = oracle.topk_f32("euclidean", corpus, Q[3], k)
    assert np.array_equal(ids[3], oids)
    assert np.array_equal(dists[3], odists)
    ctx.drop_table(14)
