"""GPU parity tests — the HIP product path vs the CPU oracle, through the
C-ABI. These are THE parity gates (tier contract): bit-exact ids/ranks,
bit-exact distances for identical per-row accumulation, scores <= 1e-5 rel
vs the Number-path (f64) reference semantics.

All tests are @pytest.mark.gpu (need a real MI355X; run via gpurun)."""
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


def stage_host(ctx, table, corpus, metric, ids=None):
    ctx.stage_corpus(table, corpus, ids=ids, metric=metric)


@pytest.mark.parametrize("metric", ["cosine", "euclidean"])
@pytest.mark.parametrize("n,d", [(1000, 128), (50_000, 768), (3000, 20)])
def test_all_distances_bitexact(ctx, metric, n, d):
    """Every distance the scan computes must equal the oracle's BITWISE
    (same restated per-row accumulation chain on both sides)."""
    corpus = oracle.gen_f32(0x5DB1, 0, n, d)
    q = oracle.gen_f32(0xBEEF, 0, 1, d)[0]
    stage_host(ctx, 1, corpus, metric)
    gpu = ctx.all_distances(1, q)
    cpu = np.array([oracle.dist_f32(metric, q, corpus[i]) for i in range(n)])
    assert gpu.shape == cpu.shape
    neq = np.nonzero(gpu != cpu)[0]
    assert neq.size == 0, (
        f"{neq.size}/{n} distances differ; first {neq[:5]}: "
        f"gpu={gpu[neq[:5]]} cpu={cpu[neq[:5]]}")
    ctx.drop_table(1)


@pytest.mark.parametrize("metric", ["cosine", "euclidean"])
@pytest.mark.parametrize("n,d,k", [(999, 128, 10), (50_000, 768, 10),
                                   (200_000, 768, 64), (5, 16, 10)])
def test_knn_bruteforce_exact(ctx, metric, n, d, k):
    corpus = oracle.gen_f32(0x5DB1, 0, n, d)
    q = oracle.gen_f32(0xBEEF, 1, 1, d)[0]
    stage_host(ctx, 2, corpus, metric)
    gids, gdists = ctx.knn_bruteforce(2, q, k)
    oids, odists = oracle.topk_f32(metric, corpus, q, k)
    assert np.array_equal(gids, oids), f"{metric} n={n}: id/rank mismatch"
    assert np.array_equal(gdists, odists), f"{metric} n={n}: dist bits differ"
    ctx.drop_table(2)


def test_knn_synthetic_staging_matches_host_staging(ctx):
    """Device-side generation (stage_synthetic) must equal host-generated +
    uploaded corpus: same generator contract on both sides."""
    n, d = 30_000, 768
    corpus = oracle.gen_f32(0x5DB1, 0, n, d)
    q = oracle.gen_f32(0xBEEF, 2, 1, d)[0]
    stage_host(ctx, 3, corpus, "cosine")
    ctx.stage_synthetic(4, n, d, metric="cosine", seed=0x5DB1)
    a = ctx.knn_bruteforce(3, q, 10)
    b = ctx.knn_bruteforce(4, q, 10)
    assert np.array_equal(a[0], b[0])
    assert np.array_equal(a[1], b[1])
    ctx.drop_table(3)
    ctx.drop_table(4)


def test_duplicates_and_ties(ctx):
    base = oracle.gen_f32(0x77, 0, 500, 64)
    corpus = np.concatenate([base, base[:20]])  # exact duplicates
    q = base[7].copy()
    stage_host(ctx, 5, corpus, "euclidean")
    gids, gdists = ctx.knn_bruteforce(5, q, 8)
    oids, odists = oracle.topk_f32("euclidean", corpus, q, 8)
    assert np.array_equal(gids, oids)
    assert gids[0] == 7 and gids[1] == 507  # dup tie -> ascending id
    assert gdists[0] == 0.0 and gdists[1] == 0.0
    ctx.drop_table(5)


def test_explicit_ids_mapping(ctx):
    corpus = oracle.gen_f32(0x88, 0, 1000, 32)
    ids = (np.arange(1000, dtype=np.uint64) * 7 + 100)
    q = oracle.gen_f32(0x99, 0, 1, 32)[0]
    stage_host(ctx, 6, corpus, "cosine", ids=ids)
    gids, gdists = ctx.knn_bruteforce(6, q, 5)
    oids, odists = oracle.topk_f32("cosine", corpus, q, 5)
    assert np.array_equal(gids, ids[oids.astype(np.int64)])
    assert np.array_equal(gdists, odists)
    ctx.drop_table(6)


def test_unsorted_ids_rejected(ctx):
    import surrealdb_amd
    corpus = oracle.gen_f32(0x11, 0, 10, 16)
    bad = np.array([5, 3, 8, 1, 2, 9, 0, 4, 6, 7], dtype=np.uint64)
    with pytest.raises(surrealdb_amd.SdbvError):
        ctx.stage_corpus(7, corpus, ids=bad, metric="cosine")


def test_gather_distance_parity(ctx):
    """HNSW frontier expansion primitive: gathered rows' distances bitwise
    equal the oracle's."""
    corpus = oracle.gen_f32(0x22, 0, 5000, 768)
    q = oracle.gen_f32(0x33, 0, 1, 768)[0]
    stage_host(ctx, 8, corpus, "euclidean")
    rows = np.array([0, 1, 17, 999, 4999, 2500, 3], dtype=np.uint32)
    got = ctx.gather_distance(8, rows, q)
    want = np.array([oracle.dist_f32("euclidean", q, corpus[r]) for r in rows])
    assert np.array_equal(got, want)
    ctx.drop_table(8)


def test_number_path_tolerance(ctx):
    """Drop-in score bar vs the reference's f64 Number-path brute force
    (the KnnTopK operator semantics): ranks equal, scores <= 1e-5 rel."""
    n, d, k = 20_000, 768, 10
    corpus = oracle.gen_f32(0x5DB1, 0, n, d)
    q = oracle.gen_f32(0xBEEF, 3, 1, d)[0]
    stage_host(ctx, 9, corpus, "cosine")
    gids, gdists = ctx.knn_bruteforce(9, q, k)
    nids, ndists = oracle.topk_number("cosine", corpus.astype(np.float64),
                                      q.astype(np.float64), k)
    assert np.array_equal(gids, nids)
    assert np.allclose(gdists, ndists, rtol=1e-5)
    ctx.drop_table(9)


def test_smoke_stats(ctx):
    ctx.stage_synthetic(10, 100_000, 768, metric="cosine")
    ctx.knn_bruteforce(10, oracle.gen_f32(1, 0, 1, 768)[0], 10)
    s = ctx.stats()
    assert s["last_scan_kernel_ms"] > 0
    assert s["last_rows_scanned"] == 100_000
    ctx.drop_table(10)


@pytest.mark.parametrize("metric", ["cosine", "euclidean"])
def test_knn_large_k_matches_oracle(ctx, metric):
    """k beyond the scan kernel's top-K window (MAX_K=64): the
    all-distances + host-selection path must keep the exact ordering
    contract (bit-exact ids/ranks/distances vs the oracle)."""
    n, d = 20_000, 128
    corpus = oracle.gen_f32(0x5DB1, 0, n, d)
    ctx.stage_corpus(17, corpus, metric=metric)
    q = oracle.gen_f32(0xBEEF, 0, 1, d)[0]
    for k in (65, 100, 500):
        gids, gdists = ctx.knn_bruteforce(17, q, k)
        oids, odists = oracle.topk_f32(metric, corpus, q, k)
        assert np.array_equal(gids, oids), f"k={k} ids"
        assert np.array_equal(gdists, odists), f"k={k} dist bits"
    ctx.drop_table(17)


# --- the six non-headline metrics (SURVEY §8 a2 closure): all-distances
# route; chains restate vector.rs:206-451 op-for-op ---

EXTRA_METRICS = ["manhattan", "chebyshev", "hamming", "pearson"]


def quantized(seed, n, d):
    """Coarsely quantized corpus: forces element collisions so hamming /
    jaccard see real overlaps (random f32 would make every element
    distinct and every distance equal)."""
    return np.round(oracle.gen_f32(seed, 0, n, d) * 0.25).astype(np.float32)


@pytest.mark.parametrize("metric", EXTRA_METRICS)
def test_extra_metric_all_distances_bitexact(ctx, metric):
    n, d = 4000, 64
    corpus = quantized(0x5DB1, n, d)
    q = quantized(0xBEEF, 1, d)[0]
    stage_host(ctx, 21, corpus, metric)
    gpu = ctx.all_distances(21, q)
    cpu = np.array([oracle.dist_f32(metric, q, corpus[i]) for i in range(n)])
    neq = np.nonzero(gpu != cpu)[0]
    assert neq.size == 0, (
        f"{metric}: {neq.size}/{n} differ; first {neq[:5]}: "
        f"gpu={gpu[neq[:5]]} cpu={cpu[neq[:5]]}")
    ctx.drop_table(21)


def test_jaccard_all_distances_bitexact(ctx):
    n, d = 4000, 64
    corpus = quantized(0xA5, n, d)
    # inject rows with heavy element overlap with q and in-row duplicates
    q = quantized(0xBEEF, 1, d)[0]
    corpus[7, :32] = q[:32]
    corpus[11, :] = q
    corpus[13, :] = corpus[13, 0]  # all-duplicate row
    stage_host(ctx, 22, corpus, "jaccard")
    gpu = ctx.all_distances(22, q)
    cpu = np.array([oracle.dist_f32("jaccard", q, corpus[i])
                    for i in range(n)])
    neq = np.nonzero(gpu != cpu)[0]
    assert neq.size == 0, (
        f"jaccard: {neq.size}/{n} differ; first {neq[:5]}: "
        f"gpu={gpu[neq[:5]]} cpu={cpu[neq[:5]]}")
    ctx.drop_table(22)


def test_minkowski_all_distances(ctx):
    """Minkowski goes through f64 pow on both sides; device libm pow may
    differ from glibc in the last ulp, so the distance bar is 1e-12 rel
    (the north_star's score bar is 1e-5) with ranks checked separately in
    test_extra_metric_knn."""
    n, d = 4000, 64
    corpus = oracle.gen_f32(0x5DB1, 0, n, d)
    q = oracle.gen_f32(0xBEEF, 1, 1, d)[0]
    stage_host(ctx, 23, corpus, "minkowski")
    import surrealdb_amd
    surrealdb_amd.lib().sdbv_table_set_order(ctx._ptr, 23, 3.0)
    gpu = ctx.all_distances(23, q)
    cpu = np.array([oracle.dist_f32("minkowski", q, corpus[i], order=3.0)
                    for i in range(n)])
    assert np.allclose(gpu, cpu, rtol=1e-12, atol=0)
    ctx.drop_table(23)


@pytest.mark.parametrize("metric,order", [("manhattan", 0.0),
                                          ("chebyshev", 0.0),
                                          ("hamming", 0.0),
                                          ("pearson", 0.0),
                                          ("jaccard", 0.0),
                                          ("minkowski", 3.0)])
def test_extra_metric_knn(ctx, metric, order):
    """Top-k through sdbv_knn_bruteforce for every metric: ids/ranks equal
    the oracle's; distances bit-exact (minkowski: 1e-12 rel, see above).
    pearson's similarity-as-distance quirk (most-negative correlation
    first) and jaccard's F32 |I|/|U| asymmetry ride through as-is."""
    n, d, k = 30_000, 64, 10
    corpus = quantized(0x5DB1, n, d) if metric in ("hamming", "jaccard") \
        else oracle.gen_f32(0x5DB1, 0, n, d)
    q = (quantized(0xBEEF, 1, d) if metric in ("hamming", "jaccard")
         else oracle.gen_f32(0xBEEF, 1, 1, d))[0]
    ctx.stage_corpus(24, corpus, metric=metric,
                     order=order if order else None)
    gids, gdists = ctx.knn_bruteforce(24, q, k)
    oids, odists = oracle.topk_f32(metric, corpus, q, k, order=order)
    assert np.array_equal(gids, oids), f"{metric}: ids/ranks"
    if metric == "minkowski":
        assert np.allclose(gdists, odists, rtol=1e-12, atol=0)
    else:
        assert np.array_equal(gdists, odists), f"{metric}: dist bits"
    ctx.drop_table(24)


def test_extra_metric_batch_route(ctx):
    """sdbv_knn_batch for a non-GEMM metric routes per query through the
    all-distances path — results equal per-query bruteforce."""
    n, d, k, b = 8000, 64, 10, 7
    corpus = oracle.gen_f32(0x5DB1, 0, n, d)
    ctx.stage_corpus(25, corpus, metric="manhattan")
    Q = oracle.gen_f32(0xBEEF, 0, b, d)
    bids, bdists = ctx.knn_batch(25, Q, k)
    for j in range(b):
        sids, sdists = ctx.knn_bruteforce(25, Q[j], k)
        assert np.array_equal(bids[j], sids)
        assert np.array_equal(bdists[j], sdists)
    ctx.drop_table(25)

