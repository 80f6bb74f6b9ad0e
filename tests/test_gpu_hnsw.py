"""GPU HNSW search parity: product knn (host graph + GPU layer-0 gather)
vs the oracle search on the identical (sequentially-built) graph, plus the
reference's recall bars on the golden datasets."""
import gzip
import json
import math
import os

import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu

GOLDEN_DIR = os.path.join(os.path.dirname(__file__), "golden")


def load_golden(name, limit):
    rows = []
    with gzip.open(os.path.join(GOLDEN_DIR, name), "rt") as f:
        for i, line in enumerate(f):
            if i >= limit:
                break
            rows.append(json.loads(line))
    return np.array(rows, dtype=np.float32)


@pytest.fixture(scope="module")
def ctx():
    import surrealdb_amd
    c = surrealdb_amd.Context()
    yield c
    c.close()


def builder_sort(ids, dists):
    """Apply the KnnResultBuilder final ordering (dist total_cmp, id) to an
    oracle result (which is in (dist, FIFO) order)."""
    from surrealdb_amd.shard import total_key
    order = np.lexsort((ids, total_key(dists)))
    return ids[order], dists[order]


def test_knn_matches_oracle_exactly(ctx):
    d, n = 768, 20_000
    rows = oracle.gen_f32(0x5DB1, 0, n, d)
    h = ctx.hnsw_create(d, metric="euclidean", m=12, efc=150, seed=0x5DB1)
    o = oracle.Hnsw(d, metric="euclidean", m=12, m0=24, efc=150,
                    ml=1.0 / math.log(12.0), seed=0x5DB1)
    h.insert_batch(rows, nthreads=1)  # sequential => graph == oracle graph
    for r in rows[:0]:
        pass
    for r in rows:
        o.insert(r)
    po, pe = h.l0_csr()
    oo, oe = o.l0_csr()
    assert np.array_equal(po, oo) and np.array_equal(pe, oe)
    h.finalize(20)
    queries = oracle.gen_f32(0xBEEF, 0, 20, d)
    for q in queries:
        gids, gdists = h.knn_search(q, 10, 64)
        oids, odists = o.search(q, 10, 64)
        oids, odists = builder_sort(oids, odists)
        assert np.array_equal(gids, oids)
        assert np.array_equal(gdists, odists)
    h.destroy()
    ctx.drop_table(20)


def test_cosine_knn_matches_oracle(ctx):
    d, n = 128, 5000
    rows = oracle.gen_f32(0x77, 0, n, d)
    h = ctx.hnsw_create(d, metric="cosine", m=8, efc=100, seed=0x11)
    o = oracle.Hnsw(d, metric="cosine", m=8, m0=16, efc=100,
                    ml=1.0 / math.log(8.0), seed=0x11)
    for r in rows:
        h.insert(r)
        o.insert(r)
    h.finalize(21)
    for q in oracle.gen_f32(0x88, 0, 10, d):
        gids, gdists = h.knn_search(q, 10, 40)
        oids, odists = builder_sort(*o.search(q, 10, 40))
        assert np.array_equal(gids, oids)
        assert np.array_equal(gdists, odists)
    h.destroy()
    ctx.drop_table(21)


def test_golden_recall_bars(ctx):
    """hnsw/mod.rs:1144-1156 restated on the product path: recall >= 0.98 @
    efs=10 and == 1.0 @ efs=40 with exact set equality vs brute force."""
    ingest = load_golden("hnsw-random-9000-20-euclidean.gz", 1000)
    queries = load_golden("hnsw-random-5000-20-euclidean.gz", 300)
    h = ctx.hnsw_create(20, metric="euclidean", m=8, efc=100,
                        ml=1.0 / math.log(8.0), seed=0x5DB1)
    h.insert_batch(ingest, nthreads=1)
    h.finalize(22)
    k = 10
    total10 = 0.0
    for q in queries:
        ids40, _ = h.knn_search(q, k, 40)
        bids, _ = oracle.topk_f32("euclidean", ingest, q, k)
        assert set(ids40.tolist()) == set(bids.tolist())
        ids10, _ = h.knn_search(q, k, 10)
        total10 += len(set(ids10.tolist()) & set(bids.tolist())) / k
    assert total10 / len(queries) >= 0.98
    h.destroy()
    ctx.drop_table(22)


def test_parallel_build_recall(ctx):
    """Bench-mode parallel build must still meet the efs=40 recall bar."""
    ingest = load_golden("hnsw-random-9000-20-euclidean.gz", 2000)
    queries = load_golden("hnsw-random-5000-20-euclidean.gz", 100)
    h = ctx.hnsw_create(20, metric="euclidean", m=8, efc=100,
                        ml=1.0 / math.log(8.0), seed=0x5DB1)
    h.insert_batch(ingest, nthreads=8)
    h.finalize(23)
    k, total = 10, 0.0
    for q in queries:
        ids, _ = h.knn_search(q, k, 40)
        bids, _ = oracle.topk_f32("euclidean", ingest, q, k)
        total += len(set(ids.tolist()) & set(bids.tolist())) / k
    # parallel build is a bench-mode extension: the graph is
    # order-nondeterministic, so the sequential ==1.0 bar relaxes slightly
    # (the reference's own bar at this ef is 1.0 for sequential builds)
    assert total / len(queries) >= 0.99, total / len(queries)
    h.destroy()
    ctx.drop_table(23)


def test_persistent_kernel_matches_perhop_and_oracle(ctx):
    """The persistent in-kernel search must return EXACTLY what the per-hop
    gather path and the oracle return (same graph, same queue semantics,
    bit-exact distances)."""
    d, n = 768, 30_000
    rows = oracle.gen_f32(0x5DB1, 0, n, d)
    h = ctx.hnsw_create(d, metric="cosine", m=16, m0=32, efc=150, seed=0x9)
    o = oracle.Hnsw(d, metric="cosine", m=16, m0=32, efc=150,
                    ml=1.0 / math.log(16.0), seed=0x9)
    h.insert_batch(rows, nthreads=1)
    for r in rows:
        o.insert(r)
    h.finalize(24)
    Q = oracle.gen_f32(0xBEEF, 0, 32, d)
    bids, bdists, bns = h.knn_search_batch(Q, 10, 64)
    for j in range(32):
        pids, pdists = h.knn_search(Q[j], 10, 64)
        assert np.array_equal(bids[j][:bns[j]], pids), f"q{j} vs per-hop"
        assert np.array_equal(bdists[j][:bns[j]], pdists), f"q{j} dists"
        oids, odists = builder_sort(*o.search(Q[j], 10, 64))
        assert np.array_equal(bids[j][:bns[j]], oids), f"q{j} vs oracle"
        assert np.array_equal(bdists[j][:bns[j]], odists)
    h.destroy()
    ctx.drop_table(24)


def test_persistent_kernel_euclidean_golden(ctx):
    ingest = load_golden("hnsw-random-9000-20-euclidean.gz", 1500)
    queries = load_golden("hnsw-random-5000-20-euclidean.gz", 200)
    h = ctx.hnsw_create(20, metric="euclidean", m=8, efc=100,
                        ml=1.0 / math.log(8.0), seed=0x5DB1)
    h.insert_batch(ingest, nthreads=1)
    h.finalize(25)
    bids, bdists, bns = h.knn_search_batch(queries, 10, 40)
    for j, q in enumerate(queries):
        bf, _ = oracle.topk_f32("euclidean", ingest, q, 10)
        assert set(bids[j][:bns[j]].tolist()) == set(bf.tolist()), f"q{j}"
        pids, pdists = h.knn_search(q, 10, 40)
        assert np.array_equal(bids[j][:bns[j]], pids)
        assert np.array_equal(bdists[j][:bns[j]], pdists)
    h.destroy()
    ctx.drop_table(25)
