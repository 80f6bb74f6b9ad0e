"""GPU HNSW search parity: product knn (host graph + GPU layer-0 gather and
the persistent in-kernel search) vs committed ORACLE FIXTURES, plus the
reference's recall bars on its golden datasets.

The oracle graph builds at 768-dim are slow, so the oracle
side of each configuration was built ONCE by tests/golden/make_hnsw_fixtures.py
(committed script) and its expected graph + builder-sorted search results are
committed as tests/golden/hnsw_fix_*.npz. Here only the product build (the
C++ host path under test) runs, and graph + GPU search results must match the
fixture bit-exactly."""
import gzip
import json
import math
import os

import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu

GOLDEN_DIR = os.path.join(os.path.dirname(__file__), "golden")

FIXTURE_NAMES = ["seq768_cos", "seq768_euc", "seq128_cos",
                 "seq768_cos_16k"]


def load_golden(name, limit):
    rows = []
    with gzip.open(os.path.join(GOLDEN_DIR, name), "rt") as f:
        for i, line in enumerate(f):
            if i >= limit:
                break
            rows.append(json.loads(line))
    return np.array(rows, dtype=np.float32)


def load_fix(name):
    return np.load(os.path.join(GOLDEN_DIR, f"hnsw_fix_{name}.npz"))


@pytest.fixture(scope="module")
def ctx():
    import surrealdb_amd
    c = surrealdb_amd.Context()
    yield c
    c.close()


@pytest.fixture(scope="module")
def built(ctx):
    """Sequential product builds for every fixture config, finalized once and
    shared by the graph/search tests below."""
    idxs = {}
    table = 30
    for name in FIXTURE_NAMES:
        fx = load_fix(name)
        d, n, m = int(fx["d"]), int(fx["n"]), int(fx["m"])
        rows = oracle.gen_f32(int(fx["data_seed"]), 0, n, d)
        h = ctx.hnsw_create(d, metric=str(fx["metric"]), m=m,
                            m0=int(fx["m0"]), efc=int(fx["efc"]),
                            ml=1.0 / math.log(m), seed=int(fx["seed"]))
        h.insert_batch(rows, nthreads=1)  # sequential => deterministic graph
        h.finalize(table)
        idxs[name] = (h, fx)
        table += 1
    yield idxs
    for h, _ in idxs.values():
        h.destroy()


def fixture_ef(fx):
    return int([k for k in fx.files if k.startswith("ids_ef")][0][6:])


@pytest.mark.parametrize("name", FIXTURE_NAMES)
def test_seq_build_graph_matches_oracle(built, name):
    """The product C++ sequential build must produce the bit-identical graph
    to the oracle build of the same configuration (independent restatements
    of hnsw/mod.rs + layer.rs + heuristic.rs with the same level-RNG)."""
    h, fx = built[name]
    assert h.n() == int(fx["n"])
    assert h.num_layers() == int(fx["num_layers"])
    po, pe = h.l0_csr()
    assert np.array_equal(po, fx["l0_offsets"]), f"{name}: CSR offsets differ"
    assert np.array_equal(pe, fx["l0_edges"]), f"{name}: CSR edges differ"


@pytest.mark.parametrize("name", FIXTURE_NAMES)
def test_perhop_search_matches_oracle(built, name):
    """Per-hop GPU search (host queue + sdbv_gather_distance expansion) must
    return exactly the oracle's builder-sorted results (ids, ranks, and
    distance BITS)."""
    h, fx = built[name]
    ef = fixture_ef(fx)
    k = int(fx["k"])
    queries = oracle.gen_f32(int(fx["query_seed"]), 0,
                             fx[f"ids_ef{ef}"].shape[0], int(fx["d"]))
    for j, q in enumerate(queries):
        gids, gdists = h.knn_search(q, k, ef)
        nn = int(fx[f"n_ef{ef}"][j])
        assert np.array_equal(gids, fx[f"ids_ef{ef}"][j][:nn]), f"{name} q{j}"
        assert np.array_equal(gdists, fx[f"dists_ef{ef}"][j][:nn]), \
            f"{name} q{j} dist bits"


@pytest.mark.parametrize("name", FIXTURE_NAMES)
def test_persistent_kernel_matches_oracle_and_perhop(built, name):
    """The persistent in-kernel batched search must return EXACTLY what the
    per-hop path and the oracle fixture return (same graph, same queue
    semantics, bit-exact distances)."""
    h, fx = built[name]
    ef = fixture_ef(fx)
    k = int(fx["k"])
    nq = fx[f"ids_ef{ef}"].shape[0]
    Q = oracle.gen_f32(int(fx["query_seed"]), 0, nq, int(fx["d"]))
    bids, bdists, bns = h.knn_search_batch(Q, k, ef)
    for j in range(nq):
        nn = int(fx[f"n_ef{ef}"][j])
        assert bns[j] == nn
        assert np.array_equal(bids[j][:nn], fx[f"ids_ef{ef}"][j][:nn]), \
            f"{name} q{j} vs oracle"
        assert np.array_equal(bdists[j][:nn], fx[f"dists_ef{ef}"][j][:nn]), \
            f"{name} q{j} dist bits"
        pids, pdists = h.knn_search(Q[j], k, ef)
        assert np.array_equal(bids[j][:nn], pids), f"{name} q{j} vs per-hop"
        assert np.array_equal(bdists[j][:nn], pdists)


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
    # (the reference's own bar at this ef is 1.0 for sequential builds;
    # typical here 0.99-1.0, with a small scheduling-flake margin)
    assert total / len(queries) >= 0.98, total / len(queries)
    h.destroy()
    ctx.drop_table(23)


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
