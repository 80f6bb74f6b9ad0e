"""Product HNSW (host graph build, CPU-side — no GPU needed) vs the oracle.

The product build (surrealdb_amd C++) and the oracle build are INDEPENDENT
restatements of the reference algorithm with the same committed level-RNG
contract; a sequential build must therefore produce the bit-identical graph.
"""
import math

import numpy as np
import pytest

import oracle


@pytest.fixture(scope="module")
def ctxless_hnsw_pair():
    # Context creation needs a GPU; the host-only index builds anywhere.
    import surrealdb_amd
    h = surrealdb_amd.hnsw_create_host(20, metric="euclidean", m=8, m0=16,
                                       efc=100, seed=0x5DB1)
    o = oracle.Hnsw(20, metric="euclidean", m=8, m0=16, efc=100,
                    ml=1.0 / math.log(8.0), seed=0x5DB1)
    yield h, o
    h.destroy()


def test_sequential_build_graph_identical(ctxless_hnsw_pair):
    h, o = ctxless_hnsw_pair
    rows = oracle.gen_f32(0x123, 0, 800, 20)
    for r in rows:
        h.insert(r)
        o.insert(r)
    assert h.n() == 800
    assert h.num_layers() == o.num_layers()
    po, pe = h.l0_csr()
    oo, oe = o.l0_csr()
    assert np.array_equal(po, oo), "layer-0 CSR offsets differ"
    assert np.array_equal(pe, oe), "layer-0 CSR edges differ"


def test_parallel_build_quality():
    """Parallel (bench-mode) build: nondeterministic graph, validated by the
    reference's recall bar (==1.0 @ efs=40 would need GPU search; here check
    structure invariants only — the GPU recall test covers quality)."""
    import surrealdb_amd
    h = surrealdb_amd.hnsw_create_host(32, metric="euclidean", m=8, m0=16,
                                       efc=100, seed=7)
    rows = oracle.gen_f32(0x321, 0, 3000, 32)
    h.insert_batch(rows, nthreads=4)
    assert h.n() == 3000
    offsets, edges = h.l0_csr()
    deg = np.diff(offsets.astype(np.int64))
    assert deg.max() <= 16 + 2  # m0 (+small concurrent-back-edge slack)
    n = 3000
    assert (edges < n).all()
    # no self-edges
    for i in range(n):
        assert not (edges[offsets[i]:offsets[i + 1]] == i).any()
    # connectivity proxy: nearly every node has at least one edge
    assert (deg > 0).mean() > 0.999
    h.destroy()


def test_host_build_matches_committed_fixture():
    """The product host build must reproduce the committed oracle fixture
    graph (tests/golden/hnsw_fix_seq128_cos.npz, built by
    make_hnsw_fixtures.py) — catches fixture/build drift without a GPU."""
    import os
    import numpy as np
    import surrealdb_amd
    fx = np.load(os.path.join(os.path.dirname(__file__), "golden",
                              "hnsw_fix_seq128_cos.npz"))
    d, n, m = int(fx["d"]), int(fx["n"]), int(fx["m"])
    rows = oracle.gen_f32(int(fx["data_seed"]), 0, n, d)
    h = surrealdb_amd.hnsw_create_host(
        d, metric=str(fx["metric"]), m=m, m0=int(fx["m0"]),
        efc=int(fx["efc"]), ml=1.0 / math.log(m), seed=int(fx["seed"]))
    h.insert_batch(rows, nthreads=1)
    assert h.num_layers() == int(fx["num_layers"])
    po, pe = h.l0_csr()
    assert np.array_equal(po, fx["l0_offsets"])
    assert np.array_equal(pe, fx["l0_edges"])
    h.destroy()


def test_host_search_matches_oracle_bitexact():
    """sdbv_hnsw_knn_host (the host-distance search path) must return
    exactly the oracle's builder-sorted results on the identical
    (sequentially built) graph — the CPU twin of the GPU search parity."""
    import numpy as np
    import surrealdb_amd
    d, n = 20, 800
    rows = oracle.gen_f32(0x123, 0, n, d)
    h = surrealdb_amd.hnsw_create_host(d, metric="euclidean", m=8, m0=16,
                                       efc=100, seed=0x5DB1)
    o = oracle.Hnsw(d, metric="euclidean", m=8, m0=16, efc=100,
                    ml=math.log(8.0) ** -1, seed=0x5DB1)
    h.insert_batch(rows, nthreads=1)
    for r in rows:
        o.insert(r)
    from surrealdb_amd.shard import total_key
    for q in oracle.gen_f32(0x99, 0, 20, d):
        gids, gdists = h.knn_search_host(q, 10, 40)
        oids, odists = o.search(q, 10, 40)
        order = np.lexsort((oids, total_key(odists)))
        assert np.array_equal(gids, oids[order])
        assert np.array_equal(gdists, odists[order])
    h.destroy()


def test_snapshot_build_recall_bars():
    """Chunked snapshot build (§8f rank 3 structure: per-chunk searches
    against the chunk-start graph — the GPU-batchable form): must meet the
    same efs=40 recall bar as the parallel build on the reference's golden
    dataset, across chunk sizes."""
    import gzip
    import json
    import os
    import numpy as np
    import surrealdb_amd
    golden = os.path.join(os.path.dirname(__file__), "golden",
                          "hnsw-random-9000-20-euclidean.gz")
    rows = []
    with gzip.open(golden, "rt") as f:
        for i, line in enumerate(f):
            if i >= 2000:
                break
            rows.append(json.loads(line))
    ingest = np.array(rows, dtype=np.float32)
    queries_all = ingest[:100] + np.float32(0.05)

    def rec_at(chunk):
        h = surrealdb_amd.hnsw_create_host(20, metric="euclidean", m=8,
                                           m0=16, efc=100, seed=0x5DB1)
        h.insert_batch_snapshot(ingest, chunk=chunk, nthreads=4)
        total = 0.0
        for q in queries_all:
            ids, _ = h.knn_search_host(q, 10, 40)
            bf, _ = oracle.topk_f32("euclidean", ingest, q, 10)
            total += len(set(ids.tolist()) & set(bf.tolist())) / 10.0
        offsets, edges = h.l0_csr()
        deg = np.diff(offsets.astype(np.int64))
        assert deg.max() <= 16 + 8  # threaded keep-back relaxation
        h.destroy()
        return total / len(queries_all)

    # Quality contract: the recall loss is bounded by the invisible
    # fraction chunk/n (chunk mates are absent from each other's snapshot
    # searches). At representative ratios (<= ~2%, the bench regime —
    # chunk 4096 of 10M rows is 0.04%) the parallel-build bar holds; the
    # threaded build is scheduling-nondeterministic, so the bar carries a
    # small flake margin (typical 0.991-0.997 here):
    assert rec_at(16) >= 0.98   # 0.8% invisible
    assert rec_at(32) >= 0.98   # 1.6% invisible
    # and the contract boundary is real: at an absurd ratio (26%) recall
    # visibly degrades — chunk must be sized << n (typical ~0.89 here)
    assert rec_at(512) < 0.97


def test_snapshot_build_matches_quality_of_parallel_768d():
    """At bench-like dimensionality the snapshot build's recall must be in
    line with the classic parallel build (both searched on the host)."""
    import numpy as np
    import surrealdb_amd
    d, n = 768, 3000
    rows = oracle.gen_f32(0x5DB1, 0, n, d)
    queries = oracle.gen_f32(0xBEEF, 0, 30, d)

    def build_recall(kind):
        h = surrealdb_amd.hnsw_create_host(d, metric="cosine", m=16, m0=32,
                                           efc=150, seed=0x5DB1)
        if kind == "snapshot":
            h.insert_batch_snapshot(rows, chunk=64, nthreads=4)
        else:
            h.insert_batch(rows, nthreads=4)
        tot = 0.0
        for q in queries:
            ids, _ = h.knn_search_host(q, 10, 64)
            bf, _, _ = oracle.topk_f32_mt("cosine", rows, q, 10)
            tot += len(set(ids.tolist()) & set(bf.tolist())) / 10.0
        h.destroy()
        return tot / len(queries)

    r_par = build_recall("parallel")
    r_snap = build_recall("snapshot")
    # uniform 768-dim data has intrinsically low recall (see DESIGN); the
    # snapshot relaxation must not degrade it materially
    assert r_snap >= r_par - 0.08, (r_snap, r_par)


def test_host_search_matches_committed_768d_fixture():
    """Product sequential build + host search against the committed
    4000x768 oracle fixture: graph CSR and every builder-sorted search
    result bit-exact — the CPU twin of the GPU fixture tests."""
    import os
    import numpy as np
    import surrealdb_amd
    fx = np.load(os.path.join(os.path.dirname(__file__), "golden",
                              "hnsw_fix_seq768_cos.npz"))
    d, n, m = int(fx["d"]), int(fx["n"]), int(fx["m"])
    rows = oracle.gen_f32(int(fx["data_seed"]), 0, n, d)
    h = surrealdb_amd.hnsw_create_host(
        d, metric=str(fx["metric"]), m=m, m0=int(fx["m0"]),
        efc=int(fx["efc"]), ml=1.0 / math.log(m), seed=int(fx["seed"]))
    h.insert_batch(rows, nthreads=1)
    po, pe = h.l0_csr()
    assert np.array_equal(po, fx["l0_offsets"])
    assert np.array_equal(pe, fx["l0_edges"])
    k = int(fx["k"])
    Q = oracle.gen_f32(int(fx["query_seed"]), 0, fx["ids_ef64"].shape[0], d)
    for j, q in enumerate(Q):
        ids, dists = h.knn_search_host(q, k, 64)
        nn = int(fx["n_ef64"][j])
        assert np.array_equal(ids, fx["ids_ef64"][j][:nn]), f"q{j}"
        assert np.array_equal(dists, fx["dists_ef64"][j][:nn]), f"q{j}"
    h.destroy()


def test_oracle_import_searches_product_graph_bitexact():
    """The oracle's graph-import path (the bench cpu_baseline leg): a
    product-built graph exported layer by layer into orc_hnsw_import must
    search bit-exactly like the product host search on the same graph."""
    import numpy as np
    import surrealdb_amd
    from surrealdb_amd.shard import total_key
    d, n = 48, 2500
    rows = oracle.gen_f32(0xE1, 0, n, d)
    h = surrealdb_amd.hnsw_create_host(d, metric="cosine", m=8, m0=16,
                                       efc=60, seed=0x5DB1)
    h.insert_batch_snapshot(rows, chunk=256, nthreads=1)
    layers = [h.layer_csr(l) for l in range(h.num_layers())]
    o = oracle.Hnsw.import_graph(d, "cosine", 8, 16, 60, h.vecs_view(),
                                 h.enter_point(), layers)
    assert o.check_props() == 0
    for q in oracle.gen_f32(0xBEEF, 0, 12, d):
        hi, hd = h.knn_search_host(q, 10, 40)
        oi, od = o.search(q, 10, 40)
        order = np.lexsort((oi, total_key(od)))
        assert np.array_equal(hi, oi[order])
        assert np.array_equal(hd, od[order])
    h.destroy()


def test_snapshot2_build_recall_bars():
    """The batched-apply schedule (snapshot2, the GPU build's host twin)
    meets the same recall bars as the interleaved snapshot build on the
    reference's golden dataset."""
    import gzip
    import json
    import os
    import numpy as np
    import surrealdb_amd
    golden = os.path.join(os.path.dirname(__file__), "golden",
                          "hnsw-random-9000-20-euclidean.gz")
    rows = []
    with gzip.open(golden, "rt") as f:
        for i, line in enumerate(f):
            if i >= 2000:
                break
            rows.append(json.loads(line))
    ingest = np.array(rows, dtype=np.float32)
    queries = ingest[:100] + np.float32(0.05)
    for chunk in (16, 32):
        h = surrealdb_amd.hnsw_create_host(20, metric="euclidean", m=8,
                                           m0=16, efc=100, seed=0x5DB1)
        h.insert_batch_snapshot2(ingest, chunk=chunk, nthreads=4)
        total = 0.0
        for q in queries:
            ids, _ = h.knn_search_host(q, 10, 40)
            bf, _ = oracle.topk_f32("euclidean", ingest, q, 10)
            total += len(set(ids.tolist()) & set(bf.tolist())) / 10.0
        offsets, edges = h.l0_csr()
        deg = np.diff(offsets.astype(np.int64))
        # threaded build: transient >m0 is documented (keep-back race);
        # strict bound asserted on the deterministic schedule elsewhere
        assert deg.max() <= 16 + 8, chunk
        assert total / len(queries) >= 0.98, (chunk, total / len(queries))
        h.destroy()


def test_snapshot2_keep_flag_and_768d_quality():
    """keep_pruned_connections through the batched schedule, plus
    768-dim quality parity with the classic parallel build."""
    import numpy as np
    import surrealdb_amd
    d, n = 768, 2000
    rows = oracle.gen_f32(0x5DB1, 0, n, d)
    queries = oracle.gen_f32(0xBEEF, 0, 20, d)

    def build_recall(kind, keep=False):
        h = surrealdb_amd.hnsw_create_host(d, metric="cosine", m=16, m0=32,
                                           efc=150, keep=keep, seed=0x5DB1)
        if kind == "snapshot2":
            h.insert_batch_snapshot2(rows, chunk=64, nthreads=4)
        else:
            h.insert_batch(rows, nthreads=4)
        tot = 0.0
        for q in queries:
            ids, _ = h.knn_search_host(q, 10, 64)
            bf, _, _ = oracle.topk_f32_mt("cosine", rows, q, 10)
            tot += len(set(ids.tolist()) & set(bf.tolist())) / 10.0
        h.destroy()
        return tot / len(queries)

    r_par = build_recall("parallel")
    r_s2 = build_recall("snapshot2")
    r_s2k = build_recall("snapshot2", keep=True)
    assert r_s2 >= r_par - 0.08, (r_s2, r_par)
    assert r_s2k >= r_par - 0.12, (r_s2k, r_par)


def test_snapshot2_deterministic_and_incremental():
    """snapshot2 at nthreads=1 is fully deterministic (same seed, same
    data -> bit-identical graphs), and two incremental calls equal...
    themselves deterministically (the GPU twin's incremental test mirrors
    this on hardware)."""
    import numpy as np
    import surrealdb_amd
    d, n = 32, 1500
    rows = oracle.gen_f32(0xFEED, 0, n, d)

    def build(split):
        h = surrealdb_amd.hnsw_create_host(d, metric="cosine", m=8, m0=16,
                                           efc=60, seed=0x77)
        if split:
            h.insert_batch_snapshot2(rows[:900], chunk=128, nthreads=1)
            h.insert_batch_snapshot2(rows[900:], chunk=128, nthreads=1)
        else:
            h.insert_batch_snapshot2(rows, chunk=128, nthreads=1)
        return h

    a, b = build(False), build(False)
    ca, cb = a.l0_csr(), b.l0_csr()
    assert np.array_equal(ca[0], cb[0]) and np.array_equal(ca[1], cb[1])
    c, e = build(True), build(True)
    cc, ce = c.l0_csr(), e.l0_csr()
    assert np.array_equal(cc[0], ce[0]) and np.array_equal(cc[1], ce[1])
    for q in oracle.gen_f32(0xB, 0, 5, d):
        i1, d1 = a.knn_search_host(q, 10, 40)
        i2, d2 = b.knn_search_host(q, 10, 40)
        assert np.array_equal(i1, i2) and np.array_equal(d1, d2)
    for h in (a, b, c, e):
        h.destroy()
