"""End-to-end restatements of the reference's own .surql KNN fixtures
(language-tests/tests/language/indexes/knn/*.surql and
reproductions/7229_*.surql) — the literal expected rows from the fixture
headers, pinned through our index/operator layer. DEFINE INDEX defaults per
the parser (syn/parser/stmt/define.rs:1106-1171): DIST EUCLIDEAN, TYPE F32,
EFC 150, M 12, M0 2*M, ML 1/ln(M).

Run on both the product (host-only mode) and the oracle — both must hit the
fixtures' exact expected distances and id order."""
import numpy as np

import oracle
import surrealdb_amd as sa


def both(d, metric="euclidean", m=12, efc=150):
    p = sa.index_create_host(d, metric=metric, m=m, m0=2 * m, efc=efc,
                             seed=0x5DB1)
    o = oracle.Index(d, metric=metric, m=m, m0=2 * m, efc=efc, seed=0x5DB1)
    return p, o


def test_hnsw_knn_surql():
    """hnsw_knn.surql: CREATE pts:1/2, CREATE pts:3 (no vector),
    DEFINE INDEX ... EFC 500 M 12 (build-from-existing), UPDATE pts:3,
    then `point <|2,100|> [2,3,4,5]` -> [(2f, pts:1), (4f, pts:2)];
    finally DELETE pts:3."""
    p, o = both(4, efc=500, m=12)
    v1 = np.array([1, 2, 3, 4], dtype=np.float32)
    v2 = np.array([4, 5, 6, 7], dtype=np.float32)
    v3 = np.array([8, 9, 10, 11], dtype=np.float32)
    q = np.array([2, 3, 4, 5], dtype=np.float32)
    for ix in (p, o):
        # DEFINE INDEX on the populated table: build-from-existing enqueues
        # each record's vectors (pts:3 has none -> nothing to enqueue,
        # content_to_vectors filters nullish, index.rs:118-129)
        ix.enqueue(1, None, v1)
        ix.enqueue(2, None, v2)
        ix.apply_pendings()
        # UPDATE pts:3 SET point = [8,9,10,11]
        ix.enqueue(3, None, v3)
        ix.apply_pendings()
        for ef in (100, 500):  # <|2,100|> and <|2,EUCLIDEAN|> (ef=500=efc)
            kinds, ids, dists = ix.knn_search(q, 2, ef)
            assert ids.tolist() == [0, 1]  # pts:1, pts:2 (doc order)
            assert dists.tolist() == [2.0, 4.0]  # exact fixture distances
        # DELETE pts:3
        ix.enqueue(3, v3, None)
        ix.apply_pendings()
        assert ix.doc_count() == 2
    p.destroy()


def test_7229_knn_k_distance_bypasses_hnsw_surql():
    """reproductions/7229: three docs, index built after, both operator
    spellings -> [(2f, pts:1), (4f, pts:2)]."""
    p, o = both(4, efc=500, m=12)
    vs = [np.array(v, dtype=np.float32) for v in
          ([1, 2, 3, 4], [4, 5, 6, 7], [8, 9, 10, 11])]
    q = np.array([2, 3, 4, 5], dtype=np.float32)
    for ix in (p, o):
        for i, v in enumerate(vs):
            ix.enqueue(i + 1, None, v)
        ix.apply_pendings()
        for ef in (100, 500):
            kinds, ids, dists = ix.knn_search(q, 2, ef)
            assert ids.tolist() == [0, 1]
            assert dists.tolist() == [2.0, 4.0]
    p.destroy()


def test_hnsw_knn_with_condition_surql():
    """hnsw_knn_with_condition.surql: DEFINE INDEX (all defaults, dim 1),
    7 docs with alternating `flag`, `WHERE flag = true AND point <|2,40|>
    [44f]` -> [(6f, pts:5), (14f, pts:3)]."""
    # The fixture is DIMENSION 1; the product staging path requires d%4==0,
    # so the restatement runs d=4 with the value in lane 0 and zeros
    # elsewhere — distance-preserving for euclidean, same graph shape.
    p, o = both(4, efc=150, m=12)
    points = [10.0, 20.0, 30.0, 40.0, 50.0, 60.0, 70.0]
    flags = [True, False, True, False, True, False, True]
    q = np.array([44.0, 0, 0, 0], dtype=np.float32)
    for ix in (p, o):
        for i, pt in enumerate(points):
            ix.enqueue(i + 1, None,
                       np.array([pt, 0, 0, 0], dtype=np.float32))
        ix.apply_pendings()
        truthy = lambda kind, doc: flags[int(doc)]
        kinds, ids, dists = ix.knn_search_filtered(q, 2, 40, truthy)
        # pts:5 = doc 4 (dist 6), pts:3 = doc 2 (dist 14)
        assert ids.tolist() == [4, 2]
        assert dists.tolist() == [6.0, 14.0]
    p.destroy()


def test_bruteforce_knn_new_executor_surql():
    """bruteforce_knn_new_executor.surql: 5 points on a line, q=[1,0], k=2
    euclidean -> [(1f, pts:2), (2f, pts:3)] — via the KnnTopK semantics
    (oracle restatement; GPU scan parity is covered by test_gpu_parity)."""
    corpus = np.array([[10, 0, 0, 0], [2, 0, 0, 0], [3, 0, 0, 0],
                       [100, 0, 0, 0], [50, 0, 0, 0]], dtype=np.float32)
    q = np.array([1, 0, 0, 0], dtype=np.float32)
    ids, dists = oracle.topk_f32("euclidean", corpus, q, 2)
    assert ids.tolist() == [1, 2]  # pts:2, pts:3 (0-based rows)
    assert dists.tolist() == [1.0, 2.0]
    # Number-path (Distance::compute over Vec<Number>, the operator's
    # actual loop) gives the same exact values on these integers
    idsn, distsn = oracle.topk_number("euclidean",
                                      corpus.astype(np.float64), q
                                      .astype(np.float64), 2)
    assert idsn.tolist() == [1, 2] and distsn.tolist() == [1.0, 2.0]


def test_bruteforce_knn_with_filter_surql():
    """bruteforce_knn_with_filter_new_executor.surql: the predicate runs
    BEFORE KnnTopK (TableScan [predicate: active = true] -> KnnTopK in the
    committed plan), so only active rows compete for the top-K:
    [(2f, pts:3), (9f, pts:1)] — pts:2 at distance 1 is inactive."""
    pts = np.array([[10, 0, 0, 0], [2, 0, 0, 0], [3, 0, 0, 0],
                    [100, 0, 0, 0], [50, 0, 0, 0]], dtype=np.float32)
    active = np.array([True, False, True, True, False])
    q = np.array([1, 0, 0, 0], dtype=np.float32)
    # boundary form: the host streams only predicate-matching rows into
    # the scan (a filtered sub-corpus with its original ids)
    sub = np.ascontiguousarray(pts[active])
    sub_ids = np.flatnonzero(active)
    ids, dists = oracle.topk_f32("euclidean", sub, q, 2)
    got = [(float(d), int(sub_ids[i])) for i, d in zip(ids, dists)]
    assert got == [(2.0, 2), (9.0, 0)]  # pts:3, pts:1 (0-based rows)


def test_bruteforce_knn_multisource_filter_surql():
    """bruteforce_knn_multisource_filter_new_executor.surql: Union of two
    tables -> Filter -> KnnTopK; expected [(2f, pts:3), (3f, pts2:2)] —
    the nearest rows overall (pts2:1 at 0.5, pts:2 at 1) are inactive."""
    union = np.array([[10, 0, 0, 0], [2, 0, 0, 0], [3, 0, 0, 0],
                      [1.5, 0, 0, 0], [4, 0, 0, 0]], dtype=np.float32)
    active = np.array([True, False, True, False, True])
    q = np.array([1, 0, 0, 0], dtype=np.float32)
    sub = np.ascontiguousarray(union[active])
    sub_ids = np.flatnonzero(active)
    ids, dists = oracle.topk_f32("euclidean", sub, q, 2)
    got = [(float(d), int(sub_ids[i])) for i, d in zip(ids, dists)]
    assert got == [(2.0, 2), (3.0, 4)]  # pts:3, pts2:2
