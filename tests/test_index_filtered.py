"""Filtered KNN (cond_filter pushdown — hnsw/filter.rs + layer.rs:110-318 +
the index.rs filtered flow): product (host-only) vs oracle bit-exact, plus
the reference semantics pinned by hand-derived consequences (cited)."""
import numpy as np
import pytest

import oracle
import surrealdb_amd as sa


def build_pair(d, n, seed=7, metric="euclidean", m=8, m0=16, efc=50):
    rows = oracle.gen_f32(0x1F, 0, n, d)
    p = sa.index_create_host(d, metric=metric, m=m, m0=m0, efc=efc, seed=seed)
    o = oracle.Index(d, metric=metric, m=m, m0=m0, efc=efc, seed=seed)
    for i, r in enumerate(rows):
        p.enqueue(i, None, r)
        o.enqueue(i, None, r)
    p.apply_pendings()
    o.apply_pendings()
    return p, o, rows


def assert_same_filtered(p, o, q, k, ef, pred, msg=""):
    pk, pi, pd = p.knn_search_filtered(q, k, ef, pred)
    ok, oi, od = o.knn_search_filtered(q, k, ef, pred)
    assert np.array_equal(pk, ok), f"{msg}: kinds"
    assert np.array_equal(pi, oi), f"{msg}: ids"
    assert np.array_equal(pd, od), f"{msg}: dist bits"
    return pk, pi, pd


@pytest.mark.parametrize("metric", ["euclidean", "cosine"])
def test_filtered_matches_oracle_and_excludes_non_truthy(metric):
    p, o, rows = build_pair(24, 300, metric=metric)
    pred = lambda kind, i: i % 3 != 0
    for j, q in enumerate(oracle.gen_f32(0xBEEF, 0, 15, 24)):
        _, ids, _ = assert_same_filtered(p, o, q, 10, 40, pred, f"q{j}")
        assert all(int(i) % 3 != 0 for i in ids), f"q{j}: non-truthy leaked"
        assert len(ids) == 10  # plenty of truthy docs in range
    p.destroy()


def test_filtered_with_pendings_and_quirks():
    d = 16
    p, o, rows = build_pair(d, 150)
    # pending update of doc 9 (truthy) and doc 4 (non-truthy under pred)
    upd9 = rows[9] + 1.0
    upd4 = rows[4] + 1.0
    for ix in (p, o):
        ix.enqueue(9, rows[9], upd9)
        ix.enqueue(4, rows[4], upd4)
    pred = lambda kind, i: i != 4
    _, ids, dists = assert_same_filtered(p, o, upd9, 6, 30, pred, "pend")
    assert ids[0] == 9 and dists[0] == 0.0  # pendings overlay, truthy
    assert 4 not in ids.tolist()  # pending id 4 filtered in search_pendings
    # after apply the same predicate holds on the graph path
    p.apply_pendings()
    o.apply_pendings()
    _, ids, dists = assert_same_filtered(p, o, upd9, 6, 30, pred, "applied")
    assert ids[0] == 9 and dists[0] == 0.0
    assert 4 not in ids.tolist()
    p.destroy()


def test_filter_cache_one_evaluation_per_id():
    """filter.rs:70-108: results are cached per VectorId — repeated checks
    during one search are free. Pinned by counting callback invocations."""
    p, o, rows = build_pair(12, 120)
    q = rows[3]
    for ix in (p, o):
        calls = {}
        ix.knn_search_filtered(
            q, 10, 60, lambda kind, i, c=calls: c.setdefault(i, 0) is not None
            and not c.update({i: c[i] + 1}) and True)
        # every consulted id was evaluated at least once; re-consults hit
        # the cache unless the id was evicted from the builder in between
        # (expire), so counts stay tiny (1 almost everywhere)
        assert calls and max(calls.values()) <= 3, calls
    p.destroy()


def test_entry_point_seed_quirk():
    """layer.rs:125-135: search_single_with_filter seeds add_if_truthy with
    SEARCH.PT as the entry point's vector — the ep enters w only when the
    query vector itself is an indexed vector (with a truthy doc). Both
    implementations must reproduce this, bit for bit."""
    d = 8
    p, o, rows = build_pair(d, 60, m=4, m0=8, efc=30)
    all_true = lambda kind, i: True
    # query = an indexed vector: normal results
    _, ids_hit, _ = assert_same_filtered(p, o, rows[11], 5, 20, all_true,
                                         "indexed query")
    assert ids_hit[0] == 11
    # query = NOT an indexed vector: still exact agreement (the ep-seed
    # lookup misses on both sides identically)
    q = rows[11] + np.float32(0.25)
    assert_same_filtered(p, o, q, 5, 20, all_true, "non-indexed query")
    p.destroy()


def test_filtered_scarce_truthy_docs():
    """With only a handful of truthy docs, the filtered search must still
    find them (candidates expand unconditionally, layer.rs:252-254) and
    both implementations agree exactly."""
    d = 16
    p, o, rows = build_pair(d, 400, efc=80)
    allowed = {7, 97, 211, 333}
    pred = lambda kind, i: i in allowed
    for j, q in enumerate(oracle.gen_f32(0xCAFE, 0, 8, d)):
        _, ids, _ = assert_same_filtered(p, o, q, 4, 80, pred, f"scarce q{j}")
        assert set(ids.tolist()) <= allowed
    p.destroy()


def test_expire_signal_fires_on_eviction():
    """filter.rs:141-151 expires: the host cache hears about ids evicted
    from the result builder; both implementations emit the same signal
    multiset. Evictions need more than k doc-entries in the builder —
    the graph path alone contributes <= k (to_vec_limit), so an
    outstanding pendings overlay provides the overflow."""
    d = 12
    p, o, rows = build_pair(d, 200)
    extra = oracle.gen_f32(0xE2, 0, 12, d)
    q = oracle.gen_f32(0xEE, 0, 1, d)[0]
    for ix in (p, o):
        for j in range(12):  # 12 pending docs near nothing in particular
            ix.enqueue(500 + j, None, extra[j])
    logs = []
    for ix in (p, o):
        log = []
        ix.knn_search_filtered(q, 5, 60, lambda kind, i: True,
                               expire=lambda kind, i: log.append((kind, i)))
        logs.append(sorted(log))
    assert logs[0] == logs[1]
    assert logs[0], "12 pendings + graph into a k=5 builder must evict"
    p.destroy()


@pytest.mark.parametrize("case", range(4))
def test_filtered_param_sweep(case):
    """Randomized filtered searches (with pendings outstanding and expire
    logs) across parameter combos — the bounded form of the exploration
    cycle in tools/explore.py."""
    rng = np.random.default_rng(77000 + case)
    d = int(rng.choice([8, 16, 32]))
    m = int(rng.choice([3, 4, 8]))
    m0 = int(rng.choice([m, 2 * m]))
    efc = int(rng.choice([8, 24, 48]))
    metric = str(rng.choice(["euclidean", "cosine"]))
    ext = bool(rng.integers(0, 2))
    keep = bool(rng.integers(0, 2))
    seed = int(rng.integers(1, 2**31))
    n = int(rng.integers(20, 150))
    mod = int(rng.integers(2, 6))
    rows = oracle.gen_f32(seed ^ 0x123, 0, 256, d)
    p = sa.index_create_host(d, metric=metric, m=m, m0=m0, efc=efc,
                             extend=ext, keep=keep, seed=seed)
    o = oracle.Index(d, metric=metric, m=m, m0=m0, efc=efc, extend=ext,
                     keep=keep, seed=seed)
    live = {}
    for i in range(n):
        key = int(rng.integers(0, 48))
        r = rng.integers(0, 4)
        if r < 2 or key not in live:
            v = rows[int(rng.integers(0, 256))]
            p.enqueue(key, live.get(key), v)
            o.enqueue(key, live.get(key), v)
            live[key] = v
        elif r == 2:
            p.enqueue(key, live[key], None)
            o.enqueue(key, live[key], None)
            del live[key]
        else:
            assert p.apply_pendings() == o.apply_pendings()
    pred = lambda kind, i: (int(i) % mod) != 0
    plog, elog = [], []
    for j in range(5):
        q = rows[int(rng.integers(0, 256))] + np.float32(0.01)
        k = int(rng.integers(1, 10))
        ef = int(rng.integers(k, 40))
        a = p.knn_search_filtered(q, k, ef, pred,
                                  expire=lambda kk, ii: plog.append((kk, ii)))
        b = o.knn_search_filtered(q, k, ef, pred,
                                  expire=lambda kk, ii: elog.append((kk, ii)))
        for x, y in zip(a, b):
            assert np.array_equal(x, y), (case, j)
    assert sorted(plog) == sorted(elog), case
    p.destroy()
