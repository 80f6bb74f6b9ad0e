"""GPU index layer: sdbv_index with a device context must return exactly
what the oracle index returns — including after writes (auto-refinalize:
device re-stage + dangling-edge scrub) and with pendings outstanding (the
per-hop GPU search with the pending-docs gate)."""
import numpy as np
import pytest

import oracle
import surrealdb_amd as sa

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctx():
    c = sa.Context()
    yield c
    c.close()


def assert_same_search(p, o, q, k, ef, msg=""):
    pk, pi, pd = p.knn_search(q, k, ef)
    ok, oi, od = o.knn_search(q, k, ef)
    assert np.array_equal(pk, ok), f"{msg}: kinds differ"
    assert np.array_equal(pi, oi), f"{msg}: ids differ"
    assert np.array_equal(pd, od), f"{msg}: distance bits differ"


@pytest.mark.parametrize("metric", ["cosine", "euclidean"])
def test_gpu_index_matches_oracle_through_writes(ctx, metric):
    d, n = 64, 3000
    rows = oracle.gen_f32(0xA1, 0, n, d)
    upd = oracle.gen_f32(0xA2, 0, n, d)
    p = sa.Index(ctx, 40, d, metric=metric, m=8, m0=16, efc=60, seed=0xF)
    o = oracle.Index(d, metric=metric, m=8, m0=16, efc=60, seed=0xF)
    for i, r in enumerate(rows):
        p.enqueue(i, None, r)
        o.enqueue(i, None, r)
    p.apply_pendings()
    o.apply_pendings()
    queries = oracle.gen_f32(0xBEEF, 0, 12, d)
    for j, q in enumerate(queries):
        assert_same_search(p, o, q, 10, 40, f"initial q{j} ({metric})")
    # writes: update a slice, delete a slice — next search re-finalizes
    for i in range(0, 200):
        p.enqueue(i, rows[i], upd[i])
        o.enqueue(i, rows[i], upd[i])
    for i in range(200, 300):
        p.enqueue(i, rows[i], None)
        o.enqueue(i, rows[i], None)
    # with pendings outstanding: pendings overlay + pending-docs gate
    for j, q in enumerate(queries[:4]):
        assert_same_search(p, o, q, 10, 40, f"pending q{j} ({metric})")
    p.apply_pendings()
    o.apply_pendings()
    # after apply: device graph re-staged with scrubbed CSR
    for j, q in enumerate(queries):
        assert_same_search(p, o, q, 10, 40, f"after-apply q{j} ({metric})")
    assert p.doc_count() == o.doc_count() == n - 100
    p.destroy()
    ctx.drop_table(40)


def test_gpu_index_filtered_matches_oracle(ctx):
    d, n = 64, 2000
    rows = oracle.gen_f32(0xB1, 0, n, d)
    p = sa.Index(ctx, 41, d, metric="cosine", m=8, m0=16, efc=60, seed=0x3)
    o = oracle.Index(d, metric="cosine", m=8, m0=16, efc=60, seed=0x3)
    for i, r in enumerate(rows):
        p.enqueue(i, None, r)
        o.enqueue(i, None, r)
    p.apply_pendings()
    o.apply_pendings()
    pred = lambda kind, i: i % 5 != 0
    for j, q in enumerate(oracle.gen_f32(0xBEEF, 0, 8, d)):
        pk, pi, pd = p.knn_search_filtered(q, 10, 40, pred)
        ok, oi, od = o.knn_search_filtered(q, 10, 40, pred)
        assert np.array_equal(pk, ok), f"q{j}: kinds"
        assert np.array_equal(pi, oi), f"q{j}: ids"
        assert np.array_equal(pd, od), f"q{j}: dist bits"
        assert all(int(i) % 5 != 0 for i in pi)
    p.destroy()
    ctx.drop_table(41)
