"""Product index layer (surrealdb_amd C++, host-only mode) vs the oracle
index: INDEPENDENT restatements of hnsw/index.rs + docs.rs + knn.rs with the
same committed level-RNG contract, so every observable — result kinds/ids/
distance bits, doc counts, graph CSR — must be bit-identical at every step.

CPU-only (ctx == NULL): the host graph search is the same code the GPU path
shares its queue semantics with; GPU parity of the graph search itself is
covered by tests/test_gpu_hnsw.py."""
import numpy as np
import pytest

import oracle
import surrealdb_amd as sa


def both(d, metric="euclidean", m=8, m0=16, efc=60, seed=0xC):
    p = sa.index_create_host(d, metric=metric, m=m, m0=m0, efc=efc, seed=seed)
    o = oracle.Index(d, metric=metric, m=m, m0=m0, efc=efc, seed=seed)
    return p, o


def assert_same_search(p, o, q, k, ef, msg=""):
    pk, pi, pd = p.knn_search(q, k, ef)
    ok, oi, od = o.knn_search(q, k, ef)
    assert np.array_equal(pk, ok), f"{msg}: kinds differ"
    assert np.array_equal(pi, oi), f"{msg}: ids differ"
    assert np.array_equal(pd, od), f"{msg}: distance bits differ"


def assert_same_graph(p, o, msg=""):
    ph, oh = p.hnsw(), o.hnsw()
    assert ph.num_layers() == oh.num_layers(), msg
    po, pe = ph.l0_csr()
    oo, oe = oh.l0_csr()
    assert np.array_equal(po, oo), f"{msg}: CSR offsets differ"
    assert np.array_equal(pe, oe), f"{msg}: CSR edges differ"


class ApplyModel:
    """Minimal model of index_pending apply semantics (index.rs:214-257),
    used only to predict doc/element counts for check_props. Notably: a
    pending's id KIND is fixed at enqueue time, so a delete enqueued before
    its key ever resolved to a DocId carries VectorId::RecordKey and its
    old-vector removal is silently skipped at apply (index.rs:229-242 —
    the removal block runs only for DocId pendings)."""

    def __init__(self):
        self.key2doc = {}
        self.docs = {}  # doc -> set of vector bytes
        self.next = 0
        self.batch = []

    def enqueue(self, key, old, new):
        self.batch.append((self.key2doc.get(key), key,
                           None if old is None else old.tobytes(),
                           None if new is None else new.tobytes()))

    def apply(self):
        for doc_at_enqueue, key, old, new in self.batch:
            if doc_at_enqueue is not None:
                if old is not None:
                    self.docs.get(doc_at_enqueue, set()).discard(old)
                if new is None and self.key2doc.get(key) == doc_at_enqueue:
                    del self.key2doc[key]
            if new is not None:
                if doc_at_enqueue is not None:
                    doc = doc_at_enqueue
                elif key in self.key2doc:
                    doc = self.key2doc[key]
                else:
                    doc = self.next
                    self.next += 1
                    self.key2doc[key] = doc
                self.docs.setdefault(doc, set()).add(new)
        self.batch = []

    def n_elements(self):
        return len({v for s in self.docs.values() for v in s})

    def n_docs(self):
        return len(self.key2doc)


@pytest.mark.parametrize("metric", ["euclidean", "cosine"])
def test_randomized_write_workload_matches_oracle(metric):
    """Randomized insert/update/delete/search workload applied identically
    to both implementations; results and graph compared at every apply."""
    d, n = 24, 90
    rows = oracle.gen_f32(0x1234, 0, n, d)
    extra = oracle.gen_f32(0x4321, 0, n, d)
    p, o = both(d, metric=metric)
    model = ApplyModel()
    rng = np.random.default_rng(7)
    live = {}  # key -> current vector
    for step in range(160):
        op = rng.integers(0, 4)
        key = int(rng.integers(0, n))
        if op == 0 or key not in live:  # insert / first write
            v = rows[key] if key not in live else extra[key]
            old = None if key not in live else live[key]
            p.enqueue(key, old, v)
            o.enqueue(key, old, v)
            model.enqueue(key, old, v)
            live[key] = v
        elif op == 1:  # update
            v = extra[key] * np.float32(0.5 + (step % 3))
            p.enqueue(key, live[key], v)
            o.enqueue(key, live[key], v)
            model.enqueue(key, live[key], v)
            live[key] = v
        elif op == 2:  # delete
            p.enqueue(key, live[key], None)
            o.enqueue(key, live[key], None)
            model.enqueue(key, live[key], None)
            del live[key]
        else:  # batch boundary: apply + full comparison
            assert p.apply_pendings() == o.apply_pendings()
            model.apply()
            assert p.doc_count() == o.doc_count() == model.n_docs()
            n_elems = model.n_elements()
            assert p.check_props(n_elems) == 0, f"step {step}"
            assert o.check_props(n_elems) == 0, f"step {step}"
            assert_same_graph(p, o, f"step {step}")
        # search with or without outstanding pendings
        q = rows[int(rng.integers(0, n))] + np.float32(0.1)
        assert_same_search(p, o, q, 10, 40, f"step {step}")
    p.apply_pendings()
    o.apply_pendings()
    model.apply()
    assert_same_graph(p, o, "final")
    for q in oracle.gen_f32(0xBEEF, 0, 10, d):
        assert_same_search(p, o, q, 10, 40, "final")
    p.destroy()


def test_duplicate_vectors_and_bits_sets_match_oracle():
    d = 12
    v = oracle.gen_f32(0x9, 0, 3, d)
    p, o = both(d, efc=30, m=4, m0=8)
    for i in range(12):  # one vector shared by 12 docs -> Vec8 -> Bits
        p.enqueue(i, None, v[0])
        o.enqueue(i, None, v[0])
    p.enqueue(70, None, v[1])
    o.enqueue(70, None, v[1])
    assert_same_search(p, o, v[0], 20, 20, "pending dups")
    p.apply_pendings()
    o.apply_pendings()
    assert p.doc_count() == o.doc_count() == 13
    assert p.check_props(2) == o.check_props(2) == 0
    assert_same_search(p, o, v[0], 20, 20, "applied dups")
    # remove one of the shared docs, then the whole shared vector
    p.enqueue(4, v[0], None)
    o.enqueue(4, v[0], None)
    p.apply_pendings()
    o.apply_pendings()
    assert_same_search(p, o, v[0], 20, 20, "after one removal")
    for i in list(range(12)) + []:
        if i == 4:
            continue
        p.enqueue(i, v[0], None)
        o.enqueue(i, v[0], None)
    p.apply_pendings()
    o.apply_pendings()
    assert p.doc_count() == o.doc_count()
    assert_same_graph(p, o, "shared vector fully removed")
    assert_same_search(p, o, v[0], 5, 20, "post-removal")
    p.destroy()


def test_record_key_pendings_and_recycling_match_oracle():
    d = 8
    rows = oracle.gen_f32(0x51, 0, 8, d)
    p, o = both(d, m=4, m0=8, efc=20, seed=1)
    for i in range(4):
        p.enqueue(800 + i, None, rows[i])
        o.enqueue(800 + i, None, rows[i])
    # searches with RecordKey-kind pendings outstanding
    assert_same_search(p, o, rows[2], 4, 16, "recordkey pendings")
    p.apply_pendings()
    o.apply_pendings()
    p.enqueue(801, rows[1], None)
    o.enqueue(801, rows[1], None)
    p.apply_pendings()
    o.apply_pendings()
    p.enqueue(950, None, rows[5])  # reuses the recycled doc id
    o.enqueue(950, None, rows[5])
    p.apply_pendings()
    o.apply_pendings()
    assert_same_search(p, o, rows[5], 3, 16, "recycled id")
    pk, pi, pd = p.knn_search(rows[5], 1, 16)
    assert (pk[0], pi[0], pd[0]) == (0, 1, 0.0)
    p.destroy()


def test_graph_remove_standalone_matches_oracle():
    """sdbv_hnsw_remove vs orc_hnsw_remove on identical sequential builds:
    bit-identical repaired graphs (CSR + entry point) after every removal."""
    d, n = 20, 300
    rows = oracle.gen_f32(0x33, 0, n, d)
    ph = sa.hnsw_create_host(d, metric="euclidean", m=8, m0=16, efc=80,
                             seed=0x22)
    oh = oracle.Hnsw(d, metric="euclidean", m=8, m0=16, efc=80, seed=0x22)
    ph.insert_batch(rows, nthreads=1)
    for r in rows:
        oh.insert(r)
    order = np.random.default_rng(3).permutation(n)
    for j, e in enumerate(order[: n // 2]):
        assert ph.remove(int(e)) is True
        assert oh.remove(int(e)) is True
        if j % 25 == 0:
            po, pe = ph.l0_csr()
            oo, oe = oh.l0_csr()
            assert np.array_equal(po, oo), f"removal {j}: offsets"
            assert np.array_equal(pe, oe), f"removal {j}: edges"
    po, pe = ph.l0_csr()
    oo, oe = oh.l0_csr()
    assert np.array_equal(po, oo) and np.array_equal(pe, oe)
    assert ph.remove(int(order[0])) is False  # double-remove is a no-op
    assert oh.remove(int(order[0])) is False
    ph.destroy()


def test_empty_and_edge_cases():
    d = 8
    p, o = both(d, m=4, m0=8, efc=20)
    # search on an empty index
    q = oracle.gen_f32(1, 0, 1, d)[0]
    assert_same_search(p, o, q, 5, 10, "empty")
    # enqueue + delete before any apply (RecordKey delete pending:
    # old_vectors present but kind is RecordKey -> no graph removal)
    v = oracle.gen_f32(2, 0, 1, d)[0]
    p.enqueue(1, None, v)
    o.enqueue(1, None, v)
    p.enqueue(1, v, None)
    o.enqueue(1, v, None)
    assert_same_search(p, o, v, 5, 10, "insert+delete pending")
    pn = p.apply_pendings()
    on = o.apply_pendings()
    assert pn == on == 2
    assert p.doc_count() == o.doc_count()
    assert_same_search(p, o, v, 5, 10, "after apply")
    p.destroy()


@pytest.mark.parametrize("case", range(4))
def test_graph_remove_param_sweep(case):
    """Randomized removal orders + reinserts across parameter combos
    (metric x ext/keep heuristics x m0=m): bit-identical repaired graphs
    between the two independent restatements at every step."""
    import math
    rng = np.random.default_rng(88000 + case)
    d = int(rng.choice([8, 16, 24]))
    m = int(rng.choice([3, 4, 8]))
    m0 = int(rng.choice([m, 2 * m]))
    efc = int(rng.choice([10, 30, 60]))
    metric = str(rng.choice(["euclidean", "cosine"]))
    ext = bool(rng.integers(0, 2))
    keep = bool(rng.integers(0, 2))
    seed = int(rng.integers(1, 2**31))
    n = int(rng.integers(30, 150))
    rows = oracle.gen_f32(seed ^ 0x77, 0, n, d)
    h = sa.hnsw_create_host(d, metric=metric, m=m, m0=m0, efc=efc,
                            extend=ext, keep=keep, seed=seed)
    o = oracle.Hnsw(d, metric=metric, m=m, m0=m0, efc=efc, extend=ext,
                    keep=keep, seed=seed, ml=1.0 / math.log(m))
    h.insert_batch(rows, nthreads=1)
    for r in rows:
        o.insert(r)
    order = rng.permutation(n)
    nrem = int(rng.integers(1, n))
    for j, e in enumerate(order[:nrem]):
        assert h.remove(int(e)) == o.remove(int(e)), (case, j)
    a, b = h.l0_csr(), o.l0_csr()
    assert np.array_equal(a[0], b[0]) and np.array_equal(a[1], b[1])
    for r in oracle.gen_f32(seed ^ 0x99, 0, 10, d):
        h.insert(r)
        o.insert(r)
    a, b = h.l0_csr(), o.l0_csr()
    assert np.array_equal(a[0], b[0]) and np.array_equal(a[1], b[1])
    h.destroy()


def test_extreme_small_params_match_oracle():
    """Corner parameters (m=2, m0=2, efc=2, ef=1): the degenerate windows
    exercise the prune/eviction edges hardest; both restatements must
    still agree bit-exactly."""
    d = 4
    rows = oracle.gen_f32(0xC0FFEE, 0, 64, d)
    p = sa.index_create_host(d, metric="euclidean", m=2, m0=2, efc=2,
                             seed=11)
    o = oracle.Index(d, metric="euclidean", m=2, m0=2, efc=2, seed=11)
    rng = np.random.default_rng(5)
    live = {}
    for i in range(120):
        key = int(rng.integers(0, 16))
        r = rng.integers(0, 5)
        if r < 2 or key not in live:
            v = rows[int(rng.integers(0, 64))]
            p.enqueue(key, live.get(key), v)
            o.enqueue(key, live.get(key), v)
            live[key] = v
        elif r == 2:
            p.enqueue(key, live[key], None)
            o.enqueue(key, live[key], None)
            del live[key]
        elif r == 3:
            assert p.apply_pendings() == o.apply_pendings()
        else:
            q = rows[int(rng.integers(0, 64))]
            assert_same_search(p, o, q, int(rng.integers(1, 5)), 1,
                               f"step {i} ef=1")
    assert p.apply_pendings() == o.apply_pendings()
    assert_same_graph(p, o, "extreme params final")
    p.destroy()
