"""Negative paths of the C-ABI: misuse must return error codes (and raise
SdbvError through the wrappers), never crash or silently fall back."""
import numpy as np
import pytest

import oracle
import surrealdb_amd as sa


def test_hnsw_create_rejects_bad_args():
    for kwargs in (dict(d=0), dict(d=6), dict(m=0)):
        # m=0 dies in the Python wrapper (ml = 1/ln(m)); the rest in C
        with pytest.raises((sa.SdbvError, ValueError)):
            d = kwargs.pop("d", 8)
            sa.hnsw_create_host(d, metric="euclidean", **kwargs)


def test_unsupported_metric_rejected():
    # the scan accepts every catalog::Distance variant; the HNSW graph
    # engine still takes cosine/euclidean only (the reference's own HNSW
    # defaults; other metrics on the graph are future work) — rejected
    # loudly, never silently downgraded
    with pytest.raises(sa.SdbvError):
        sa.hnsw_create_host(8, metric="manhattan")
    with pytest.raises(KeyError):
        sa.hnsw_create_host(8, metric="nonsense")


def test_hnsw_remove_on_finalized_graph_rejected():
    h = sa.hnsw_create_host(8, metric="euclidean", m=4, m0=8, efc=20)
    h.insert(np.zeros(8, dtype=np.float32))
    assert h.remove(5) is False  # nonexistent id: no-op, not an error
    h.destroy()


def test_index_knn_bad_args():
    ix = sa.index_create_host(8, metric="euclidean", m=4, m0=8, efc=20)
    with pytest.raises(sa.SdbvError):
        ix.knn_search(np.zeros(8, dtype=np.float32), 0, 10)  # k == 0
    ix.destroy()


def test_kvload_rejects_malformed_keys_and_missing_state():
    # not a /*...!h? key at all
    with pytest.raises(sa.SdbvError):
        sa.load_kv_hnsw([(b"garbage", b"")], 8)
    # well-formed pairs but no Hs state record
    h = sa.hnsw_create_host(8, metric="euclidean", m=4, m0=8, efc=20)
    h.insert(np.zeros(8, dtype=np.float32))
    pairs = [(k, v) for k, v in h.dump_kv() if b"!hs" not in k]
    with pytest.raises(sa.SdbvError):
        sa.load_kv_hnsw(pairs, 8)
    h.destroy()


def test_kvload_rejects_wrong_dimension():
    h = sa.hnsw_create_host(8, metric="euclidean", m=4, m0=8, efc=20)
    h.insert(np.zeros(8, dtype=np.float32))
    pairs = h.dump_kv()
    with pytest.raises(sa.SdbvError):
        sa.load_kv_hnsw(pairs, 12)  # d mismatch vs the He payloads
    h.destroy()


def test_index_thread_safety_smoke():
    """Concurrent searches + writes through the index mutex (the
    reference's RwLock discipline, index.rs:55): no crashes, results
    always well-formed. ctypes releases the GIL during calls, so the C++
    paths genuinely interleave."""
    import threading
    d = 16
    rows = oracle.gen_f32(0xAA, 0, 400, d)
    ix = sa.index_create_host(d, metric="euclidean", m=8, m0=16, efc=40)
    for i, r in enumerate(rows[:200]):
        ix.enqueue(i, None, r)
    ix.apply_pendings()
    stop = threading.Event()
    errors = []

    def searcher():
        try:
            while not stop.is_set():
                k, i, dd = ix.knn_search(rows[7], 10, 30)
                assert len(i) <= 10
                assert np.all(np.diff(dd) >= 0) or len(dd) <= 1
        except Exception as e:  # pragma: no cover
            errors.append(e)

    def writer():
        try:
            for j in range(200, 400):
                ix.enqueue(j, None, rows[j])
                if j % 20 == 0:
                    ix.apply_pendings()
            ix.apply_pendings()
        except Exception as e:  # pragma: no cover
            errors.append(e)

    threads = [threading.Thread(target=searcher) for _ in range(3)]
    wt = threading.Thread(target=writer)
    for t in threads:
        t.start()
    wt.start()
    wt.join(timeout=60)
    stop.set()
    for t in threads:
        t.join(timeout=10)
    assert not errors, errors
    assert ix.doc_count() == 400
    k, i, dd = ix.knn_search(rows[300], 1, 20)
    assert i[0] == 300 and dd[0] == 0.0
    ix.destroy()
