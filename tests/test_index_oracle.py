"""Oracle index layer (hnsw/index.rs + docs.rs + knn.rs) semantics, pinned
against the reference's own tests where it has them (test_ids, the
insert/find/delete collection pattern) and against hand-derived consequences
of the reference code for the pendings overlay (each cited by file:line).

CPU-only: the oracle index searches through its own restated graph."""
import math

import numpy as np
import pytest

import oracle


def test_ids64_reference_sequence():
    """knn.rs:669-717 test_ids restated verbatim: variant transitions
    (Some/None) and contents/order at every step."""
    ids = oracle.Ids64()
    expect = []
    for d in (10, 20, 30, 40, 50, 60, 70, 80):
        assert ids.insert(d) is True  # Empty->One->Vec2..Vec8: all Some
        expect.append(d)
        assert ids.export() == (expect, False)
    assert ids.insert(90) is True  # Vec8 -> Bits
    assert ids.export() == ([10, 20, 30, 40, 50, 60, 70, 80, 90], True)
    assert ids.insert(100) is False  # Bits: in-place, None
    assert ids.export() == ([10, 20, 30, 40, 50, 60, 70, 80, 90, 100], True)
    assert ids.remove(10) is False  # Bits in-place (len 9 != 8): None
    assert ids.export() == ([20, 30, 40, 50, 60, 70, 80, 90, 100], True)
    assert ids.remove(20) is True  # Bits -> Vec8 (ascending)
    assert ids.export() == ([30, 40, 50, 60, 70, 80, 90, 100], False)
    for d, rest in ((30, [40, 50, 60, 70, 80, 90, 100]),
                    (40, [50, 60, 70, 80, 90, 100]),
                    (50, [60, 70, 80, 90, 100]),
                    (60, [70, 80, 90, 100]),
                    (70, [80, 90, 100]),
                    (80, [90, 100]),
                    (90, [100]),
                    (100, [])):
        assert ids.remove(d) is True
        assert ids.export() == (rest, False)


def test_ids64_insertion_order_and_quirks():
    # Vec* variants keep INSERTION order (knn.rs:203-216 iter)
    ids = oracle.Ids64()
    for d in (7, 3, 9, 1):
        ids.insert(d)
    assert ids.export() == ([7, 3, 9, 1], False)
    # duplicate insert: None, unchanged (knn.rs:234-236)
    assert ids.insert(3) is False
    assert ids.export() == ([7, 3, 9, 1], False)
    # Vec2 non-member removal: `find(|i| i != d).map(One)` drops the second
    # element (knn.rs:266-268 — reference behaviour, restated as-is)
    v2 = oracle.Ids64()
    v2.insert(5)
    v2.insert(6)
    assert v2.remove(99) is True
    assert v2.export() == ([5], False)
    # Vec3+ non-member removal: no variant, unchanged (knn.rs:269-277)
    v3 = oracle.Ids64()
    for d in (1, 2, 3):
        v3.insert(d)
    assert v3.remove(99) is False
    assert v3.export() == ([1, 2, 3], False)


@pytest.mark.parametrize("unique", [True, False])
def test_insert_find_delete_collection(unique):
    """The reference's index test pattern (hnsw/mod.rs:793-880):
    index() -> index_pendings() -> check_hnsw_properties per doc on insert;
    search finds each doc; then delete each doc the same way down to empty."""
    d, n = 20, 120
    rows = oracle.gen_f32(0xAB, 0, n, d)
    if not unique:
        rows[n // 2:] = rows[:n - n // 2]  # duplicate vectors across docs
    ix = oracle.Index(d, metric="euclidean", m=8, m0=16, efc=60, seed=0xC)
    vec_docs = {}
    for i, r in enumerate(rows):
        ix.enqueue(1000 + i, None, r)
        assert ix.apply_pendings() == 1
        vec_docs.setdefault(r.tobytes(), set()).add(i)
        assert ix.check_props(len(vec_docs)) == 0
    assert ix.doc_count() == n
    # find: knn=1..min(20,n) result counts == min(knn, n_docs-ish); with
    # duplicates each element expands to all its docs, so exact counts hold
    # at doc granularity (find_collection_hnsw_index, mod.rs:816-857)
    for i in (0, 3, n // 2, n - 1):
        kinds, ids, dists = ix.knn_search(rows[i], 10, 500)
        assert len(ids) == 10
        assert (kinds == 0).all()
        # the doc itself is among the zero-distance results
        zero = set(ids[dists == 0.0].tolist())
        # doc ids were allocated sequentially 0..n-1 in enqueue order
        assert i in zero or (not unique and (i + n // 2) % n in zero) or \
            (not unique and (i - n // 2) % n in zero)
    # delete one by one (delete_hnsw_index_collection, mod.rs:859-880)
    for i, r in enumerate(rows):
        ix.enqueue(1000 + i, r, None)
        assert ix.apply_pendings() == 1
        docs = vec_docs[r.tobytes()]
        docs.discard(i)
        if not docs:
            del vec_docs[r.tobytes()]
        assert ix.check_props(len(vec_docs)) == 0, (i, len(vec_docs))
    assert ix.doc_count() == 0
    kinds, ids, dists = ix.knn_search(rows[0], 5, 50)
    assert len(ids) == 0


def test_pendings_overlay_semantics():
    """index.rs:366-421 search_pendings + :340-364 search_graph: outstanding
    pendings are searched directly; DocId pendings exclude their graph
    elements from the expansion frontier only (layer.rs:209-212 pushes to w
    outside the check — restated as-is, so the stale graph entry of an
    updated doc still surfaces until apply)."""
    d = 16
    rows = oracle.gen_f32(0xE1, 0, 40, d)
    ix = oracle.Index(d, metric="euclidean", m=8, m0=16, efc=50, seed=3)
    for i, r in enumerate(rows):
        ix.enqueue(i, None, r)
    # nothing applied: results come purely from pendings, kind=RecordKey
    kinds, ids, dists = ix.knn_search(rows[7], 3, 20)
    assert (kinds == 1).all() and ids[0] == 7 and dists[0] == 0.0
    ix.apply_pendings()
    # update doc 7 while pending: pendings contribute the NEW vector at the
    # recorded DocId; the graph still contributes the OLD vector (stale
    # entry via w) until apply_pendings
    newv = rows[7] + 2.0
    ix.enqueue(7, rows[7], newv)
    kinds, ids, dists = ix.knn_search(newv, 4, 20)
    assert kinds[0] == 0 and ids[0] == 7 and dists[0] == 0.0
    stale = [(i, dv) for i, (dd, dv) in enumerate(zip(ids, dists))
             if dd == 7 and dv > 0.0]
    assert stale, "stale graph entry for the pending-updated doc must " \
        "surface (layer.rs:209-212 w.push outside the pending check)"
    assert math.isclose(stale[0][1], math.sqrt(d * 4.0), rel_tol=1e-6)
    # latest pending wins (non_deleted insert overwrite, index.rs:398-403)
    newer = rows[7] - 1.0
    ix.enqueue(7, newv, newer)
    kinds, ids, dists = ix.knn_search(newer, 2, 20)
    assert ids[0] == 7 and dists[0] == 0.0
    kinds, ids, dists = ix.knn_search(newv, 2, 20)
    assert not ((ids == 7) & (dists == 0.0)).any()
    # a pure-delete pending removes the doc from the pendings overlay but
    # the graph entry still surfaces via w until apply (same quirk)
    ix.apply_pendings()
    ix.enqueue(3, rows[3], None)
    kinds, ids, dists = ix.knn_search(rows[3], 3, 20)
    assert ids[0] == 3 and dists[0] == 0.0  # stale graph entry
    ix.apply_pendings()
    kinds, ids, dists = ix.knn_search(rows[3], 3, 20)
    assert not ((ids == 3) & (dists == 0.0)).any()


def test_record_key_resolution_and_recycling():
    """docs.rs:64-90 resolve + :113-135 remove: doc ids allocate
    sequentially, deletes recycle, the smallest recycled id is reused."""
    d = 8
    rows = oracle.gen_f32(0x5, 0, 6, d)
    ix = oracle.Index(d, metric="euclidean", m=4, m0=8, efc=20, seed=1)
    for i in range(4):
        ix.enqueue(500 + i, None, rows[i])
    ix.apply_pendings()
    # delete docs 1 and 2 (keys 501, 502)
    ix.enqueue(501, rows[1], None)
    ix.enqueue(502, rows[2], None)
    ix.apply_pendings()
    assert ix.doc_count() == 2
    # next two inserts reuse doc ids 1 then 2
    ix.enqueue(900, None, rows[4])
    ix.enqueue(901, None, rows[5])
    ix.apply_pendings()
    kinds, ids, dists = ix.knn_search(rows[4], 1, 20)
    assert (kinds[0], ids[0], dists[0]) == (0, 1, 0.0)
    kinds, ids, dists = ix.knn_search(rows[5], 1, 20)
    assert (kinds[0], ids[0], dists[0]) == (0, 2, 0.0)
    # re-keying: enqueue for an existing key resolves to its DocId at
    # enqueue time (index.rs:158-163)
    ix.enqueue(900, rows[4], rows[0] * 0.5)
    assert ix.pending_count() == 1
    kinds, ids, dists = ix.knn_search(rows[0] * 0.5, 1, 20)
    assert (kinds[0], ids[0]) == (0, 1)  # DocId kind: key was resolved


def test_shared_vector_doc_expansion_and_bits_drop():
    """docs.rs VecDocs: docs sharing a vector share one graph element; the
    element expands to ALL its docs in results (add_graph_result). At >8
    docs the Ids64 goes Bits and further adds/removes are dropped by the
    caller-persists contract (docs.rs:376-381, :437-447 — restated as-is)."""
    d = 8
    v = oracle.gen_f32(0x77, 0, 2, d)
    ix = oracle.Index(d, metric="euclidean", m=4, m0=8, efc=20, seed=2)
    for i in range(12):
        ix.enqueue(i, None, v[0])
    ix.enqueue(50, None, v[1])
    ix.apply_pendings()
    assert ix.check_props(2) == 0  # 12+1 docs, but only 2 graph elements
    kinds, ids, dists = ix.knn_search(v[0], 20, 20)
    zero_docs = sorted(ids[dists == 0.0].tolist())
    # docs 0..8 persisted (Vec8 -> Bits at the 9th), 9..11 dropped
    assert zero_docs == list(range(9))
    # removing one doc from the 9-doc Bits set drops it to exactly 8 ->
    # Some(Vec8), persisted (knn.rs:314-324). Through this API a Bits set
    # can never exceed 9 docs (the 10th+ adds are dropped above), so the
    # unpersisted-Bits-removal branch (len != 8 -> None) is unreachable via
    # VecDocs — it is pinned directly in test_ids64_reference_sequence.
    ix.enqueue(0, v[0], None)
    ix.apply_pendings()
    kinds, ids, dists = ix.knn_search(v[0], 20, 20)
    zero_docs = sorted(ids[dists == 0.0].tolist())
    assert zero_docs == list(range(1, 9))
    # and the set being Vec8 again, a new doc CAN be added (-> Bits, 9
    # docs); the new key recycles the freed doc id 0 (docs.rs:78-90)
    ix.enqueue(60, None, v[0])
    ix.apply_pendings()
    kinds, ids, dists = ix.knn_search(v[0], 20, 20)
    zero_docs = sorted(ids[dists == 0.0].tolist())
    assert zero_docs == list(range(9))


def test_graph_remove_repairs_and_recall():
    """hnsw/mod.rs:398-455 remove + layer.rs:408-460 neighbour repair:
    after deleting half the corpus through the index, properties hold and
    efs=40 search still finds the brute-force top-10 of the remainder."""
    d, n = 20, 400
    rows = oracle.gen_f32(0xDE1, 0, n, d)
    ix = oracle.Index(d, metric="euclidean", m=8, m0=16, efc=100, seed=9)
    for i, r in enumerate(rows):
        ix.enqueue(i, None, r)
    ix.apply_pendings()
    for i in range(0, n, 2):
        ix.enqueue(i, rows[i], None)
    ix.apply_pendings()
    assert ix.check_props(n // 2) == 0
    remaining = np.ascontiguousarray(rows[1::2])
    hits = 0.0
    queries = oracle.gen_f32(0xBEEF, 0, 40, d)
    for q in queries:
        kinds, ids, dists = ix.knn_search(q, 10, 40)
        assert (np.asarray(ids) % 2 == 1).all(), "deleted docs must not appear"
        bf, _ = oracle.topk_f32("euclidean", remaining, q, 10)
        bf_docs = set((np.asarray(bf) * 2 + 1).tolist())
        hits += len(set(ids.tolist()) & bf_docs) / 10.0
    assert hits / len(queries) >= 0.95, hits / len(queries)
    # enter-point deletion: remove elements until empty, properties hold
    ix2 = oracle.Index(d, metric="euclidean", m=4, m0=8, efc=30, seed=4)
    for i in range(10):
        ix2.enqueue(i, None, rows[i])
    ix2.apply_pendings()
    for i in range(10):
        ix2.enqueue(i, rows[i], None)
        ix2.apply_pendings()
        assert ix2.check_props(9 - i) == 0
    kinds, ids, dists = ix2.knn_search(rows[0], 3, 10)
    assert len(ids) == 0
