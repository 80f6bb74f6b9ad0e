"""HNSW oracle vs the reference's own golden datasets and recall bars.

Restates test_recall_euclidean (hnsw/mod.rs:1144-1156): ingest from
hnsw-random-9000-20-euclidean.gz, query from hnsw-random-5000-20-euclidean.gz,
M=8, M0=16, EFC=100, euclidean F32; recall >= 0.98 at efs=10 and == 1.0 at
efs=40 with EXACT result-set equality vs brute force (the reference asserts
set equality whenever recall == 1.0, hnsw/mod.rs:1121-1123).
"""
import gzip
import json
import math
import os

import numpy as np
import pytest

import oracle

GOLDEN_DIR = os.path.join(os.path.dirname(__file__), "golden")


def load_golden(name, limit):
    rows = []
    with gzip.open(os.path.join(GOLDEN_DIR, name), "rt") as f:
        for i, line in enumerate(f):
            if i >= limit:
                break
            rows.append(json.loads(line))
    # loaded as F32 vectors (the reference test uses VectorType::F32)
    return np.array(rows, dtype=np.float32)


@pytest.fixture(scope="module")
def golden_index():
    # The reference's OWN scale: test_recall_euclidean passes ingest
    # limit 1000 and query limit 300 (hnsw/mod.rs:1146-1152 —
    # `test_recall("hnsw-random-9000-20-euclidean.gz", 1000,
    # "hnsw-random-5000-20-euclidean.gz", 300, p, &[(10,0.98),(40,1.0)])`).
    # The recall bars are defined at THESE limits only: at the full 9000
    # rows recall@efs10 drops below 0.98 for this algorithm regardless of
    # implementation (measured 0.959 on the bit-exact restatement) — see
    # test_recall_full_files_floor below, which pins the full-file
    # behaviour without inventing a bar the reference never states.
    ingest = load_golden("hnsw-random-9000-20-euclidean.gz", 1000)
    queries = load_golden("hnsw-random-5000-20-euclidean.gz", 300)
    h = oracle.Hnsw(20, metric="euclidean", m=8, m0=16, efc=100,
                    ml=1.0 / math.log(8.0), seed=0x5DB1)
    for row in ingest:
        h.insert(row)
    return h, ingest, queries


def test_props(golden_index):
    h, ingest, _ = golden_index
    assert h.check_props() == 0


def test_recall_efs10(golden_index):
    h, ingest, queries = golden_index
    k = 10
    total = 0.0
    for q in queries:
        ids, _ = h.search(q, k, 10)
        bids, _ = oracle.topk_f32("euclidean", ingest, q, k)
        total += len(set(ids.tolist()) & set(bids.tolist())) / k
    recall = total / len(queries)
    assert recall >= 0.98, recall


def test_recall_efs40_exact(golden_index):
    h, ingest, queries = golden_index
    k = 10
    for qi, q in enumerate(queries):
        ids, dists = h.search(q, k, 40)
        bids, bdists = oracle.topk_f32("euclidean", ingest, q, k)
        assert set(ids.tolist()) == set(bids.tolist()), f"query {qi}"
        # result ordering contract: ascending (total_cmp dist, id)
        assert np.array_equal(np.sort(ids), np.sort(bids))
        assert np.allclose(np.sort(dists), np.sort(bdists), rtol=1e-12)


def test_heuristic_variants_recall():
    """Restates test_recall_euclidean_keep_pruned_connections (ingest 750 /
    queries 200) and test_recall_euclidean_full (500 / 100) at the
    reference's own per-variant limits (hnsw/mod.rs:1158-1184)."""
    for extend, keep, n_ing, n_q in [(False, True, 750, 200),
                                     (True, True, 500, 100)]:
        ingest = load_golden("hnsw-random-9000-20-euclidean.gz", n_ing)
        queries = load_golden("hnsw-random-5000-20-euclidean.gz", n_q)
        h = oracle.Hnsw(20, metric="euclidean", m=8, m0=16, efc=100,
                        extend=extend, keep=keep, ml=1.0 / math.log(8.0))
        for row in ingest:
            h.insert(row)
        assert h.check_props() == 0
        k, total = 10, 0.0
        for q in queries:
            ids, _ = h.search(q, k, 40)
            bids, _ = oracle.topk_f32("euclidean", ingest, q, k)
            total += len(set(ids.tolist()) & set(bids.tolist())) / k
        assert total / len(queries) == 1.0, (extend, keep)


def test_recall_full_files_floor():
    """The FULL golden files (9000 ingest / 5000 queries — a scale the
    reference's own tests never run): no reference bar exists here, so pin
    a floor on the measured behaviour of the bit-exact restatement
    (recall@10 efs=10 was 0.9593 when this was written) to catch build or
    search regressions at scale."""
    ingest = load_golden("hnsw-random-9000-20-euclidean.gz", 9000)
    queries = load_golden("hnsw-random-5000-20-euclidean.gz", 5000)
    h = oracle.Hnsw(20, metric="euclidean", m=8, m0=16, efc=100,
                    ml=1.0 / math.log(8.0), seed=0x5DB1)
    for row in ingest:
        h.insert(row)
    assert h.check_props() == 0
    k, total = 10, 0.0
    for q in queries:
        ids, _ = h.search(q, k, 10)
        bids, _ = oracle.topk_f32("euclidean", ingest, q, k)
        total += len(set(ids.tolist()) & set(bids.tolist())) / k
    recall = total / len(queries)
    assert recall >= 0.95, recall


def test_upper_layer_structure(golden_index):
    h, _, _ = golden_index
    assert h.num_layers() >= 2  # 1000 elements at ml=1/ln(8) -> upper layers
    offsets, edges = h.l0_csr()
    assert offsets[-1] == len(edges)
    n = len(offsets) - 1
    deg = np.diff(offsets.astype(np.int64))
    assert deg.max() <= 16  # m0
    assert (edges < n).all()
