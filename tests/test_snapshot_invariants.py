"""Structural invariants of the batched snapshot build (v3 host twin)
across randomized configurations — the check_hnsw_props contract
(hnsw/mod.rs:561-570: degree <= m_max, no self-edges) plus membership
consistency, determinism, and a recall floor vs brute force. The GPU
build is bit-identical to this twin (tests/test_gpu_insert.py), so these
invariants transfer."""
import numpy as np
import pytest

import oracle
import surrealdb_amd as sa

CASES = [
    # (seed, n, d, metric, m, m0, efc, keep, chunk)
    (0x11, 1200, 16, "euclidean", 8, 16, 60, False, 128),
    (0x22, 900, 20, "cosine", 8, 16, 100, False, 64),
    (0x33, 1500, 32, "cosine", 12, 24, 80, True, 256),
    (0x44, 700, 24, "euclidean", 4, 8, 40, False, 96),
    (0x55, 2000, 16, "cosine", 16, 32, 120, False, 512),
    (0x66, 1100, 8, "euclidean", 6, 12, 50, True, 100),
]


@pytest.mark.parametrize("seed,n,d,metric,m,m0,efc,keep,chunk", CASES)
def test_snapshot2_structural_invariants(seed, n, d, metric, m, m0, efc,
                                         keep, chunk):
    # nthreads=1: the strict check_hnsw_props degree bound only holds for
    # deterministic schedules — parallel applies can leave a node
    # transiently above m_max (the documented keep-back relaxation, the
    # reason finalize sizes its scratch from the ACTUAL max degree)
    rows = oracle.gen_f32(seed, 0, n, d)
    h = sa.hnsw_create_host(d, metric=metric, m=m, m0=m0, efc=efc,
                            keep=keep, seed=seed)
    h.insert_batch_snapshot2(rows, chunk=chunk, nthreads=1)
    nl = h.num_layers()
    ep = h.enter_point()
    assert 0 <= ep < n
    pops = []
    for l in range(nl):
        offsets, edges, in_layer = h.layer_csr(l)
        deg = np.diff(offsets.astype(np.int64))
        m_max = m0 if l == 0 else m
        # check_hnsw_props: degree cap and no self-edges
        assert deg.max() <= m_max, (l, int(deg.max()))
        for i in np.nonzero(deg > 0)[0][:200]:
            es = edges[offsets[i]:offsets[i + 1]]
            assert i not in es, f"self-edge at layer {l} node {i}"
            assert len(set(es.tolist())) == len(es), "duplicate edge"
        # nodes with edges must be members; members nest downward
        assert np.all(in_layer[np.nonzero(deg > 0)[0]] == 1)
        pops.append(int(in_layer.sum()))
        if l > 0:
            _, _, below = h.layer_csr(l - 1)
            mem = np.nonzero(in_layer)[0]
            assert np.all(below[mem] == 1), f"layer {l} member not below"
    # layer populations decay roughly geometrically
    assert pops[0] == n
    for l in range(1, nl):
        assert pops[l] <= pops[l - 1]
    # recall floor vs brute force (quality contract at chunk << n)
    if chunk * 10 <= n:
        queries = oracle.gen_f32(seed ^ 0xBEEF, 0, 30, d)
        tot = 0.0
        for q in queries:
            ids, _ = h.knn_search_host(q, 10, 40)
            bf, _ = oracle.topk_f32(metric, rows, q, 10)
            tot += len(set(ids.tolist()) & set(bf.tolist())) / 10.0
        # uniform random high-dim data has an intrinsic ceiling (DESIGN
        # recall note); d <= 32 here keeps it high
        assert tot / len(queries) >= 0.85, tot / len(queries)
    h.destroy()
