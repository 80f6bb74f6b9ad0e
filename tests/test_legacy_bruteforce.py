"""Legacy two-phase brute-force seam (idx/planner/knn.rs KnnPriorityList +
executor.rs:283-311): the OTHER brute-force implementation next to KnnTopK.

Its observable contract (restated here in python, cited):
- `add` keeps a BTreeMap<dist, set-of-records>, evicting only whole tie
  GROUPS and only while >= knn docs remain (knn.rs:69-79);
- `build` then takes ascending distance groups up to knn docs, choosing an
  UNSPECIFIED subset of the cutoff group (`docs.iter().take(left)` over a
  HashSet — nondeterministic, knn.rs:82-105).

Therefore any result containing (a) every doc strictly below the cutoff
distance, plus (b) exactly `knn - #below` docs AT the cutoff distance, is
reference-conformant. `sdbv_knn_bruteforce` / orc_topk (KnnTopK semantics:
ties broken by earlier row) produce exactly such a result, so ONE boundary
serves both executor seams — pinned by this test."""
import numpy as np

import oracle


def knn_priority_list(dists, knn):
    """Deterministic restatement of KnnPriorityList (knn.rs:43-105) over
    (dist, row) pairs; returns (below_cutoff_set, cutoff_dist, n_at_cutoff)
    — the parts of build() that are deterministic."""
    groups = {}
    docs = 0
    # add with group eviction (knn.rs:54-79)
    order = {}
    for row, d in enumerate(dists):
        if docs < knn or d < max(order):
            groups.setdefault(d, set()).add(row)
            order[d] = None
            docs = sum(len(s) for s in groups.values())
            if docs > knn:
                worst = max(groups)
                if docs - len(groups[worst]) >= knn:
                    del groups[worst]
                    del order[worst]
                    docs = sum(len(s) for s in groups.values())
    # build (knn.rs:82-105): ascending groups, cutoff group truncated
    below = set()
    left = knn
    for d in sorted(groups):
        g = groups[d]
        if len(g) > left:
            return below, d, left
        below |= g
        left -= len(g)
        if left == 0:
            return below, None, 0
    return below, None, 0


def test_strict_k_scan_is_conformant_with_ties():
    d, n, k = 16, 400, 10
    base = oracle.gen_f32(0x71, 0, 40, d)
    # tie-heavy corpus: every vector appears 10x
    corpus = np.ascontiguousarray(np.repeat(base, 10, axis=0))
    for qi in range(8):
        q = base[qi] + np.float32(0.01)
        full = np.array([oracle.dist_f32("euclidean", q, corpus[r])
                         for r in range(n)])
        below, cutoff, n_at_cut = knn_priority_list(full, k)
        ids, dists = oracle.topk_f32("euclidean", corpus, q, k)
        got = set(ids.tolist())
        # (a) everything strictly below the cutoff is present
        assert below <= got, (below - got, cutoff)
        # (b) the remainder sits exactly AT the cutoff distance
        extra = got - below
        assert len(got) == k and len(extra) == n_at_cut
        if cutoff is not None:
            assert all(full[r] == cutoff for r in extra)


def test_exact_agreement_without_ties():
    d, n, k = 24, 500, 10
    corpus = oracle.gen_f32(0x72, 0, n, d)
    for qi in range(5):
        q = oracle.gen_f32(0x73, 0, 5, d)[qi]
        full = np.array([oracle.dist_f32("cosine", q, corpus[r])
                         for r in range(n)])
        below, cutoff, n_at_cut = knn_priority_list(full, k)
        assert cutoff is None and len(below) == k  # unique dists: no group
        ids, _ = oracle.topk_f32("cosine", corpus, q, k)
        assert set(ids.tolist()) == below
