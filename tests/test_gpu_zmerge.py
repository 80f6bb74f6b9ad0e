"""Two-level k_merge tree (round 2): the only scan shape not re-validated
on hardware mid-round (it engages above 4096 merge candidates — the 10M
bench shape). Ordered LAST in the GPU suite so a regression here cannot
hide the foundational gates under -x (the round-1 lesson)."""
import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctx():
    import surrealdb_amd
    c = surrealdb_amd.Context()
    yield c
    c.close()


def test_tree_merge_path_matches_oracle(ctx):
    """The two-level k_merge tree engages when nblocks > 32 and
    nblocks*k > 4096. 98304 rows at k=64 -> 96 scan blocks, 6144
    candidates: exercises level-1 slices (including the ragged last one)
    + the level-2 fold, bit-exact vs the oracle."""
    n, d, k = 98_304, 768, 64
    corpus = oracle.gen_f32(0x5DB1, 0, n, d)
    ctx.stage_corpus(30, corpus, metric="cosine")
    for qi in range(3):
        q = oracle.gen_f32(0xBEEF, qi, 1, d)[0]
        gids, gdists = ctx.knn_bruteforce(30, q, k)
        oids, odists = oracle.topk_f32("cosine", corpus, q, k)
        assert np.array_equal(gids, oids), f"q{qi}: ids"
        assert np.array_equal(gdists, odists), f"q{qi}: dist bits"
    ctx.drop_table(30)


def test_tree_merge_at_bench_scale_shape(ctx):
    """The bench configuration's merge shape (k=10, ~2930 scan blocks at
    3M rows -> 29300 candidates through the tree) vs the oracle on the
    same synthetic corpus — the driver-visible correctness check for the
    exact shape the 10M scan line runs."""
    n, d, k = 3_000_000, 768, 10
    ctx.stage_synthetic(31, n, d, metric="cosine", seed=0x5DB1)
    corpus = oracle.gen_f32(0x5DB1, 0, n, d)
    q = oracle.gen_f32(0xBEEF, 0, 1, d)[0]
    gids, gdists = ctx.knn_bruteforce(31, q, k)
    oids, odists, _ = oracle.topk_f32_mt("cosine", corpus, q, k)
    assert np.array_equal(gids, oids)
    assert np.array_equal(gdists, odists)
    ctx.drop_table(31)
