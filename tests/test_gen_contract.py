"""Pins the synthetic-data contract (oracle orc_gen_f32 == device k_gen_cm).

tests/golden/gen_v1.npz holds committed generator output; any change to the
generator breaks this test, which would silently invalidate GPU<->oracle
parity fixtures across machines.
"""
import os

import numpy as np

import oracle

GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "gen_v1.npz")


def test_gen_matches_committed_golden():
    g = np.load(GOLDEN)
    got = oracle.gen_f32(0x5DB1, 0, 8, 16)
    assert got.dtype == np.float32
    assert np.array_equal(got, g["a"])
    got2 = oracle.gen_f32(0x5DB1, 1000000, 4, 768)
    assert np.array_equal(got2, g["b"])


def test_gen_range_and_no_zero_rows():
    v = oracle.gen_f32(0x5DB1, 0, 1000, 64)
    assert v.min() >= -20.0 and v.max() < 20.0
    # all-zero rows are rejected by the reference (knn.rs:538-543); our
    # generator cannot produce one — assert so
    assert (np.abs(v).sum(axis=1) > 0).all()


def test_gen_row_offset_consistency():
    # shard-generation must be window-consistent: rows [100,110) generated
    # directly equal rows [100,110) of a larger generation
    big = oracle.gen_f32(7, 0, 200, 32)
    win = oracle.gen_f32(7, 100, 10, 32)
    assert np.array_equal(big[100:110], win)


def test_synth_numpy_matches_oracle():
    """The product-side numpy generator (surrealdb_amd.synth) must be
    bit-identical to the oracle C generator."""
    from surrealdb_amd.synth import gen_f32 as np_gen, gen_f32_numpy
    a = oracle.gen_f32(0x5DB1, 999_983, 64, 768)
    b = np_gen(0x5DB1, 999_983, 64, 768)          # product C generator
    c = gen_f32_numpy(0x5DB1, 999_983, 64, 768)   # numpy restatement
    assert np.array_equal(a, b)
    assert np.array_equal(a, c)
