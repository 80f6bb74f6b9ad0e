"""Oracle vs the reference's own exact-constant tests.

Constants are copied VERBATIM from the reference test suite
(surrealdb/core/src/idx/trees/vector.rs:723-772, test_distance): the reference
asserts both Distance::compute (Number path) and Distance::calculate (typed
F64) equal these exact f64 values. Our oracle must reproduce them bit-exactly.
"""
import numpy as np
import pytest

import oracle

A = np.array([1.0, 2.0, 3.0])
B = np.array([2.0, 3.0, 4.0])

# (metric, order, expected) — vector.rs test constants
CASES = [
    ("chebyshev", 0.0, 1.0),                      # vector.rs:726
    ("cosine", 0.0, 0.007416666029069652),        # vector.rs:732
    ("euclidean", 0.0, 1.7320508075688772),       # vector.rs:738
    ("hamming", 0.0, 3.0),                        # vector.rs:744
    ("jaccard", 0.0, 0.5),                        # vector.rs:750
    ("manhattan", 0.0, 3.0),                      # vector.rs:755
    ("minkowski", 3.0, 1.4422495703074083),       # vector.rs:764
    ("pearson", 0.0, 1.0),                        # vector.rs:771
]


@pytest.mark.parametrize("metric,order,expected", CASES)
def test_typed_f64_exact(metric, order, expected):
    got = oracle.dist_f64(metric, A, B, order=order)
    assert got == expected, f"{metric}: {got!r} != {expected!r}"


@pytest.mark.parametrize("metric,order,expected", CASES)
def test_number_path_exact(metric, order, expected):
    got = oracle.dist_number(metric, A, B, order=order)
    assert got == expected, f"{metric}: {got!r} != {expected!r}"


@pytest.mark.parametrize("metric,order,expected", CASES)
def test_typed_f32_close(metric, order, expected):
    # F32 typed path on the same small integers (exactly f32-representable):
    # identical except rounding inside f32 accumulation — must agree to 1e-6.
    got = oracle.dist_f32(metric, A.astype(np.float32), B.astype(np.float32),
                          order=order)
    assert got == pytest.approx(expected, rel=1e-6, abs=1e-9)


def test_distance_collection_properties():
    """Restates test_distance_collection (vector.rs:697-721): seeded sweeps
    produce finite, non-NaN distances with a low zero-rate."""
    rng = np.random.default_rng(42)
    for metric in ["chebyshev", "cosine", "euclidean", "hamming", "manhattan",
                   "pearson"]:
        num_zero = 0
        size = 30
        for i in range(size):
            v1 = rng.uniform(-20, 20, 256).astype(np.float32)
            v2 = rng.uniform(-20, 20, 256).astype(np.float32)
            d = oracle.dist_f32(metric, v1, v2)
            assert np.isfinite(d), f"{metric} i={i}"
            if d == 0.0:
                num_zero += 1
        assert num_zero / size < 0.1, metric


def test_number_vs_typed_f64_agree():
    """vector.rs:688-694 asserts compute == calculate for F64 inputs."""
    rng = np.random.default_rng(7)
    for metric, order in [("cosine", 0.0), ("euclidean", 0.0),
                          ("manhattan", 0.0), ("chebyshev", 0.0),
                          ("minkowski", 3.0), ("pearson", 0.0)]:
        a = rng.uniform(-20, 20, 768)
        b = rng.uniform(-20, 20, 768)
        t = oracle.dist_f64(metric, a, b, order=order)
        n = oracle.dist_number(metric, a, b, order=order)
        # typed F64 uses unrolled-8 accumulation, Number path is sequential:
        # equal up to last-ulp differences
        assert n == pytest.approx(t, rel=1e-12), (metric, t, n)


def test_jaccard_f64_f32_asymmetry():
    """The reference computes jaccard as 1 - |I|/|U| on F64 (vector.rs:326)
    but |I|/|U| on F32 (vector.rs:339). Restated faithfully."""
    a = np.array([1.0, 2.0, 3.0])
    b = np.array([2.0, 3.0, 4.0])
    s64 = oracle.dist_f64("jaccard", a, b)
    s32 = oracle.dist_f32("jaccard", a.astype(np.float32), b.astype(np.float32))
    assert s64 == 0.5  # 1 - 2/4
    assert s32 == 0.5  # 2/4  (equal only by coincidence on this input)
    a2 = np.array([1.0, 2.0, 3.0, 5.0])
    b2 = np.array([2.0, 3.0, 4.0, 5.0])
    assert oracle.dist_f64("jaccard", a2, b2) == 1.0 - 3.0 / 5.0
    assert oracle.dist_f32("jaccard", a2.astype(np.float32),
                           b2.astype(np.float32)) == 3.0 / 5.0


def test_dot_unrolled_contract():
    """The restated ndarray unrolled_dot must differ from naive sequential
    summation only in accumulation order (same value in f64 on exact ints),
    and must be deterministic."""
    rng = np.random.default_rng(3)
    a = rng.integers(-10, 10, 64).astype(np.float32)
    b = rng.integers(-10, 10, 64).astype(np.float32)
    # exact integers: any order gives the exact value
    assert oracle.lib().orc_dot_f32(
        a.ctypes.data_as(__import__("ctypes").POINTER(__import__("ctypes").c_float)),
        b.ctypes.data_as(__import__("ctypes").POINTER(__import__("ctypes").c_float)),
        64) == float(np.dot(a.astype(np.float64), b.astype(np.float64)))


def test_dot_chain_bitexact_vs_strict_simulation():
    """Pins the oracle's fp behaviour against a strict per-op np.float32
    simulation of the restated ndarray chain — catches any build-flag change
    (e.g. fp contraction) that would silently fuse mul+add into fma."""
    import ctypes
    import math
    a = oracle.gen_f32(0x5DB1, 0, 1, 768)[0]
    b = oracle.gen_f32(0xBEEF, 0, 1, 768)[0]
    p = np.zeros(8, np.float32)
    for i in range(0, 768, 8):
        for j in range(8):
            p[j] = np.float32(p[j] + np.float32(a[i + j] * b[i + j]))
    s = np.float32(0)
    s = np.float32(s + np.float32(p[0] + p[4]))
    s = np.float32(s + np.float32(p[1] + p[5]))
    s = np.float32(s + np.float32(p[2] + p[6]))
    s = np.float32(s + np.float32(p[3] + p[7]))
    f32p = ctypes.POINTER(ctypes.c_float)
    got = oracle.lib().orc_dot_f32(a.ctypes.data_as(f32p),
                                   b.ctypes.data_as(f32p), 768)
    assert np.float32(got).tobytes() == s.tobytes()
    acc = np.float32(0)
    for i in range(768):
        d = np.float32(a[i] - b[i])
        acc = np.float32(acc + np.float32(d * d))
    assert math.sqrt(float(acc)) == oracle.dist_f32("euclidean", a, b)
