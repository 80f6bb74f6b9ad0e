"""Vectorised host-side generator of the committed synthetic-data contract.

Bit-identical to the device generator (csrc/sdbv.hip k_gen_cm) and the oracle
(orc_gen_f32): element(seed, row*d+j) = splitmix64 -> f64 [0,1) -> [-20,20) f32.
Product-side (bench/input prep) — no oracle dependency.
"""
import numpy as np

_C1 = np.uint64(0x9E3779B97F4A7C15)
_C2 = np.uint64(0xBF58476D1CE4E5B9)
_C3 = np.uint64(0x94D049BB133111EB)


def _splitmix64(z):
    z = (z + _C1).astype(np.uint64)
    z = ((z ^ (z >> np.uint64(30))) * _C2).astype(np.uint64)
    z = ((z ^ (z >> np.uint64(27))) * _C3).astype(np.uint64)
    return z ^ (z >> np.uint64(31))


def gen_f32(seed, row0, nrows, d):
    with np.errstate(over="ignore"):
        gidx = (np.uint64(row0) + np.arange(nrows, dtype=np.uint64)[:, None] ) * np.uint64(d) \
            + np.arange(d, dtype=np.uint64)[None, :]
        x = _splitmix64(np.uint64(seed) + gidx.ravel())
    u = (x >> np.uint64(11)).astype(np.float64) * (2.0 ** -53)
    return (-20.0 + 40.0 * u).astype(np.float32).reshape(nrows, d)
