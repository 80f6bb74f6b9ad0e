"""Host-side generator of the committed synthetic-data contract.

Primary path: the product C library's parallel generator (sdbv_gen_f32,
bit-identical to the device staging generator and the oracle).
`gen_f32_numpy` is a pure-numpy restatement kept for cross-checking the
contract in tests (numpy's uint64 ops are too slow for bulk use).
"""
import ctypes

import numpy as np

_C1 = np.uint64(0x9E3779B97F4A7C15)
_C2 = np.uint64(0xBF58476D1CE4E5B9)
_C3 = np.uint64(0x94D049BB133111EB)


def gen_f32(seed, row0, nrows, d):
    from . import lib
    L = lib()
    if not hasattr(L.sdbv_gen_f32, "_typed"):
        L.sdbv_gen_f32.argtypes = [ctypes.c_uint64, ctypes.c_uint64,
                                   ctypes.c_uint64, ctypes.c_uint32,
                                   ctypes.POINTER(ctypes.c_float)]
        L.sdbv_gen_f32._typed = True
    out = np.empty((nrows, d), dtype=np.float32)
    L.sdbv_gen_f32(seed, row0, nrows, d,
                   out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)))
    return out


def _splitmix64(z):
    z = (z + _C1).astype(np.uint64)
    z = ((z ^ (z >> np.uint64(30))) * _C2).astype(np.uint64)
    z = ((z ^ (z >> np.uint64(27))) * _C3).astype(np.uint64)
    return z ^ (z >> np.uint64(31))


def gen_f32_numpy(seed, row0, nrows, d):
    with np.errstate(over="ignore"):
        gidx = (np.uint64(row0) + np.arange(nrows, dtype=np.uint64)[:, None]) \
            * np.uint64(d) + np.arange(d, dtype=np.uint64)[None, :]
        x = _splitmix64(np.uint64(seed) + gidx.ravel())
    u = (x >> np.uint64(11)).astype(np.float64) * (2.0 ** -53)
    return (-20.0 + 40.0 * u).astype(np.float32).reshape(nrows, d)
