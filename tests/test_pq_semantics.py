"""DoublePriorityQueue semantics (knn.rs:15-123) — the reference's own
test_double_priority_queue sequence (knn.rs:735-790) replayed VERBATIM
against BOTH restatements (oracle OrcPQ, product hnsw::PQ), including the
FIFO-within-equal-distance pop_first order and the latest-of-max-key
pop_last order."""
import ctypes

import pytest

import oracle
import surrealdb_amd as sa


class PQ:
    def __init__(self, lib, prefix):
        self.lib = lib
        self.p = prefix
        fn = getattr(lib, prefix + "new")
        fn.restype = ctypes.c_void_p
        getattr(lib, prefix + "len").restype = ctypes.c_uint64
        self.q = ctypes.c_void_p(fn())

    def _call(self, name, *argtypes):
        return getattr(self.lib, self.p + name)

    def push(self, d, i):
        f = self._call("push")
        f.argtypes = [ctypes.c_void_p, ctypes.c_double, ctypes.c_uint64]
        f(self.q, d, i)

    def len(self):
        f = self._call("len")
        f.argtypes = [ctypes.c_void_p]
        return f(self.q)

    def _pair(self, name):
        f = self._call(name)
        f.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.c_double),
                      ctypes.POINTER(ctypes.c_uint64)]
        d = ctypes.c_double(0)
        i = ctypes.c_uint64(0)
        if not f(self.q, ctypes.byref(d), ctypes.byref(i)):
            return None
        return (d.value, i.value)

    def peek_first(self):
        return self._pair("peek_first")

    def pop_first(self):
        return self._pair("pop_first")

    def pop_last(self):
        return self._pair("pop_last")

    def peek_last_dist(self):
        f = self._call("peek_last_dist")
        f.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.c_double)]
        d = ctypes.c_double(0)
        if not f(self.q, ctypes.byref(d)):
            return None
        return d.value

    def free(self):
        f = self._call("free")
        f.argtypes = [ctypes.c_void_p]
        f(self.q)


@pytest.fixture(params=["oracle", "product"])
def pq_factory(request):
    if request.param == "oracle":
        return lambda: PQ(oracle.lib(), "orc_test_pq_")
    return lambda: PQ(sa.lib(), "sdbv_test_pq_")


def test_double_priority_queue_reference_sequence(pq_factory):
    """knn.rs:735-790, step for step."""
    q = pq_factory()
    # DoublePriorityQueue::from(2.0, 2)
    q.push(2.0, 2)
    q.push(3.0, 4)
    q.push(3.0, 3)
    q.push(1.0, 1)
    assert q.len() == 4
    assert q.peek_first() == (1.0, 1)
    assert q.peek_last_dist() == 3.0
    assert q.pop_first() == (1.0, 1)
    assert q.len() == 3
    assert q.peek_first() == (2.0, 2)
    assert q.peek_last_dist() == 3.0
    assert q.pop_first() == (2.0, 2)
    assert q.len() == 2
    assert q.peek_first() == (3.0, 4)  # FIFO within the 3.0 group
    assert q.peek_last_dist() == 3.0
    assert q.pop_first() == (3.0, 4)
    assert q.len() == 1
    assert q.peek_first() == (3.0, 3)
    assert q.peek_last_dist() == 3.0
    assert q.pop_first() == (3.0, 3)
    assert q.len() == 0
    assert q.peek_first() is None
    assert q.peek_last_dist() is None
    q.free()

    q = pq_factory()
    q.push(2.0, 2)
    q.push(3.0, 4)
    q.push(3.0, 3)
    q.push(1.0, 1)
    assert q.pop_last() == (3.0, 3)  # LATEST push of the max key
    assert q.len() == 3
    assert q.peek_first() == (1.0, 1)
    assert q.peek_last_dist() == 3.0
    assert q.pop_last() == (3.0, 4)
    assert q.len() == 2
    assert q.peek_first() == (1.0, 1)
    assert q.peek_last_dist() == 2.0
    assert q.pop_last() == (2.0, 2)
    assert q.len() == 1
    assert q.peek_first() == (1.0, 1)
    assert q.peek_last_dist() == 1.0
    assert q.pop_last() == (1.0, 1)
    assert q.len() == 0
    assert q.peek_first() is None
    assert q.peek_last_dist() is None
    q.free()


def test_interleaved_push_pop_fifo(pq_factory):
    """Pushes after pops keep FIFO order within a key (the BTreeMap-of-
    VecDeque contract the sorted-(key, seq) product queue must preserve)."""
    q = pq_factory()
    q.push(5.0, 10)
    q.push(5.0, 11)
    assert q.pop_first() == (5.0, 10)
    q.push(5.0, 12)
    q.push(4.0, 9)
    assert q.pop_first() == (4.0, 9)
    assert q.pop_first() == (5.0, 11)
    q.push(5.0, 13)
    assert q.pop_last() == (5.0, 13)
    assert q.pop_first() == (5.0, 12)
    assert q.pop_first() is None
    q.free()
