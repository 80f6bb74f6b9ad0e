"""UndirectedGraph semantics (graph.rs:43-130) — the reference's own
test_undirected_graph sequence (graph.rs:166-233) replayed verbatim against
the oracle's layer-graph primitives (the same primitives the build and
remove paths use internally)."""
import ctypes

import numpy as np

import oracle


class Graph:
    def __init__(self, m_max=10):
        L = oracle.lib()
        L.orc_test_graph_new.restype = ctypes.c_void_p
        L.orc_test_graph_add_empty_node.restype = ctypes.c_int
        L.orc_test_graph_remove.restype = ctypes.c_int
        L.orc_test_graph_edges.restype = ctypes.c_int
        u64p = ctypes.POINTER(ctypes.c_uint64)
        L.orc_test_graph_add_bidir.argtypes = [ctypes.c_void_p,
                                               ctypes.c_uint64, u64p,
                                               ctypes.c_uint32]
        L.orc_test_graph_set_node.argtypes = [ctypes.c_void_p,
                                              ctypes.c_uint64, u64p,
                                              ctypes.c_uint32]
        L.orc_test_graph_remove.argtypes = [ctypes.c_void_p, ctypes.c_uint64,
                                            u64p, ctypes.c_uint32]
        L.orc_test_graph_edges.argtypes = [ctypes.c_void_p, ctypes.c_uint64,
                                           u64p, ctypes.c_uint32]
        self.L = L
        self.g = ctypes.c_void_p(L.orc_test_graph_new(m_max))

    def _arr(self, vals):
        a = np.asarray(list(vals), dtype=np.uint64)
        return a, a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64))

    def add_empty_node(self, i):
        return bool(self.L.orc_test_graph_add_empty_node(self.g, i))

    def add_bidir(self, i, edges):
        a, p = self._arr(edges)
        self.L.orc_test_graph_add_bidir(self.g, i, p, len(a))

    def set_node(self, i, edges):
        a, p = self._arr(edges)
        self.L.orc_test_graph_set_node(self.g, i, p, len(a))

    def remove(self, i):
        out = np.zeros(64, dtype=np.uint64)
        n = self.L.orc_test_graph_remove(
            self.g, i, out.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            64)
        if n < 0:
            return None
        return sorted(out[:n].tolist())

    def edges(self, i):
        out = np.zeros(64, dtype=np.uint64)
        n = self.L.orc_test_graph_edges(
            self.g, i, out.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            64)
        if n < 0:
            return None
        return sorted(out[:n].tolist())

    def check(self, expected):
        """g.check(vec![(node, edges)]) — node present with exactly these
        edge contents (graph.rs:146-158)."""
        for node, edges in expected:
            got = self.edges(node)
            assert got == sorted(edges), (node, got, edges)

    def free(self):
        self.L.orc_test_graph_free(self.g)


def test_undirected_graph_reference_sequence():
    """graph.rs:166-233, step for step."""
    g = Graph(10)
    assert g.add_empty_node(0) is True
    g.check([(0, [])])
    assert g.add_empty_node(0) is False  # adding the same node
    g.check([(0, [])])
    g.add_bidir(1, [0])
    g.check([(0, [1]), (1, [0])])
    g.add_bidir(2, [0, 1])
    g.check([(0, [1, 2]), (1, [0, 2]), (2, [0, 1])])
    g.add_bidir(3, [1, 2])
    g.check([(0, [1, 2]), (1, [0, 2, 3]), (2, [0, 1, 3]), (3, [1, 2])])
    # change the edges of a node (one-directional — 1 and 2 keep their
    # edge to 3: the insert-time pruning asymmetry source)
    g.set_node(3, [0])
    g.check([(0, [1, 2]), (1, [0, 2, 3]), (2, [0, 1, 3]), (3, [0])])
    # remove node 2: returns its edges; back-edges cleaned; 3's list does
    # NOT contain 2 so it is untouched
    assert g.remove(2) == [0, 1, 3]
    g.check([(0, [1]), (1, [0, 3]), (3, [0])])
    assert g.edges(2) is None  # get_edges -> None
    assert g.remove(2) is None  # remove again
    # set a non-existing node: creates it (one-directional)
    g.set_node(2, [1])
    g.check([(0, [1]), (1, [0, 3]), (2, [1]), (3, [0])])
    g.free()


def test_implicit_node_creation_on_back_edge():
    """graph.rs:52-64: add_node_and_bidirectional_edges creates a MISSING
    edge target (`nodes.entry(e).or_insert_with`) — the behaviour that
    makes upper-layer seeds members of lower layers."""
    g = Graph(10)
    g.add_bidir(5, [9])  # 9 never added explicitly
    g.check([(5, [9]), (9, [5])])
    assert g.remove(9) == [5]
    g.check([(5, [])])
    g.free()
