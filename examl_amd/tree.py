"""Unrooted binary phylogeny with ExaML's node numbering and the post-order
traversal builder (the computeTraversalInfo role,
examl/newviewGenericSpecial.c:691).

Node ids: tips 1..n, inner nodes n+1..2n-2; CLV slot of inner node p is
p - n - 1 (the xVector indexing of examl/newviewGenericSpecial.c:1221).
"""

import numpy as np

from . import INNER_INNER, TIP_INNER, TIP_TIP, TravEntry

DEFAULTZ = 0.9  # examl/axml.h:94


class PhyloTree:
    def __init__(self, ntips):
        assert ntips >= 4
        self.ntips = ntips
        self.nnodes = 2 * ntips - 1  # ids 1..2n-2 used
        # adjacency: node -> {neighbor: z}
        self.adj = {i: {} for i in range(1, self.nnodes)}

    def is_tip(self, v):
        return v <= self.ntips

    def clv_slot(self, v):
        assert not self.is_tip(v)
        return v - self.ntips - 1

    def add_edge(self, a, b, z):
        self.adj[a][b] = z
        self.adj[b][a] = z

    def del_edge(self, a, b):
        del self.adj[a][b]
        del self.adj[b][a]

    def set_z(self, a, b, z):
        """Scalar set: with per-partition branch lengths (-M) this sets
        ALL partitions' values (treeReadLen semantics, treeIO.c)."""
        cur = self.adj[a][b]
        if isinstance(cur, np.ndarray):
            cur[:] = z
        else:
            self.adj[a][b] = z
            self.adj[b][a] = z

    def get_z(self, a, b):
        z = self.adj[a][b]
        return float(z[0]) if isinstance(z, np.ndarray) else z

    # -- per-partition branch lengths (-M, tr->numBranches > 1) ------------

    def expand_branches(self, nb):
        """Switch every edge to an nb-vector branch length (aliased in both
        adjacency directions, like p->z/p->back->z sharing updates)."""
        for a, b in self.edges():
            z = self.adj[a][b]
            if not isinstance(z, np.ndarray):
                v = np.full(nb, float(z))
                self.adj[a][b] = v
                self.adj[b][a] = v

    def get_zv(self, a, b):
        z = self.adj[a][b]
        return z if isinstance(z, np.ndarray) else np.array([z])

    def edges(self):
        out = []
        for a in self.adj:
            for b in self.adj[a]:
                if a < b:
                    out.append((a, b))
        return out

    @staticmethod
    def random(ntips, seed=7, z=DEFAULTZ, rng_z=False):
        """Random topology by sequential edge splitting (seeded)."""
        rng = np.random.default_rng(seed)

        def draw_z():
            return float(rng.uniform(0.2, 0.98)) if rng_z else z

        t = PhyloTree(ntips)
        inner = ntips + 1
        t.add_edge(1, inner, draw_z())
        t.add_edge(2, inner, draw_z())
        t.add_edge(3, inner, draw_z())
        next_inner = inner + 1
        for tip in range(4, ntips + 1):
            a, b = t.edges()[rng.integers(0, len(t.edges()))]
            zab = t.get_z(a, b)
            m = next_inner
            next_inner += 1
            t.del_edge(a, b)
            t.add_edge(a, m, zab)
            t.add_edge(m, b, draw_z())
            t.add_edge(m, tip, draw_z())
        assert next_inner == 2 * ntips - 1
        return t

    @staticmethod
    def caterpillar(ntips, z=DEFAULTZ):
        """Maximally deep (chain) topology — drives the 2^-256 rescale path
        (CLV magnitudes shrink multiplicatively with depth)."""
        t = PhyloTree(ntips)
        first = ntips + 1
        t.add_edge(1, first, z)
        t.add_edge(2, first, z)
        prev = first
        for k in range(3, ntips):
            m = ntips + k - 1
            t.add_edge(prev, m, z)
            t.add_edge(k, m, z)
            prev = m
        t.add_edge(ntips, prev, z)
        return t

    # -- traversal ----------------------------------------------------------

    def _collect(self, node, parent, out):
        """Post-order entries for the subtree of `node` seen from `parent`
        (full traversal: every inner node emitted, children before parents —
        the partialTraversal=FALSE behavior of computeTraversalInfo)."""
        if self.is_tip(node):
            return
        children = [w for w in self.adj[node] if w != parent]
        assert len(children) == 2
        q, r = children
        # match computeTraversalInfo: if exactly one child is a tip it is
        # stored as q (newviewGenericSpecial.c:742-749)
        if self.is_tip(r) and not self.is_tip(q):
            q, r = r, q
        self._collect(q, node, out)
        self._collect(r, node, out)
        e = TravEntry()
        e.pNumber, e.qNumber, e.rNumber = node, q, r
        e.qz, e.rz = self.get_z(node, q), self.get_z(node, r)
        e.x3Slot = self.clv_slot(node)
        if self.is_tip(q) and self.is_tip(r):
            e.tipCase = TIP_TIP
            e.x1Slot, e.x2Slot = q, r  # tip rows
        elif self.is_tip(q):
            e.tipCase = TIP_INNER
            e.x1Slot = q  # tip row
            e.x2Slot = self.clv_slot(r)
        else:
            e.tipCase = INNER_INNER
            e.x1Slot = self.clv_slot(q)
            e.x2Slot = self.clv_slot(r)
        out.append(e)

    def full_traversal(self, root_edge=None):
        """(entries, (p, q, z)) for a full-tree evaluation at `root_edge`
        (default: the branch at tip 1, like tr->start in the reference)."""
        if root_edge is None:
            p = 1
            q = next(iter(self.adj[1]))
        else:
            p, q = root_edge
        out = []
        self._collect(p, q, out)
        self._collect(q, p, out)
        return out, (p, q, self.get_z(p, q))
