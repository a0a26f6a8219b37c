"""The SPR tree-search driver (-f d, BIG_RAPID_MODE): computeBIGRAPID and
its machinery (searchAlgo.c) restated over the engine layer — the primary
CALLER of the conditional-likelihood hot path.

The reference represents each inner node as a ring of three `node`
structs whose `next` pointers are fixed at setup and whose `back`
pointers are rewired by SPR surgery (hookup).  SprTree reproduces that
exactly: RingNode objects with static rings, surgery touches only backs.
TreeSearch drives the same engines over SprTree through the PhyloTree
interface plus ring-ordered children.

Restated pieces (file:line in searchAlgo.c unless noted):
  removeNodeBIG:440  insertBIG:482  insertRestoreBIG:574
  restoreTopologyOnly:610  testInsertBIG:683  addTraverseBIG:784
  rearrangeBIG:804  treeOptimizeRapid:914  testInsertRestoreBIG:1037
  restoreTreeFast:1095  localSmooth:280  nodeRectifier (trash.c:54)
  determineRearrangementSetting:1752  computeBIGRAPID:1914
  infoList:318-380  bestlist (topologies.c:187-700)
"""

import math

import numpy as np

from . import INNER_INNER, TIP_INNER, TIP_TIP, TravEntry
from .search import SMOOTHINGS, TreeSearch
from .tree import DEFAULTZ

# checkpoint states for -R resume (axml.h:655-657)
CKP_REARR_SETTING, CKP_FAST_SPRS, CKP_SLOW_SPRS = 1, 2, 3

UNLIKELY = -1.0e300
ZMIN, ZMAX = 1.0e-15, 1.0 - 1.0e-6
ITERATIONS = 10  # axml.h:90, NR iterations per insertion's makenewz


def _cdiv(a, b):
    """C floating division semantics for lhAVG/lhDEC with lhDEC==0
    (the reference divides by a zero int: NaN/inf, never a trap) — a NaN
    or +inf cutoff simply disables the cutoff comparison."""
    if b:
        return a / b
    if a == 0.0:
        return math.nan
    return math.inf if a > 0 else -math.inf


class RingNode:
    __slots__ = ("number", "next", "back", "z")

    def __init__(self, number):
        self.number = number
        self.next = self
        self.back = None
        self.z = DEFAULTZ

    def __repr__(self):
        return f"RN({self.number})"


def hookup(p, q, z):
    """hookup (axml.c:478): under -M z is a per-partition vector and the
    reference copies it into both nodes — each edge owns its storage."""
    p.back = q
    q.back = p
    if isinstance(z, np.ndarray):
        z = z.copy()
    p.z = q.z = z


def _branch_split(zqr, zqs, zrs):
    """The three-way branch-length split of insertBIG (:512-530)."""
    lzqr = math.log(zqr) if zqr > ZMIN else math.log(ZMIN)
    lzqs = math.log(zqs) if zqs > ZMIN else math.log(ZMIN)
    lzrs = math.log(zrs) if zrs > ZMIN else math.log(ZMIN)
    lzsum = 0.5 * (lzqr + lzqs + lzrs)
    lzq = lzsum - lzrs
    lzr = lzsum - lzqs
    lzs = lzsum - lzqr
    lzmax = math.log(ZMAX)
    if lzq > lzmax:
        lzq, lzr, lzs = lzmax, lzqr, lzqs
    elif lzr > lzmax:
        lzr, lzq, lzs = lzmax, lzqr, lzrs
    elif lzs > lzmax:
        lzs, lzq, lzr = lzmax, lzqs, lzrs
    return math.exp(lzq), math.exp(lzr), math.exp(lzs)


def _sqrtz(z):
    """sqrt + [zmin,zmax] clamp, scalar or per-partition (insertBIG:541)."""
    if isinstance(z, np.ndarray):
        return np.clip(np.sqrt(z), ZMIN, ZMAX)
    z = math.sqrt(z)
    return min(max(z, ZMIN), ZMAX)


class SprTree:
    """Ring-based unrooted tree with the PhyloTree interface (so the
    unchanged TreeSearch drives the engines over it) plus the surgery
    primitives of searchAlgo.c.  nodep[i] is the canonical ring member
    (reassigned by node_rectifier, like tr->nodep)."""

    def __init__(self, ntips):
        self.ntips = ntips
        self.nnodes = 2 * ntips - 1
        # nodep: the tr->nodep iteration array — node_rectifier PERMUTES
        # the inner entries (nodep[i].number need not equal i afterwards,
        # exactly like the reference).  ring: a fixed member per node
        # NUMBER, used for all number->ring lookups.
        self.nodep = [None] * (2 * ntips)
        self.ring = [None] * (2 * ntips)
        for i in range(1, ntips + 1):
            self.nodep[i] = self.ring[i] = RingNode(i)
        for i in range(ntips + 1, 2 * ntips - 1):
            # setupTree ring construction (axml.c:631-655): members built
            # j=1..3 with p->next = previous; nodep[i] = last member
            m1, m2, m3 = RingNode(i), RingNode(i), RingNode(i)
            m3.next = m2
            m2.next = m1
            m1.next = m3
            self.nodep[i] = self.ring[i] = m3
        self.start = 1  # node number of tr->start

    # -- construction from a PhyloTree --------------------------------

    @classmethod
    def from_phylo(cls, t):
        """Build rings from a parsed PhyloTree: preorder from tip 1, the
        first-reached member of each inner node faces its parent and the
        children attach in the PhyloTree adjacency order (which mirrors
        the reference's treeReadLen ring order)."""
        st = cls(t.ntips)
        used = {}  # inner number -> next free member (cycled via .next)

        def member_for(num, toward_parent):
            if num <= t.ntips:
                return st.ring[num]
            if num not in used:
                used[num] = st.ring[num]  # canonical faces the parent
                return used[num]
            used[num] = used[num].next
            return used[num]

        seen = set()

        def walk(p, parent):
            seen.add(p)
            for w in t.adj[p]:
                if w == parent:
                    continue
                mp = member_for(p, False)
                mw = member_for(w, True)
                hookup(mp, mw, t.get_z(p, w))
                if w not in seen:
                    walk(w, p)

        # root the construction at tip 1's edge
        first = next(iter(t.adj[1]))
        hookup(st.nodep[1], member_for(first, True), t.get_z(1, first))
        seen.update((1,))
        walk(first, 1)
        return st

    # -- PhyloTree interface ------------------------------------------

    def is_tip(self, v):
        return v <= self.ntips

    def clv_slot(self, v):
        assert not self.is_tip(v)
        return v - self.ntips - 1

    def members(self, p):
        m = self.ring[p]
        yield m
        if not self.is_tip(p):
            yield m.next
            yield m.next.next

    def find_member(self, p, q):
        for m in self.members(p):
            if m.back is not None and m.back.number == q:
                return m
        raise KeyError((p, q))

    def get_z(self, a, b):
        z = self.find_member(a, b).z
        return float(z[0]) if isinstance(z, np.ndarray) else z

    def get_zv(self, a, b):
        z = self.find_member(a, b).z
        return z if isinstance(z, np.ndarray) else np.array([z])

    def set_z(self, a, b, z):
        m = self.find_member(a, b)
        m.z = m.back.z = z

    def expand_branches(self, nb):
        """Per-partition branch vectors (-M): every edge's two members
        share one nb-vector, like p->z/p->back->z."""
        for i in range(1, 2 * self.ntips - 1):
            for m in self.members(i):
                if m.back is not None and not isinstance(m.z, np.ndarray):
                    v = np.full(nb, float(m.z))
                    m.z = m.back.z = v

    def ring_children(self, p, parent):
        """the (q, r) of computeTraversalInfo: ring order from the
        parent-facing member."""
        m = self.find_member(p, parent)
        return m.next.back.number, m.next.next.back.number

    class _Nbrs:
        def __init__(self, t, p):
            self.t = t
            self.p = p

        def __iter__(self):
            for m in self.t.members(self.p):
                if m.back is not None:
                    yield m.back.number

        def __getitem__(self, q):
            # raw member z: -M updates mutate the shared vector in place
            return self.t.find_member(self.p, q).z

    class _Adj:
        def __init__(self, t):
            self.t = t

        def __getitem__(self, p):
            return SprTree._Nbrs(self.t, p)

    @property
    def adj(self):
        return SprTree._Adj(self)

    def edges(self):
        out = []
        seen = set()
        for i in range(1, 2 * self.ntips - 1):
            for m in self.members(i):
                if m.back is None:
                    continue
                key = (min(i, m.back.number), max(i, m.back.number))
                if key not in seen:
                    seen.add(key)
                    out.append(key)
        return out


class Topol:
    """topol (topologies.c): standard-order link list.  links[k] =
    [member_p, member_q, z, descend, sibling, valptr]."""

    __slots__ = ("links", "likelihood", "start")

    def __init__(self):
        self.links = []
        self.likelihood = UNLIKELY
        self.start = None


def _save_subtree(p, tpl, ntips):
    """saveSubtree (topologies.c:229): returns the link index for branch
    p--p->back with children merged in ascending min-tip order."""
    links = tpl.links
    z = p.z.copy() if isinstance(p.z, np.ndarray) else p.z
    r = [p, p.back, z, 0, 0, None]
    ri = len(links)
    links.append(r)
    q = p.back
    if q.number <= ntips:
        r[5] = q.number
    else:
        s = q.next
        while True:
            t = _save_subtree(s, tpl, ntips)
            t0, t1 = 0, r[3]
            while t1 and links[t1][5] < links[t][5]:
                t0 = t1
                t1 = links[t1][4]
            if t0:
                links[t0][4] = t
            else:
                r[3] = t
            links[t][4] = t1
            s = s.next
            if s is q:
                break
        r[5] = links[r[3]][5]
    return ri


def _min_subtree_tip(p0, ntips):
    if p0.number <= ntips:
        return p0
    p = p0.next
    best = _min_subtree_tip(p.back, ntips)
    p = p.next
    while p is not p0:
        t = _min_subtree_tip(p.back, ntips)
        if t.number < best.number:
            best = t
        p = p.next
    return best


def _min_tree_tip(p, ntips):
    a = _min_subtree_tip(p, ntips)
    b = _min_subtree_tip(p.back, ntips)
    return a if a.number < b.number else b


def save_tree(st, likelihood, tpl):
    """saveTree (topologies.c:291)."""
    tpl.links = []
    start_m = st.find_member(st.start, next(iter(st.adj[st.start])))
    _save_subtree(_min_tree_tip(start_m, st.ntips), tpl, st.ntips)
    tpl.likelihood = likelihood
    tpl.start = st.start


def _cmp_subtopol(l1, p1, l2, p2):
    """cmpSubtopol (topologies.c:452)."""
    if not p1[3] and not p2[3]:
        return (p1[5] > p2[5]) - (p1[5] < p2[5])
    if not p1[3]:
        return -1
    if not p2[3]:
        return 1
    d1, d2 = l1[p1[3]], l2[p2[3]]
    while True:
        c = _cmp_subtopol(l1, d1, l2, d2)
        if c:
            return c
        if not d1[4] and not d2[4]:
            return 0
        if not d1[4]:
            return -1
        if not d2[4]:
            return 1
        d1, d2 = l1[d1[4]], l2[d2[4]]


def cmp_topol(t1, t2):
    r1, r2 = t1.links[0], t2.links[0]
    v1, v2 = r1[0].number, r2[0].number  # root tip values
    if v1 != v2:
        return -1 if v1 < v2 else 1
    return _cmp_subtopol(t1.links, r1, t2.links, r2)


def restore_tree(tpl, st, ts):
    """restoreTree (topologies.c:331): clear all backs, re-hookup the
    saved links, full evaluate."""
    for i in range(1, 2 * st.ntips - 1):
        for m in st.members(i):
            m.back = None
    for r in tpl.links:
        hookup(r[0], r[1], r[2])
    st.start = tpl.start
    return ts.evaluate_generic(full=True)


class BestList:
    """bestlist (topologies.c:370-655): trees ordered by score, duplicate
    topologies detected via the standard-order comparison."""

    def __init__(self, keep, st):
        self.keep = keep
        self.st = st
        self.by_score = []  # list of Topol, best first
        self.best = UNLIKELY
        self.worst = UNLIKELY

    def reset(self):
        self.by_score = []
        self.best = UNLIKELY
        self.worst = UNLIKELY

    @property
    def nvalid(self):
        return len(self.by_score)

    def save(self, ts, keep_identical):
        """saveBestTree semantics: returns the 1-based score rank or 0."""
        lh = ts.likelihood
        tpl = Topol()
        save_tree(self.st, lh, tpl)
        for t in self.by_score:
            if cmp_topol(tpl, t) == 0:
                if not keep_identical:
                    return 0
                # replace the stored copy (branch lengths/score updated)
                self.by_score.remove(t)
                break
        else:
            if self.nvalid >= self.keep and lh < self.worst:
                return 0
        # insert by score (descending)
        pos = 0
        while pos < len(self.by_score) and \
                self.by_score[pos].likelihood >= lh:
            pos += 1
        self.by_score.insert(pos, tpl)
        if len(self.by_score) > self.keep:
            self.by_score.pop()
        self.best = self.by_score[0].likelihood
        if len(self.by_score) == self.keep:
            self.worst = self.by_score[-1].likelihood
        return pos + 1

    def recall(self, rank, ts):
        rank = max(1, min(rank, self.nvalid))
        if rank > 0:
            return restore_tree(self.by_score[rank - 1], self.st, ts)
        return None


class RfConvergence:
    """The -D search convergence criterion: a two-slot bipartition store
    (tr->h) restating bipartitionList.c — insertHashRF:385,
    cleanupHashTable:181, convergenceCriterion:541.  Keys are the tip
    sets on the far side of each internal edge as seen from tip 1
    (bitVectorInitravSpecial:472 roots the traversal at nodep[1]->back,
    so an inserted vector never contains tip 1); values are the 2-bit
    slot masks of e->treeVector[0]."""

    def __init__(self, st):
        self.st = st
        self.table = {}  # frozenset(tip numbers) -> slot mask (bits 1|2)

    def _bipartitions(self):
        """Subtree tip sets of every inner node except the one adjacent
        to tip 1 — the mxtips-3 internal edges (bCounter assert)."""
        st = self.st
        ntips = st.ntips
        out = []

        def down(m):
            # m: the parent-facing member of the node entered
            if m.number <= ntips:
                return frozenset((m.number,))
            s = down(m.next.back) | down(m.next.next.back)
            out.append(s)  # the edge (m, m.back); m.back is never a tip
            return s

        root = st.ring[1].back  # nodep[1]->back: its edge to tip 1 is
        down(root.next.back)    # not an internal bipartition
        s2 = down(root.next.next.back)
        # merge: out already holds both child subtrees' contributions
        del s2
        assert len(out) == ntips - 3, len(out)
        return out

    def store(self, iteration):
        """Record the current tree in slot iteration%2, first clearing
        that slot's stale bits (cleanupHashTable) when iteration > 1."""
        slot = iteration % 2
        if iteration > 1:
            self.cleanup(slot)
        bit = 1 << slot
        for b in self._bipartitions():
            self.table[b] = self.table.get(b, 0) | bit

    def cleanup(self, slot):
        keep = 2 >> slot  # slot 0 keeps bit 2, slot 1 keeps bit 1
        self.table = {k: v & keep for k, v in self.table.items()
                      if v & keep}

    def rrf(self):
        """convergenceCriterion: bipartitions in exactly one slot over
        2*(mxtips-3)."""
        rf = sum(1 for v in self.table.values() if v == 1 or v == 2)
        return rf / (2.0 * (self.st.ntips - 3))

    def clear(self):
        self.table = {}

    def seed_from_newick(self, newick, slot, taxa_names):
        """Re-populate one slot from a stored tr->tree0/tree1 topology
        string on -R restart (readCheckpoint, searchAlgo.c:1545-1580:
        treeReadTopologyString + bitVectorInitravSpecial into slot
        0/1)."""
        from .examl_io import parse_newick_topology
        if isinstance(newick, bytes):
            newick = newick.split(b"\0", 1)[0].decode()
        t = parse_newick_topology(newick.strip(), taxa_names)
        out = []
        start = 1
        back = next(iter(t.adj[start]))

        def down(m, parent):
            if t.is_tip(m):
                return frozenset((m,))
            s = frozenset()
            for w in t.adj[m]:
                if w != parent:
                    s |= down(w, m)
            out.append(s)
            return s

        for w in t.adj[back]:
            if w != start:
                down(w, back)
        assert len(out) == len(taxa_names) - 3, len(out)
        bit = 1 << slot
        for b in out:
            self.table[b] = self.table.get(b, 0) | bit


class SprSearch:
    """computeBIGRAPID over a TreeSearch on an SprTree."""

    def __init__(self, ts, do_cutoff=True, big_cutoff=False, stepwidth=5,
                 max_rearrange=21, log=None, convergence_criterion=False,
                 save_best_trees=0):
        self.ts = ts
        self.st = ts.tree
        assert isinstance(self.st, SprTree)
        self.thorough = False
        self.do_cutoff = do_cutoff
        self.big_cutoff = big_cutoff
        self.stepwidth = stepwidth
        self.max_rearrange = max_rearrange
        self.log = log or (lambda *_: None)
        # -D: RF-distance stopping criterion (tr->searchConvergenceCriterion)
        self.convergence_criterion = convergence_criterion
        self.rfconv = RfConvergence(self.st)
        # -g: constraintVector by node number (treeReadLenMULT labels);
        # None when no constraint tree is in use
        self.constraint = None
        # -B: keep the N best distinct ML trees seen during the search
        # (tr->saveBestTrees / bestML, searchAlgo.c:1944)
        self.save_best_trees = save_best_trees
        self.best_ml = (BestList(save_best_trees, self.st)
                        if save_best_trees > 0 else None)
        # optional writeCheckpoint hook: called as writer(state, fields)
        # at the reference's write points (searchAlgo.c:2155/:2425)
        self.checkpoint_writer = None
        # -D bookkeeping for checkpoints: topology strings captured at
        # the rfconv store points (tr->tree0/tree1, searchAlgo.c:2178);
        # topology_string_fn is set by the CLI to a Tree2String-style
        # callback
        self.topology_string_fn = None
        self.slot_tree_strings = [None, None]
        self.good_trees = []  # good-tree lnls after compute_big_rapid
        # tr-> search state
        self.start_lh = 0.0
        self.end_lh = 0.0
        self.best_of_node = UNLIKELY
        self.remove_node = None   # ring member
        self.insert_node = None   # ring member
        self.zqr = DEFAULTZ
        self.current_zqr = DEFAULTZ
        self.lzq = self.lzr = self.lzs = DEFAULTZ
        self.current_lzq = self.current_lzr = self.current_lzs = DEFAULTZ
        self.lh_cutoff = 0.0
        self.lh_avg = 0.0
        self.lh_dec = 0
        self.it_count = 0
        # infoList (n=50)
        self.ilist_n = 50
        self.ilist = []

    # ---- member-based traversal (computeTraversalInfo on rings) ------
    # Node numbers are ambiguous mid-surgery (makenewzGeneric is called on
    # node pairs that are not yet adjacent, removeNodeBIG:452), so these
    # drive the engines through member handles directly.

    def _collect_m(self, m, partial, out):
        self.ts._collect_ring(m, partial, out)

    def _evaluate_branch(self, m):
        """evaluateGeneric(tr, m, FALSE) at the branch m--m->back:
        each side collected only if its x flag is unset (:940)."""
        ts = self.ts
        out = []
        if ts.oriented.get(m.number) is not m:
            self._collect_m(m, True, out)
        mb = m.back
        if ts.oriented.get(mb.number) is not mb:
            self._collect_m(mb, True, out)
        return ts.evaluate_generic(full=False, p=m.number, q=mb.number,
                                   entries=out, z=m.z)

    def _newview(self, m):
        """newviewGeneric(tr, m, FALSE): CLV at m's node oriented toward
        m->back."""
        out = []
        self._collect_m(m, True, out)
        self.ts._run(out)

    def _makenewz(self, m1, m2, z0):
        """makenewzGeneric on a (possibly not-yet-hooked) member pair;
        sides collected only when their x flag is unset (:1386)."""
        ts = self.ts
        out = []
        if ts.oriented.get(m1.number) is not m1:
            self._collect_m(m1, True, out)
        if ts.oriented.get(m2.number) is not m2:
            self._collect_m(m2, True, out)
        if ts.NB > 1:
            return ts.makenewz_generic_vec(m1.number, m2.number, z0,
                                           ITERATIONS, mask=False,
                                           entries=out)
        return ts.makenewz_generic(m1.number, m2.number, z0, ITERATIONS,
                                   entries=out)

    def _update(self, m):
        """searchAlgo update(tr, p): optimize branch m--m->back."""
        self.ts.update(m.number, m.back.number)

    def local_smooth(self, p, maxtimes):
        """localSmooth (searchAlgo.c:280)."""
        ts = self.ts
        if self.st.is_tip(p.number):
            return False
        nb1 = ts.NB == 1
        if nb1:
            ts.partition_converged = False
        else:
            ts.partition_converged[:] = False
        while maxtimes > 0:
            maxtimes -= 1
            if nb1:
                ts.partition_smoothed = True
            else:
                ts.partition_smoothed[:] = True
            q = p
            while True:
                self._update(q)
                q = q.next
                if q is p:
                    break
            smoothed = (ts.partition_smoothed if nb1
                        else all(ts.partition_smoothed))
            if smoothed:  # allSmoothed
                if nb1:
                    ts.partition_converged = True
                else:
                    ts.partition_converged[:] = True
                break
        if nb1:
            ts.partition_smoothed = False
            ts.partition_converged = False
        else:
            ts.partition_smoothed[:] = False
            ts.partition_converged[:] = False
        return True

    # ---- SPR surgery -------------------------------------------------

    def remove_node_big(self, p):
        """removeNodeBIG (searchAlgo.c:440)."""
        q = p.next.back
        r = p.next.next.back
        z0 = q.z * r.z
        result = self._makenewz(q, r, z0)
        self.zqr = result
        hookup(q, r, result)
        p.next.back = p.next.next.back = None
        return q

    def remove_node_restore_big(self, p):
        """removeNodeRestoreBIG (:464)."""
        q = p.next.back
        r = p.next.next.back
        self._newview(q)
        self._newview(r)
        hookup(q, r, self.current_zqr)
        p.next.back = p.next.next.back = None
        return q

    def insert_big(self, p, q):
        """insertBIG (:482)."""
        r = q.back
        s = p.back
        if self.thorough:
            nb = self.ts.NB
            qz = q.z
            dz = np.full(nb, DEFAULTZ) if nb > 1 else DEFAULTZ
            zqr = self._makenewz(q, r, qz)
            zqs = self._makenewz(q, s, dz)
            zrs = self._makenewz(r, s, dz)
            if nb > 1:
                e1 = np.empty(nb)
                e2 = np.empty(nb)
                e3 = np.empty(nb)
                for i in range(nb):
                    e1[i], e2[i], e3[i] = _branch_split(zqr[i], zqs[i],
                                                        zrs[i])
                hookup(p.next, q, e1)
                hookup(p.next.next, r, e2)
                hookup(p, s, e3)
            else:
                e1, e2, e3 = _branch_split(zqr, zqs, zrs)
                hookup(p.next, q, e1)
                hookup(p.next.next, r, e2)
                hookup(p, s, e3)
        else:
            z = _sqrtz(q.z)
            hookup(p.next, q, z)
            hookup(p.next.next, r, z)
        self._newview(p)
        if self.thorough:
            self.local_smooth(p, SMOOTHINGS)
            self.lzq = p.next.z
            self.lzr = p.next.next.z
            self.lzs = p.z
        return True

    def insert_restore_big(self, p, q):
        """insertRestoreBIG (:574)."""
        r = q.back
        s = p.back
        if self.thorough:
            hookup(p.next, q, self.current_lzq)
            hookup(p.next.next, r, self.current_lzr)
            hookup(p, s, self.current_lzs)
        else:
            z = _sqrtz(q.z)
            hookup(p.next, q, z)
            hookup(p.next.next, r, z)
        self._newview(p)
        return True

    def _constraint_checker(self, p):
        """checker (searchAlgo.c:69): first non--9 group label found in
        the subtree behind member p."""
        cv = self.constraint
        group = cv[p.number]
        if p.number <= self.st.ntips:
            return group
        if group != -9:
            return group
        group = self._constraint_checker(p.next.back)
        if group != -9:
            return group
        return self._constraint_checker(p.next.next.back)

    def test_insert_big(self, p, q):
        """testInsertBIG (:683); with a -g constraint tree the insertion
        is gated by the group labels of p, q and r (:697-722)."""
        r = q.back
        qz, pz = q.z, p.z
        start_lh = self.end_lh
        if self.constraint is not None:
            cv = self.constraint
            do_it = False
            r_num = cv[r.number]
            q_num = cv[q.number]
            p_num = cv[p.number]
            if p_num == -9:
                p_num = self._constraint_checker(p.back)
            if p_num == -9:
                do_it = True
            else:
                if q_num == -9:
                    q_num = self._constraint_checker(q)
                if r_num == -9:
                    r_num = self._constraint_checker(r)
                if p_num == r_num or p_num == q_num:
                    do_it = True
            if not do_it:
                return True
        self.insert_big(p, q)
        lnl = self._evaluate_branch(p.next.next)
        if lnl > self.best_of_node:
            self.best_of_node = lnl
            self.insert_node = q
            self.remove_node = p
            self.current_zqr = self.zqr
            self.current_lzr = self.lzr
            self.current_lzq = self.lzq
            self.current_lzs = self.lzs
        if lnl > self.end_lh:
            self.insert_node = q
            self.remove_node = p
            self.current_zqr = self.zqr
            self.end_lh = lnl
        hookup(q, r, qz)
        p.next.back = p.next.next.back = None
        if self.thorough:
            s = p.back
            hookup(p, s, pz)
        if self.do_cutoff and lnl < start_lh:
            self.lh_avg += start_lh - lnl
            self.lh_dec += 1
            return not (start_lh - lnl >= self.lh_cutoff)
        return True

    def add_traverse_big(self, p, q, mintrav, maxtrav):
        """addTraverseBIG (:784)."""
        mintrav -= 1
        if mintrav <= 0:
            if not self.test_insert_big(p, q):
                return
        maxtrav -= 1
        if not self.st.is_tip(q.number) and maxtrav > 0:
            self.add_traverse_big(p, q.next.back, mintrav, maxtrav)
            self.add_traverse_big(p, q.next.next.back, mintrav, maxtrav)

    def rearrange_big(self, p, mintrav, maxtrav):
        """rearrangeBIG (:804)."""
        st = self.st
        if maxtrav < 1 or mintrav > maxtrav:
            return 0
        q = p.back
        if not st.is_tip(p.number):
            p1 = p.next.back
            p2 = p.next.next.back
            if not st.is_tip(p1.number) or not st.is_tip(p2.number):
                p1z, p2z = p1.z, p2.z
                self.remove_node_big(p)
                if not st.is_tip(p1.number):
                    self.add_traverse_big(p, p1.next.back, mintrav, maxtrav)
                    self.add_traverse_big(p, p1.next.next.back, mintrav,
                                          maxtrav)
                if not st.is_tip(p2.number):
                    self.add_traverse_big(p, p2.next.back, mintrav, maxtrav)
                    self.add_traverse_big(p, p2.next.next.back, mintrav,
                                          maxtrav)
                hookup(p.next, p1, p1z)
                hookup(p.next.next, p2, p2z)
                self._newview(p)
        if not st.is_tip(q.number) and maxtrav > 0:
            q1 = q.next.back
            q2 = q.next.next.back
            if ((not st.is_tip(q1.number) and
                 (not st.is_tip(q1.next.back.number) or
                  not st.is_tip(q1.next.next.back.number))) or
                (not st.is_tip(q2.number) and
                 (not st.is_tip(q2.next.back.number) or
                  not st.is_tip(q2.next.next.back.number)))):
                q1z, q2z = q1.z, q2.z
                self.remove_node_big(q)
                mintrav2 = max(mintrav, 2)
                if not st.is_tip(q1.number):
                    self.add_traverse_big(q, q1.next.back, mintrav2, maxtrav)
                    self.add_traverse_big(q, q1.next.next.back, mintrav2,
                                          maxtrav)
                if not st.is_tip(q2.number):
                    self.add_traverse_big(q, q2.next.back, mintrav2, maxtrav)
                    self.add_traverse_big(q, q2.next.next.back, mintrav2,
                                          maxtrav)
                hookup(q.next, q1, q1z)
                hookup(q.next.next, q2, q2z)
                self._newview(q)
        return 1

    def test_insert_restore_big(self, p, q):
        """testInsertRestoreBIG (:1037)."""
        if self.thorough:
            self.insert_big(p, q)
            self._evaluate_branch(p.next.next)
        else:
            self.insert_restore_big(p, q)
            # re-establish the CLVs the reference spins on x flags for
            x = p.next.next
            y = p.back
            ts, st = self.ts, self.st
            if not st.is_tip(x.number) and ts.oriented.get(x.number) is not x:
                self._newview(x)
            if not st.is_tip(y.number) and ts.oriented.get(y.number) is not y:
                self._newview(y)
            self.ts.likelihood = self.end_lh
        return True

    def restore_tree_fast(self):
        """restoreTreeFast (:1095)."""
        self.remove_node_restore_big(self.remove_node)
        self.test_insert_restore_big(self.remove_node, self.insert_node)

    def restore_topology_only(self, bt, best_ml=None):
        """restoreTopologyOnly (:610): record the best insertion for this
        node into bt (and, under -B, into bestML with keep_identical
        FALSE, :664) without computing anything."""
        p = self.remove_node
        q = self.insert_node
        current_lh = self.ts.likelihood
        p1 = p.next.back
        p2 = p.next.next.back
        p1z, p2z = p1.z, p2.z
        hookup(p1, p2, self.current_zqr)
        p.next.back = p.next.next.back = None
        qz, pz = q.z, p.z
        r = q.back
        s = p.back
        if self.thorough:
            hookup(p.next, q, self.current_lzq)
            hookup(p.next.next, r, self.current_lzr)
            hookup(p, s, self.current_lzs)
        else:
            z = _sqrtz(q.z)
            hookup(p.next, q, z)
            hookup(p.next.next, r, z)
        self.ts.likelihood = self.best_of_node
        bt.save(self.ts, True)
        if best_ml is not None:
            best_ml.save(self.ts, False)
        self.ts.likelihood = current_lh
        hookup(q, r, qz)
        p.next.back = p.next.next.back = None
        if self.thorough:
            hookup(p, s, pz)
        hookup(p.next, p1, p1z)
        hookup(p.next.next, p2, p2z)

    # ---- infoList ----------------------------------------------------

    def reset_info_list(self):
        self.ilist = []

    def insert_info_list(self, node, likelihood):
        """insertInfoList (:352): replace the minimum if better."""
        if len(self.ilist) < self.ilist_n:
            self.ilist.append([node, likelihood])
            return
        mn = min(range(len(self.ilist)), key=lambda i: self.ilist[i][1])
        if likelihood > self.ilist[mn][1]:
            self.ilist[mn] = [node, likelihood]

    # ---- driver pieces -----------------------------------------------

    def node_rectifier(self):
        """nodeRectifier (trash.c:54): canonicalize nodep[] to the
        preorder parent-facing members from nodep[1]->back."""
        st = self.st
        st.start = 1
        count = [0]
        old = {i: st.nodep[i] for i in range(st.ntips + 1, 2 * st.ntips - 1)}

        def reorder(p):
            if st.is_tip(p.number):
                return
            st.nodep[st.ntips + 1 + count[0]] = p
            count[0] += 1
            reorder(p.next.back)
            reorder(p.next.next.back)

        reorder(st.nodep[1].back)
        assert count[0] == st.ntips - 2, (count[0], old and None)

    def tree_optimize_rapid(self, mintrav, maxtrav, bt, best_ml=None):
        """treeOptimizeRapid (:914); best_ml is tr->saveBestTrees' bestML
        list (-B), fed at :979/:1015 and through restoreTopologyOnly."""
        ts, st = self.ts, self.st
        self.node_rectifier()
        maxtrav = min(maxtrav, st.ntips - 3)
        self.reset_info_list()
        bt.reset()
        self.start_lh = self.end_lh = ts.likelihood
        if self.do_cutoff:
            if self.big_cutoff:
                if self.it_count == 0:
                    self.lh_cutoff = 0.5 * (ts.likelihood / -1000.0)
                else:
                    self.lh_cutoff = 0.5 * _cdiv(self.lh_avg, self.lh_dec)
            else:
                if self.it_count == 0:
                    self.lh_cutoff = ts.likelihood / -1000.0
                else:
                    self.lh_cutoff = _cdiv(self.lh_avg, self.lh_dec)
            self.it_count += 1
            self.lh_avg = 0.0
            self.lh_dec = 0
        for i in range(1, 2 * st.ntips - 1):
            self.best_of_node = UNLIKELY
            if self.rearrange_big(st.nodep[i], mintrav, maxtrav):
                if self.thorough:
                    if self.end_lh > self.start_lh:
                        self.restore_tree_fast()
                        self.start_lh = self.end_lh = ts.likelihood
                        bt.save(ts, True)
                        if best_ml is not None:
                            best_ml.save(ts, False)
                    elif self.best_of_node != UNLIKELY:
                        self.restore_topology_only(bt, best_ml)
                else:
                    self.insert_info_list(st.nodep[i], self.best_of_node)
                    if self.end_lh > self.start_lh:
                        self.restore_tree_fast()
                        self.start_lh = self.end_lh = ts.likelihood
        if not self.thorough:
            self.thorough = True
            for node, _lh in list(self.ilist):
                self.best_of_node = UNLIKELY
                if self.rearrange_big(node, mintrav, maxtrav):
                    if self.end_lh > self.start_lh:
                        self.restore_tree_fast()
                        self.start_lh = self.end_lh = ts.likelihood
                        bt.save(ts, True)
                        if best_ml is not None:
                            best_ml.save(ts, False)
                    elif self.best_of_node != UNLIKELY:
                        self.restore_topology_only(bt, best_ml)
            self.thorough = False
        return self.start_lh

    def determine_rearrangement_setting(self, best_t, bt,
                                        checkpoint=None):
        """determineRearrangementSetting (:1752); checkpoint resumes a
        REARR_SETTING-state -R restart (:1769-1781)."""
        ts = self.ts
        MAX_FAST = 26
        maxtrav, best_trav = 5, 5
        start_lh = ts.likelihood
        impr = True
        cutoff = self.do_cutoff
        if checkpoint is not None:
            maxtrav = checkpoint.maxtrav
            best_trav = checkpoint.best_trav
            start_lh = checkpoint.start_lh
            impr = bool(checkpoint.impr)
            cutoff = bool(checkpoint.cutoff)
            # readCheckpoint's tr-field restore (:1525-1533)
            ts.likelihood = checkpoint.tr_likelihood
            self.lh_cutoff = checkpoint.tr_lh_cutoff
            self.lh_avg = checkpoint.tr_lh_avg
            self.lh_dec = checkpoint.tr_lh_dec
            self.it_count = checkpoint.tr_it_count
        self.do_cutoff = False
        bt.reset()
        assert not self.thorough
        while impr and maxtrav < MAX_FAST:
            best_t.recall(1, ts)
            self.node_rectifier()
            if self.checkpoint_writer is not None:
                self.checkpoint_writer(CKP_REARR_SETTING, dict(
                    maxtrav=maxtrav, best_trav=best_trav,
                    start_lh=start_lh, impr=int(impr),
                    cutoff=int(cutoff), tr_likelihood=ts.likelihood,
                    tr_lh_cutoff=self.lh_cutoff, tr_lh_avg=self.lh_avg,
                    tr_lh_dec=float(self.lh_dec),
                    tr_it_count=self.it_count,
                    tr_do_cutoff=int(self.do_cutoff)))
            maxtrav_eff = min(maxtrav, self.st.ntips - 3)
            self.start_lh = self.end_lh = ts.likelihood
            for i in range(1, 2 * self.st.ntips - 1):
                self.best_of_node = UNLIKELY
                if self.rearrange_big(self.st.nodep[i], 1, maxtrav_eff):
                    if self.end_lh > self.start_lh:
                        self.restore_tree_fast()
                        self.start_lh = self.end_lh = ts.likelihood
            ts.tree_evaluate(0.25)
            bt.save(ts, True)
            if self.best_ml is not None:
                self.best_ml.save(ts, False)
            if ts.likelihood > start_lh:
                start_lh = ts.likelihood
                best_trav = maxtrav
                impr = True
            else:
                impr = False
            if self.do_cutoff:
                self.lh_cutoff = _cdiv(self.lh_avg, self.lh_dec)
                self.it_count += 1
                self.lh_avg = 0.0
                self.lh_dec = 0
            maxtrav += 5
            self.log(f"rearrangement radius {maxtrav - 5}: "
                     f"{ts.likelihood:.6f}")
        bt.recall(1, ts)
        self.do_cutoff = cutoff
        return best_trav

    def seed_rfconv_from_checkpoint(self, ck, taxa_names):
        """-R resume with -D: re-populate the RF-convergence table from
        the checkpoint's tree0/tree1 topology strings, with the
        reference's gating (readCheckpoint, searchAlgo.c:1545-1580:
        tree0 when the resumed phase's iteration count > 0, tree1 when
        > 1)."""
        if not self.convergence_criterion:
            return
        if ck.state == CKP_FAST_SPRS:
            it = ck.fast_iterations
        elif ck.state == CKP_SLOW_SPRS:
            it = ck.thorough_iterations
        else:
            return
        for slot, tstr in ((0, ck.tree0), (1, ck.tree1)):
            if it > slot:
                self.rfconv.seed_from_newick(tstr, slot, taxa_names)
                self.slot_tree_strings[slot] = \
                    tstr.split(b"\0", 1)[0].decode() \
                    if isinstance(tstr, bytes) else tstr

    def compute_big_rapid(self, estimate_model=True, initial_trav=None,
                          checkpoint=None):
        """computeBIGRAPID (:1914).  checkpoint: a parsed reference -R
        checkpoint (state FAST_SPRS/SLOW_SPRS) whose tree and model
        state the caller has already restored into this search — the
        loops resume at the reference's START_FAST_SPRS /
        START_SLOW_SPRS labels (:2074/:2346) with the stored loop
        variables."""
        ts, st = self.ts, self.st
        self.lh_avg = 0.0
        self.lh_dec = 0
        best_t = BestList(1, st)
        bt = BestList(20, st)
        epsilon = 0.01
        self.thorough = False
        ck = checkpoint
        resume_fast = ck is not None and ck.state == CKP_FAST_SPRS
        resume_slow = ck is not None and ck.state == CKP_SLOW_SPRS
        resume_rearr = ck is not None and ck.state == CKP_REARR_SETTING
        if ck is not None:
            assert resume_fast or resume_slow or resume_rearr, ck.state
            # restart(): tree+models restored by the caller; the
            # preamble is skipped (modOpt(10) ran before the earliest
            # REARR_SETTING checkpoint was written)
            ts.evaluate_generic(full=True)
            self.log(f"restart with likelihood: {ts.likelihood:.6f}")
            best_t.save(ts, True)
            if resume_rearr:
                best_trav = self.determine_rearrangement_setting(
                    best_t, bt, checkpoint=ck)
                self.log(f"best rearrangement radius: {best_trav}")
                if estimate_model:
                    ts.mod_opt(5.0)
                else:
                    ts.tree_evaluate(1.0)
                best_t.save(ts, True)
            else:
                best_trav = ck.best_trav
        else:
            # main()'s preamble before computeBIGRAPID (axml.c:2760-2764)
            ts.evaluate_generic(full=True)
            ts.tree_evaluate(1.0)
            if estimate_model:
                ts.mod_opt(10.0)
            else:
                ts.tree_evaluate(2.0)
            best_t.save(ts, True)
            if initial_trav is None:
                best_trav = self.determine_rearrangement_setting(best_t, bt)
                self.log(f"best rearrangement radius: {best_trav}")
            else:
                best_trav = initial_trav
            if estimate_model:
                ts.mod_opt(5.0)
            else:
                ts.tree_evaluate(1.0)
            best_t.save(ts, True)
        impr = True
        if self.do_cutoff:
            self.it_count = 0
        fast_iterations = 0
        lh = previous_lh = UNLIKELY
        while impr and not resume_slow:
            if resume_fast:
                # START_FAST_SPRS restore (:2080-2105)
                fast_iterations = ck.fast_iterations
                best_trav = ck.best_trav
                epsilon = ck.epsilon
                ts.likelihood = ck.tr_likelihood
                self.lh_cutoff = ck.tr_lh_cutoff
                self.lh_avg = ck.tr_lh_avg
                self.lh_dec = ck.tr_lh_dec
                self.it_count = ck.tr_it_count
                impr = bool(ck.impr)
                resume_fast = False
            else:
                best_t.recall(1, ts)
            if self.checkpoint_writer is not None:
                self.checkpoint_writer(CKP_FAST_SPRS, dict(
                    fast_iterations=fast_iterations, best_trav=best_trav,
                    thorough=0, impr=int(impr), epsilon=epsilon,
                    tr_likelihood=ts.likelihood,
                    tr_lh_cutoff=self.lh_cutoff, tr_lh_avg=self.lh_avg,
                    tr_lh_dec=float(self.lh_dec),
                    tr_it_count=self.it_count,
                    tr_do_cutoff=int(self.do_cutoff)))
            # -D check at the top of each fast cycle (searchAlgo.c:2160):
            # store the current best tree in slot fastIterations%2, then
            # compare against the previous cycle's tree.
            if self.convergence_criterion:
                self.rfconv.store(fast_iterations)
                if self.topology_string_fn is not None:
                    self.slot_tree_strings[fast_iterations % 2] = \
                        self.topology_string_fn()
                if fast_iterations > 0:
                    rrf = self.rfconv.rrf()
                    if rrf <= 0.01:  # 1% cutoff
                        self.log(f"converged fast cycle {fast_iterations}"
                                 f": {rrf:.6f}")
                        break
                    self.log("convergence fast cycle "
                             f"{fast_iterations - 1}->{fast_iterations}"
                             f": {rrf:.6f}")
            fast_iterations += 1
            ts.tree_evaluate(1.0)
            best_t.save(ts, True)
            lh = previous_lh = ts.likelihood
            self.log(f"fast SPR cycle {fast_iterations}: {lh:.6f}")
            self.tree_optimize_rapid(1, best_trav, bt, self.best_ml)
            impr = False
            for i in range(1, bt.nvalid + 1):
                bt.recall(i, ts)
                ts.tree_evaluate(0.25)
                difference = abs(ts.likelihood - previous_lh)
                if ts.likelihood > lh and difference > epsilon:
                    impr = True
                    lh = ts.likelihood
                    best_t.save(ts, True)
        if not resume_slow:
            if self.convergence_criterion:
                # both exits empty the table (searchAlgo.c:2202/2303)
                self.rfconv.clear()
            self.thorough = True
            impr = True
            best_t.recall(1, ts)
            ts.evaluate_generic(full=True)
            if estimate_model:
                ts.mod_opt(1.0)
            else:
                ts.tree_evaluate(1.0)
        rearrangements_min = 1
        rearrangements_max = self.stepwidth
        thorough_iterations = 0
        while True:
            if resume_slow:
                # START_SLOW_SPRS restore (:2346-2379)
                impr = bool(ck.impr)
                self.thorough = bool(ck.thorough)
                best_trav = ck.best_trav
                rearrangements_max = ck.rearrangements_max
                rearrangements_min = ck.rearrangements_min
                thorough_iterations = ck.thorough_iterations
                fast_iterations = ck.fast_iterations
                epsilon = ck.epsilon
                ts.likelihood = ck.tr_likelihood
                self.lh_cutoff = ck.tr_lh_cutoff
                self.lh_avg = ck.tr_lh_avg
                self.lh_dec = ck.tr_lh_dec
                self.it_count = ck.tr_it_count
                resume_slow = False
            else:
                best_t.recall(1, ts)
            if self.checkpoint_writer is not None:
                self.checkpoint_writer(CKP_SLOW_SPRS, dict(
                    fast_iterations=fast_iterations,
                    thorough_iterations=thorough_iterations,
                    best_trav=best_trav, thorough=1, impr=int(impr),
                    rearrangements_min=rearrangements_min,
                    rearrangements_max=rearrangements_max,
                    lh=lh, previous_lh=previous_lh, epsilon=epsilon,
                    tr_likelihood=ts.likelihood,
                    tr_lh_cutoff=self.lh_cutoff, tr_lh_avg=self.lh_avg,
                    tr_lh_dec=float(self.lh_dec),
                    tr_it_count=self.it_count,
                    tr_do_cutoff=int(self.do_cutoff)))
            if impr:
                rearrangements_min = 1
                rearrangements_max = self.stepwidth
                # -D check (searchAlgo.c:2438): slot thoroughIterations%2
                if self.convergence_criterion:
                    self.rfconv.store(thorough_iterations)
                    if self.topology_string_fn is not None:
                        self.slot_tree_strings[thorough_iterations % 2] = \
                            self.topology_string_fn()
                    if thorough_iterations > 0:
                        rrf = self.rfconv.rrf()
                        if rrf <= 0.01:  # goto cleanup
                            self.log("converged thorough cycle "
                                     f"{thorough_iterations}: {rrf:.6f}")
                            break
                        self.log("convergence thorough cycle "
                                 f"{thorough_iterations - 1}->"
                                 f"{thorough_iterations}: {rrf:.6f}")
                thorough_iterations += 1
            else:
                rearrangements_max += self.stepwidth
                rearrangements_min += self.stepwidth
                if rearrangements_max > self.max_rearrange:
                    break
            ts.tree_evaluate(1.0)
            previous_lh = lh = ts.likelihood
            # saveBestTree refreshes the stored copy's branch lengths
            # (searchAlgo.c:2519) — the cleanup path re-evaluates THESE
            best_t.save(ts, True)
            self.log(f"thorough SPR cycle {thorough_iterations} "
                     f"[{rearrangements_min},{rearrangements_max}]: "
                     f"{lh:.6f}")
            self.tree_optimize_rapid(rearrangements_min, rearrangements_max,
                                     bt, self.best_ml)
            impr = False
            for i in range(1, bt.nvalid + 1):
                bt.recall(i, ts)
                ts.tree_evaluate(0.25)
                difference = abs(ts.likelihood - previous_lh)
                if ts.likelihood > lh and difference > epsilon:
                    impr = True
                    lh = ts.likelihood
                    best_t.save(ts, True)
        ts.evaluate_generic(full=True)
        self.log(f"likelihood of best tree: {ts.likelihood:.6f}")
        final = ts.likelihood
        # -B epilogue (searchAlgo.c:2577): re-load each good tree and
        # report it; leaves the tree at the last entry, like the
        # reference, after the result tree has been produced.
        self.good_trees = []
        if self.best_ml is not None:
            for i in range(1, self.best_ml.nvalid + 1):
                self.best_ml.recall(i, ts)
                self.good_trees.append(ts.likelihood)
        return final


class _NewickStream:
    """Character-level reader with the reference's treeGetCh semantics
    (treeIO.c:60: skip whitespace, return next char)."""

    def __init__(self, text):
        self.text = text
        self.pos = 0

    def getch(self):
        while self.pos < len(self.text):
            c = self.text[self.pos]
            self.pos += 1
            if not c.isspace():
                return c
        return ""

    def ungetc(self):
        self.pos -= 1

    def read_name(self):
        out = []
        while self.pos < len(self.text):
            c = self.text[self.pos]
            if c in "(),:;[]":
                break
            out.append(c)
            self.pos += 1
        return "".join(out).strip()

    def flush_len(self):
        """treeFlushLen: consume an optional :branchLength."""
        c = self.getch()
        if c == ":":
            while self.pos < len(self.text) and \
                    self.text[self.pos] in "0123456789.eE+-":
                self.pos += 1
        elif c:
            self.ungetc()

    def flush_label(self):
        self.read_name()


def read_constraint_tree(text, taxa, seed):
    """treeReadLenMULT (treeIO.c:1033) + getStartingTree (:1162) for -g:
    parse a multifurcating constraint tree over ALL taxa, resolving each
    multifurcation randomly with the reference's srand/rand stream
    (randomInt(10000), :916), and label every node with its constraint
    group in constraintVector.  Returns (SprTree, constraint dict).

    Only unrooted constraints (>= 3 top-level children) are supported,
    like the reference's non-rooted path."""
    import ctypes
    libc = ctypes.CDLL(None)
    libc.srand(ctypes.c_uint(seed))

    def random_int(n):
        return libc.rand() % n

    ntips = len(taxa)
    tip_no = {name: i + 1 for i, name in enumerate(taxa)}
    st = SprTree(ntips)
    cv = {i: -1 for i in range(2 * ntips)}
    state = {"nextnode": ntips + 1, "ntips": 0, "partCount": 0}
    s = _NewickStream(text)

    def next_inner():
        n = state["nextnode"]
        state["nextnode"] += 1
        assert n <= 2 * ntips - 2, "rooted constraint tree not supported"
        return st.ring[n]

    def resolution():
        rn = random_int(10000)
        return 0.0 if rn == 0 else rn / 10000.0

    def add_element(p, pc):
        # addElementLenMULT (treeIO.c:921)
        cv[p.number] = pc
        ch = s.getch()
        if ch == "(":
            state["partCount"] += 1
            old = state["partCount"]
            q = next_inner()
            cv[q.number] = state["partCount"]
            add_element(q.next, old)
            assert s.getch() == ","
            add_element(q.next.next, old)
            hookup(p, q, DEFAULTZ)
            while True:
                ch = s.getch()
                if ch != ",":
                    break
                r = next_inner()
                cv[r.number] = state["partCount"]  # CURRENT count (:974)
                if resolution() < 0.5:
                    t = q.next.back
                    r.back = q.next
                    q.next.back = r
                    r.next.back = t
                    t.back = r.next
                    add_element(r.next.next, old)
                else:
                    t = q.next.next.back
                    r.back = q.next.next
                    q.next.next.back = r
                    r.next.back = t
                    t.back = r.next
                    add_element(r.next.next, old)
            assert ch == ")", "missing ) in constraint tree"
            s.flush_label()
        else:
            s.ungetc()
            n = tip_no[s.read_name()]
            q = st.ring[n]
            cv[q.number] = pc
            state["ntips"] += 1
            hookup(p, q, DEFAULTZ)
        s.flush_len()

    p = next_inner()
    while s.getch() != "(":
        pass
    add_element(p, 0)
    assert s.getch() == ","
    add_element(p.next, 0)
    ch = s.getch()
    assert ch == ",", "rooted (bifurcating-top) constraint unsupported"
    add_element(p.next.next, 0)
    while True:
        ch = s.getch()
        if ch != ",":
            break
        r = next_inner()
        cv[r.number] = 0
        if resolution() < 0.5:
            t = p.next.next.back
            r.back = p.next.next
            p.next.next.back = r
            r.next.back = t
            t.back = r.next
            add_element(r.next.next, 0)
        else:
            t = p.next.back
            r.back = p.next
            p.next.back = r
            r.next.back = t
            t.back = r.next
            add_element(r.next.next, 0)
    assert ch == ")"
    s.ungetc()
    assert s.getch() == ")"
    s.flush_label()
    s.flush_len()
    assert s.getch() == ";"
    assert state["ntips"] == ntips, "constraint must contain all taxa"
    st.start = 1  # getStartingTree: tr->start = tr->nodep[1]
    return st, cv
