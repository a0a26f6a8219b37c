"""Quartet likelihood evaluation (-f q, QUARTET_CALCULATION): the
computeQuartets flow of examl/quartets.c:349 restated over the engine
layer.  For each selected quartet {t1<t2<t3<t4} the three unrooted
4-taxon topologies are built on two scratch inner nodes, their five
branch lengths optimized by nniSmooth (quartets.c:176, 16 rounds of the
searchAlgo update()), and the lnL evaluated at the (inner2, t4) branch
(quartets.c:262).

The model/engine state is whatever the caller optimized beforehand (the
reference runs getStartingTree + treeEvaluate(1) + modOpt(0.1) first —
our -f E pipeline)."""

import numpy as np

from .search import TreeSearch
from .tree import PhyloTree, DEFAULTZ

ALL_QUARTETS = 0
RANDOM_QUARTETS = 1
GROUPED_QUARTETS = 2


def randum(seed):
    """The reference's PRNG (axml.c:353), returning (value, new_seed)."""
    mult0 = 1549
    seed0 = seed & 4095
    s = mult0 * seed0
    newseed0 = s & 4095
    s >>= 12
    seed1 = (seed >> 12) & 4095
    mult1 = 406
    s += mult0 * seed1 + mult1 * seed0
    newseed1 = s & 4095
    s >>= 12
    seed2 = (seed >> 24) & 255
    s += mult0 * seed2 + mult1 * seed1
    newseed2 = s & 255
    newseed = newseed2 << 24 | newseed1 << 12 | newseed0
    res = 0.00390625 * (newseed2 + 0.000244140625
                        * (newseed1 + 0.000244140625 * newseed0))
    return res, newseed


class QuartetTree(PhyloTree):
    """4-taxon scratch tree on the full alignment's numbering: tips keep
    their original ids (rows into the engines' tip data), the two inner
    nodes are mxtips+1 / mxtips+2 (CLV slots 0/1 — the reference reuses
    tr->nodep[mxtips+1..2], quartets.c:375)."""

    def __init__(self, mxtips, p1, p2, p3, p4, nb=1):
        self.ntips = mxtips
        self.nnodes = mxtips + 3
        q1, q2 = mxtips + 1, mxtips + 2
        self.adj = {v: {} for v in (p1, p2, p3, p4, q1, q2)}
        # hookupDefault (axml.c:489): all five branches at defaultz
        self.add_edge(q1, q2, DEFAULTZ)
        self.add_edge(q1, p1, DEFAULTZ)
        self.add_edge(q1, p2, DEFAULTZ)
        self.add_edge(q2, p3, DEFAULTZ)
        self.add_edge(q2, p4, DEFAULTZ)
        if nb > 1:
            self.expand_branches(nb)


def _nni_smooth(ts, q1, q2, p1, p2, p3, p4, maxtimes):
    """nniSmooth (quartets.c:176): optimize the five quartet branches in
    the fixed order (q1,q2), (q1,p1), (q1,p2), (q2,p3), (q2,p4)."""
    nb = ts.NB
    if nb == 1:
        ts.partition_converged = False
    else:
        ts.partition_converged[:] = False
    while maxtimes > 0:
        maxtimes -= 1
        if nb == 1:
            ts.partition_smoothed = True
        else:
            ts.partition_smoothed[:] = True
        ts.update(q1, q2)
        ts.update(p1, q1)   # update(tr, q1->next): branch q1 - p1
        ts.update(p2, q1)
        ts.update(p3, q2)
        ts.update(p4, q2)
        if nb == 1:
            if ts.partition_smoothed:
                break
        else:
            result = True
            for i in range(nb):
                if not ts.partition_smoothed[i]:
                    result = False
                else:
                    ts.partition_converged[i] = True
            if result:
                break
    if nb == 1:
        ts.partition_smoothed = False
        ts.partition_converged = False
    else:
        ts.partition_smoothed[:] = False
        ts.partition_converged[:] = False


def quartet_likelihood(engines, mxtips, p1, p2, p3, p4, per_gene_bl=False):
    """quartetLikelihood (quartets.c:217): tree ((p1,p2),(p3,p4)), five
    branches nniSmooth'ed 16 rounds, lnL at the (q2,p4) branch."""
    nb = len(engines) if per_gene_bl else 1
    qt = QuartetTree(mxtips, p1, p2, p3, p4, nb=nb)
    q1, q2 = mxtips + 1, mxtips + 2
    ts = TreeSearch(qt, engines, per_gene_bl=per_gene_bl)
    ts.newview_generic(q1, q2)
    ts.newview_generic(q2, q1)
    _nni_smooth(ts, q1, q2, p1, p2, p3, p4, 16)
    # evaluateGeneric(tr, q1->back->next->next, FALSE): the branch between
    # q2's third ring node and p4
    return ts.evaluate_generic(full=False, p=p4)


def _all_three(engines, mxtips, t1, t2, t3, t4, out, per_gene_bl=False):
    """computeAllThreeQuartets (quartets.c:283)."""
    for a, b, c, d in ((t1, t2, t3, t4), (t1, t3, t2, t4), (t1, t4, t2, t3)):
        lnl = quartet_likelihood(engines, mxtips, a, b, c, d,
                                 per_gene_bl=per_gene_bl)
        out.append((a, b, c, d, lnl))


def compute_quartets(engines, mxtips, random_quartets=0, seed=0,
                     groups=None, per_gene_bl=False, start_counter=0):
    """computeQuartets (quartets.c:349) minus the file plumbing:
    returns [(a, b, c, d, lnL)] in the reference's emission order.

    random_quartets == 0 -> ALL_QUARTETS; groups -> GROUPED_QUARTETS
    (four disjoint 1-based taxon lists); otherwise RANDOM_QUARTETS with
    the reference PRNG and sub-sampling fraction.  start_counter: a -R
    resume replays the deterministic enumeration (and, for RANDOM, the
    PRNG stream from the checkpoint's stored INITIAL seed) but skips
    the first start_counter evaluations (quartets.c:520/:560/:598)."""
    out = []
    n_quartets = mxtips * (mxtips - 1) * (mxtips - 2) * (mxtips - 3) // 24
    count = 0
    if groups is not None:
        for i1 in groups[0]:
            for i2 in groups[1]:
                for i3 in groups[2]:
                    for i4 in groups[3]:
                        if count >= start_counter:
                            _all_three(engines, mxtips, i1, i2, i3, i4,
                                       out, per_gene_bl)
                        count += 1
        return out
    if random_quartets == 0 or random_quartets >= n_quartets:
        for t1 in range(1, mxtips + 1):
            for t2 in range(t1 + 1, mxtips + 1):
                for t3 in range(t2 + 1, mxtips + 1):
                    for t4 in range(t3 + 1, mxtips + 1):
                        if count >= start_counter:
                            _all_three(engines, mxtips, t1, t2, t3, t4,
                                       out, per_gene_bl)
                        count += 1
        return out
    fraction = random_quartets / n_quartets
    while True:
        for t1 in range(1, mxtips + 1):
            for t2 in range(t1 + 1, mxtips + 1):
                for t3 in range(t2 + 1, mxtips + 1):
                    for t4 in range(t3 + 1, mxtips + 1):
                        r, seed = randum(seed)
                        if r < fraction:
                            if count >= start_counter:
                                _all_three(engines, mxtips, t1, t2, t3,
                                           t4, out, per_gene_bl)
                            count += 1
                        if count == random_quartets:
                            return out
