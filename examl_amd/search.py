"""The search/optimizer layer (ExaML's L3) restated over the C-ABI engines:
branch-length smoothing (searchAlgo.c: update/smooth/smoothTree/
treeEvaluate) and model-parameter optimization (optimizeModel.c:
brakGeneric/brentGeneric/optParamGeneric/optRates/optAlphas/optBaseFreqs/
modOpt), faithful to the reference's control flow so that the -f E
(TREE_EVALUATION) pipeline reproduces the reference's final lnL on the
same inputs.

Backend-agnostic: drives any per-partition engine object exposing
newview_traversal / evaluate_root / sum_root / core_derivs / upload_model
and a mutable .model — the HIP DnaGammaEngine or the test-only oracle
engine."""

import math

import numpy as np

from . import INNER_INNER, TIP_INNER, TIP_TIP, TravEntry, ZMIN, ZMAX

# constants — examl/axml.h:89-95,164-197, optimizeModel.c:45-49
SMOOTHINGS = 32
NEWZPERCYCLE = 1
DELTAZ = 0.00001
ALPHA_MIN, ALPHA_MAX = 0.02, 1000.0
RATE_MIN, RATE_MAX = 0.0000001, 1000000.0
FREQ_MIN = 0.001
ITMAX = 100
BRENT_ZEPS = 1.0e-5
BRENT_CGOLD = 0.3819660
MNBRAK_GOLD = 1.618034
MNBRAK_TINY = 1.0e-20
MNBRAK_GLIMIT = 100.0
UNLIKELY = -1.0e300

RATE_F, ALPHA_F, FREQ_F, LXRATE_F, LXWEIGHT_F = 0, 1, 2, 3, 4
LG4X_RATE_MIN, LG4X_RATE_MAX = 0.0000001, 1000.0  # axml.h:178


def _sign(a, b):
    return abs(a) if b > 0.0 else -abs(a)


class TreeSearch:
    """One tree + one engine per partition (joint branch lengths,
    numBranches=1)."""

    def __init__(self, tree, engines, opt_freq_flags=None,
                 auto_flags=None, empirical_freqs=None, rate_het="GAMMA",
                 max_categories=25, per_gene_bl=False):
        self.tree = tree
        self.engines = engines
        self.M = len(engines)
        self.execute_model = [True] * self.M
        self.per_partition_lnl = [0.0] * self.M
        self.likelihood = None
        self.oriented = {}  # inner node -> the parent neighbor its CLV faces
        self.start = 1  # tr->start = tr->nodep[1]
        # -M (perGeneBranchLengths): numBranches == NumberOfModels
        # (topLevelMakenewz's assert, makenewzGenericSpecial.c)
        self.NB = self.M if per_gene_bl else 1
        if per_gene_bl:
            tree.expand_branches(self.M)
            self.partition_smoothed = np.ones(self.NB, dtype=bool)
            self.partition_converged = np.zeros(self.NB, dtype=bool)
        else:
            self.partition_smoothed = True
            self.partition_converged = False
        self.opt_freq_flags = opt_freq_flags or [False] * self.M
        # AUTO protein model selection state (optimizeModel.c:2669)
        self.auto_flags = auto_flags or [False] * self.M
        self.empirical_freqs = empirical_freqs or [None] * self.M
        self.auto_prot_models = [4] * self.M  # WAG default, models.c:4222
        self.prot_freqs = [1] * self.M  # AUTO: 1 = fixed/model freqs
        # freqExponents state (models.c:4227: init 0.0)
        self.freq_exponents = [np.zeros(e.model.states) for e in engines]
        # CAT (PSR) state (models.c:4194-4201: one category, all rates 1)
        self.rate_het = rate_het
        self.max_categories = max_categories
        self.rate_cat_invocations = 1  # optimizeRateCategoryInvocations
        if rate_het == "CAT":
            self.cat_patrat = [np.ones(e.width) for e in engines]
            self.cat_lhs = [np.zeros(e.width) for e in engines]
        self._last_full = None
        # fused multi-partition executors (mseg): one launch per
        # (traversal level x tipCase) over ALL partitions — used when
        # every engine is the plain dense GAMMA engine of one states
        # family on a GPU (the per-partition loop otherwise)
        self.fused = None
        if (self.M > 1 and rate_het == "GAMMA"
                and all(type(e).__name__ == "DnaGammaEngine"
                        for e in engines)
                and len({e.states for e in engines}) == 1
                and getattr(engines[0], "device", None) is not None
                and engines[0].device.type == "cuda"):
            try:
                from .engine import MultiEngine
                self.fused = MultiEngine(engines)
            except Exception:
                self.fused = None

    # ------------------------------------------------------------------
    # traversal construction (computeTraversalInfo,
    # newviewGenericSpecial.c:691)
    # ------------------------------------------------------------------

    def _children(self, p, parent):
        t = self.tree
        if hasattr(t, "ring_children"):
            # ring order from the parent-facing member (computeTraversalInfo
            # uses p->next->back / p->next->next->back)
            return t.ring_children(p, parent)
        nbrs = [w for w in t.adj[p] if w != parent]
        assert len(nbrs) == 2
        q, r = nbrs
        return q, r

    def _collect(self, p, parent, partial, out):
        t = self.tree
        if hasattr(t, "find_member"):
            # ring tree (SPR flows): the x flag lives on a ring MEMBER —
            # validity must follow the member, not the node number, to
            # reproduce the reference's stale-CLV reuse after surgery
            # (getxnode/computeTraversalInfo semantics)
            self._collect_ring(t.find_member(p, parent), partial, out)
            return
        if t.is_tip(p):
            return
        q, r = self._children(p, parent)
        q_tip, r_tip = t.is_tip(q), t.is_tip(r)
        if q_tip and r_tip:
            tc = TIP_TIP
        elif q_tip or r_tip:
            if r_tip:  # tip data stored for q (newviewGenericSpecial.c:744)
                q, r = r, q
            if self.oriented.get(r) != p or not partial:
                self._collect(r, p, partial, out)
            tc = TIP_INNER
        else:
            if self.oriented.get(q) != p or not partial:
                self._collect(q, p, partial, out)
            if self.oriented.get(r) != p or not partial:
                self._collect(r, p, partial, out)
            tc = INNER_INNER
        e = TravEntry()
        e.tipCase = tc
        e.pNumber, e.qNumber, e.rNumber = p, q, r
        e.qz, e.rz = t.get_z(p, q), t.get_z(p, r)
        if self.NB > 1:
            # per-partition branch lengths (qz[i]/rz[i] of axml.h:434)
            e.qzv, e.rzv = t.get_zv(p, q).copy(), t.get_zv(p, r).copy()
        e.x3Slot = t.clv_slot(p)
        e.x1Slot = q if t.is_tip(q) else t.clv_slot(q)
        e.x2Slot = r if t.is_tip(r) else t.clv_slot(r)
        out.append(e)
        self.oriented[p] = parent

    def _collect_gated(self, p, parent, partial, out):
        """the x-gated top of evaluateGeneric/makenewzGeneric
        (evaluateGenericSpecial.c:940: if(!p->x) computeTraversalInfo):
        on ring trees a valid top member skips its whole side."""
        t = self.tree
        if partial and hasattr(t, "find_member") and not t.is_tip(p):
            m = t.find_member(p, parent)
            if self.oriented.get(p) is m:
                return
            self._collect_ring(m, partial, out)
            return
        self._collect(p, parent, partial, out)

    def _collect_ring(self, m, partial, out):
        """computeTraversalInfo (newviewGenericSpecial.c:691) on ring
        member m: children q/r are m->next->back / m->next->next->back;
        a child is valid iff the x flag (self.oriented) sits on exactly
        that member."""
        t = self.tree
        if t.is_tip(m.number):
            return
        q = m.next.back
        r = m.next.next.back
        q_tip, r_tip = t.is_tip(q.number), t.is_tip(r.number)
        if q_tip and r_tip:
            tc = TIP_TIP
        elif q_tip or r_tip:
            if r_tip:
                q, r = r, q
            if self.oriented.get(r.number) is not r or not partial:
                self._collect_ring(r, partial, out)
            tc = TIP_INNER
        else:
            if self.oriented.get(q.number) is not q or not partial:
                self._collect_ring(q, partial, out)
            if self.oriented.get(r.number) is not r or not partial:
                self._collect_ring(r, partial, out)
            tc = INNER_INNER
        e = TravEntry()
        e.tipCase = tc
        e.pNumber, e.qNumber, e.rNumber = m.number, q.number, r.number
        qz, rz = q.z, r.z
        e.qz = float(qz[0]) if isinstance(qz, np.ndarray) else qz
        e.rz = float(rz[0]) if isinstance(rz, np.ndarray) else rz
        if self.NB > 1:
            e.qzv = np.asarray(qz, dtype=float).copy()
            e.rzv = np.asarray(rz, dtype=float).copy()
        e.x3Slot = t.clv_slot(m.number)
        e.x1Slot = q.number if t.is_tip(q.number) else t.clv_slot(q.number)
        e.x2Slot = r.number if t.is_tip(r.number) else t.clv_slot(r.number)
        out.append(e)
        self.oriented[m.number] = m

    def _per_partition_entries(self, entries, m):
        out = []
        for e in entries:
            c = TravEntry()
            c.tipCase, c.pNumber, c.qNumber, c.rNumber = \
                e.tipCase, e.pNumber, e.qNumber, e.rNumber
            c.x1Slot, c.x2Slot, c.x3Slot = e.x1Slot, e.x2Slot, e.x3Slot
            c.qz, c.rz = e.qzv[m], e.rzv[m]
            out.append(c)
        return out

    def _run(self, entries):
        if not entries:
            return
        if self.fused is not None:
            if self.NB == 1:
                self.fused.newview_traversal(entries,
                                             active=self.execute_model)
            else:
                qz = np.array([e.qzv[:self.NB] for e in entries])
                rz = np.array([e.rzv[:self.NB] for e in entries])
                self.fused.newview_traversal(entries,
                                             active=self.execute_model,
                                             qz_ov=qz, rz_ov=rz)
            return
        if self.NB == 1:
            for m, eng in enumerate(self.engines):
                if self.execute_model[m]:
                    eng.newview_traversal(entries)
        else:
            for m, eng in enumerate(self.engines):
                if self.execute_model[m]:
                    eng.newview_traversal(
                        self._per_partition_entries(entries, m))

    # ------------------------------------------------------------------
    # L1 entry points over the engines
    # ------------------------------------------------------------------

    def newview_generic(self, p, parent):
        """newviewGeneric(tr, p, FALSE) with p->back == parent
        (newviewGenericSpecial.c:1523)."""
        out = []
        self._collect(p, parent, True, out)
        self._run(out)

    def evaluate_generic(self, full=True, p=None, q=None, entries=None,
                         z=None):
        """evaluateGeneric(tr, p (default tr->start), fullTraversal)
        (evaluateGenericSpecial.c:897).  q selects the branch explicitly
        (the reference's p->back); `entries`/`z` let a caller (the SPR
        driver) supply a pre-built traversal for a branch that cannot be
        resolved through node numbers mid-surgery."""
        t = self.tree
        if p is None:
            p = self.start
        if q is None:
            q = next(iter(t.adj[p]))  # p->back
        if entries is None:
            out = []
            self._collect_gated(p, q, not full, out)
            self._collect_gated(q, p, not full, out)
        else:
            out = entries
        self._run(out)
        if z is None:
            zv = t.get_zv(p, q)
        elif isinstance(z, np.ndarray):
            zv = z
        else:
            zv = np.array([z])
        z = float(zv[0])
        if full:
            # td[0] of the last full traversal: evaluatePartialGeneric walks
            # exactly these entries (evaluatePartialGenericSpecial.c:259)
            self._last_full = (out, p, q, zv.copy())
        # launch all partitions, then ONE host sync for the readbacks
        # (z = pz[m] per partition under -M, evaluateGenericSpecial.c:449)
        if self.fused is not None:
            vec = self.fused.evaluate_root(
                t, p, q, zv if self.NB > 1 else z,
                active=self.execute_model).cpu()
            for m in range(self.M):
                if self.execute_model[m]:
                    self.per_partition_lnl[m] = float(vec[m])
            self.likelihood = sum(self.per_partition_lnl)
            return self.likelihood
        outs = []
        for m, eng in enumerate(self.engines):
            if self.execute_model[m]:
                outs.append((m, eng.evaluate_root(
                    t, p, q, float(zv[m]) if self.NB > 1 else z)))
        if outs and not isinstance(outs[0][1], float):
            import torch
            vals = torch.cat([o[1] for o in outs]).cpu()
            for k, (m, _) in enumerate(outs):
                self.per_partition_lnl[m] = float(vals[k])
        else:
            for m, v in outs:
                self.per_partition_lnl[m] = v
        self.likelihood = sum(self.per_partition_lnl)
        return self.likelihood

    def makenewz_generic(self, p, q, z0, maxiter, entries=None):
        """makenewzGeneric (makenewzGenericSpecial.c:1355, mask=FALSE) +
        topLevelMakenewz (:1133) for numBranches=1, derivatives summed over
        executing partitions (execCore, :1075)."""
        if entries is None:
            out = []
            self._collect_gated(p, q, True, out)
            self._collect_gated(q, p, True, out)
        else:
            out = entries
        self._run(out)
        if self.fused is not None:
            self.fused.sum_root(self.tree, p, q,
                                active=self.execute_model)
        else:
            for m, eng in enumerate(self.engines):
                if self.execute_model[m]:
                    eng.sum_root(self.tree, p, q)

        z = float(z0)
        zprev = z
        zstep = (1.0 - ZMAX) * z + ZMIN
        curvat_ok = True
        outer_converged = False
        it = maxiter
        while not outer_converged:
            if curvat_ok:
                curvat_ok = False
                zprev = z
                zstep = (1.0 - ZMAX) * z + ZMIN
            z = min(max(z, ZMIN), ZMAX)
            lz = math.log(z)
            dlnL = d2lnL = 0.0
            if self.fused is not None:
                v = self.fused.core_derivs_vec(
                    lz, active=self.execute_model).cpu().numpy()
                dlnL = float(v[0::2].sum())
                d2lnL = float(v[1::2].sum())
            else:
                outs = [eng.core_derivs_async(lz)
                        for m, eng in enumerate(self.engines)
                        if self.execute_model[m]]
                if outs and not isinstance(outs[0], tuple):
                    import torch
                    vals = torch.stack(outs).cpu()
                    dlnL = float(vals[:, 0].sum())
                    d2lnL = float(vals[:, 1].sum())
                else:
                    for a, b in outs:
                        dlnL += a
                        d2lnL += b
            if (d2lnL >= 0.0) and (z < ZMAX):
                zprev = z = 0.37 * z + 0.63
                continue
            curvat_ok = True
            if d2lnL < 0.0:
                tantmp = -dlnL / d2lnL
                if tantmp < 100:
                    z *= math.exp(tantmp)
                    z = max(z, ZMIN)
                    z = min(z, 0.25 * zprev + 0.75)
                else:
                    z = 0.25 * zprev + 0.75
            z = min(z, ZMAX)
            it -= 1
            if abs(z - zprev) > zstep:
                if it < -20:
                    z = float(z0)
                    outer_converged = True
            else:
                outer_converged = True
        return z

    # ------------------------------------------------------------------
    # branch-length smoothing (searchAlgo.c:127-270,2635)
    # ------------------------------------------------------------------

    def makenewz_generic_vec(self, p, q, z0, maxiter, mask, entries=None):
        """topLevelMakenewz for numBranches == NumberOfModels (-M):
        per-partition NR with curvatOK/outerConverged masks
        (makenewzGenericSpecial.c:849-1063) and the partitionConverged
        execute mask of makenewzGeneric(mask=TRUE) (:1369-1378)."""
        NB = self.NB
        if entries is None:
            out = []
            self._collect(p, q, True, out)
            self._collect(q, p, True, out)
        else:
            out = entries
        if mask:
            for i in range(NB):
                self.execute_model[i] = not self.partition_converged[i]
        z = np.array(z0, dtype=float)
        zprev = z.copy()
        zstep = np.zeros(NB)
        corelz = np.zeros(NB)
        dl = np.zeros(NB)
        d2 = np.zeros(NB)
        miter = np.full(NB, maxiter)
        outer = np.zeros(NB, dtype=bool)
        curvat = np.ones(NB, dtype=bool)
        first = True
        while True:
            for i in range(NB):
                if not outer[i] and curvat[i]:
                    curvat[i] = False
                    zprev[i] = z[i]
                    zstep[i] = (1.0 - ZMAX) * z[i] + ZMIN
            for i in range(NB):
                if not outer[i] and not curvat[i]:
                    z[i] = min(max(z[i], ZMIN), ZMAX)
                    corelz[i] = math.log(z[i])
            for m in range(NB):
                if self.execute_model[m]:
                    self.execute_model[m] = not curvat[m]
            if first:
                self._run(out)
                if self.fused is not None:
                    self.fused.sum_root(self.tree, p, q,
                                        active=self.execute_model)
                else:
                    for m, eng in enumerate(self.engines):
                        if self.execute_model[m]:
                            eng.sum_root(self.tree, p, q)
                first = False
            dl[:] = 0.0
            d2[:] = 0.0
            if self.fused is not None:
                v = self.fused.core_derivs_vec(
                    corelz, active=self.execute_model).cpu().numpy()
                for m in range(NB):
                    if self.execute_model[m]:
                        dl[m], d2[m] = v[2 * m], v[2 * m + 1]
            else:
                outs = [(m, eng.core_derivs_async(float(corelz[m])))
                        for m, eng in enumerate(self.engines)
                        if self.execute_model[m]]
                if outs and not isinstance(outs[0][1], tuple):
                    import torch
                    vals = torch.stack([o[1] for o in outs]).cpu()
                    for k, (m, _) in enumerate(outs):
                        dl[m], d2[m] = float(vals[k][0]), float(vals[k][1])
                else:
                    for m, v in outs:
                        dl[m], d2[m] = v
            for i in range(NB):
                if not outer[i] and not curvat[i]:
                    if d2[i] >= 0.0 and z[i] < ZMAX:
                        zprev[i] = z[i] = 0.37 * z[i] + 0.63
                    else:
                        curvat[i] = True
            for i in range(NB):
                if curvat[i] and not outer[i]:
                    if d2[i] < 0.0:
                        tantmp = -dl[i] / d2[i]
                        if tantmp < 100:
                            z[i] *= math.exp(tantmp)
                            if z[i] < ZMIN:
                                z[i] = ZMIN
                            if z[i] > 0.25 * zprev[i] + 0.75:
                                z[i] = 0.25 * zprev[i] + 0.75
                        else:
                            z[i] = 0.25 * zprev[i] + 0.75
                    if z[i] > ZMAX:
                        z[i] = ZMAX
                    miter[i] -= 1
                    if abs(z[i] - zprev[i]) > zstep[i]:
                        if miter[i] < -20:
                            z[i] = z0[i]
                            outer[i] = True
                        else:
                            outer[i] = False
                    else:
                        outer[i] = True
            if outer.all():
                break
        self.execute_model = [True] * self.M
        return z

    def update(self, p, parent):
        """update(tr, p) with q = p->back = parent (searchAlgo.c:127)."""
        t = self.tree
        if self.NB == 1:
            z0 = t.get_z(p, parent)
            z = self.makenewz_generic(p, parent, z0, NEWZPERCYCLE)
            if not self.partition_converged:
                if abs(z - z0) > DELTAZ:
                    self.partition_smoothed = False
                t.set_z(p, parent, z)
        else:
            z0 = t.get_zv(p, parent).copy()
            z = self.makenewz_generic_vec(p, parent, z0, NEWZPERCYCLE,
                                          mask=True)
            zarr = t.adj[p][parent]  # aliased both directions
            for i in range(self.NB):
                if not self.partition_converged[i]:
                    if abs(z[i] - z0[i]) > DELTAZ:
                        self.partition_smoothed[i] = False
                    zarr[i] = z[i]

    def smooth(self, p, parent):
        """smooth(tr, p) with p->back == parent (searchAlgo.c:196)."""
        t = self.tree
        self.update(p, parent)
        if not t.is_tip(p):
            for w in self._children(p, parent):
                self.smooth(w, p)
            if self.NB > 1:
                # masked newview (newviewGenericSpecial.c:1559-1573)
                self.execute_model = [not c for c in self.partition_converged]
                self.newview_generic(p, parent)
                self.execute_model = [True] * self.M
            else:
                self.newview_generic(p, parent)

    def smooth_tree(self, maxtimes):
        """smoothTree (searchAlgo.c:237)."""
        t = self.tree
        p = self.start
        if self.NB == 1:
            self.partition_converged = False
            while maxtimes > 0:
                maxtimes -= 1
                self.partition_smoothed = True
                self.smooth(next(iter(t.adj[p])), p)  # smooth(tr, p->back)
                # p == tr->start is a tip: the second descent is skipped
                if self.partition_smoothed:  # allSmoothed
                    self.partition_converged = True
                    break
            self.partition_converged = False
        else:
            self.partition_converged[:] = False
            while maxtimes > 0:
                maxtimes -= 1
                self.partition_smoothed[:] = True
                self.smooth(next(iter(t.adj[p])), p)
                # allSmoothed (searchAlgo.c:222): flags converged partitions
                # even when the sweep as a whole is not yet smoothed
                result = True
                for i in range(self.NB):
                    if not self.partition_smoothed[i]:
                        result = False
                    else:
                        self.partition_converged[i] = True
                if result:
                    break
            self.partition_converged[:] = False

    def tree_evaluate(self, smooth_factor):
        """treeEvaluate (searchAlgo.c:2635)."""
        self.smooth_tree(int(SMOOTHINGS * smooth_factor))
        return self.evaluate_generic(full=True)

    # ------------------------------------------------------------------
    # model-parameter optimization (optimizeModel.c)
    # ------------------------------------------------------------------

    def _change_param(self, m, rate_number, value, which):
        """changeModelParameters (optimizeModel.c:419)."""
        eng = self.engines[m]
        model = eng.model
        if which == RATE_F:
            # setRateModel (optimizeModel.c:78); DNA: position < 5
            rates = model.rates6 if model.states == 4 else model.rates190
            assert 0 <= rate_number < len(rates) - (1 if model.states == 4
                                                    else 0)
            assert RATE_MIN <= value <= RATE_MAX
            rates[rate_number] = value
            model.reinit()
            eng.upload_model()
        elif which == ALPHA_F:
            model.set_alpha(value)
        elif which == LXRATE_F:
            # optimizeModel.c:451: gammaRates[rate]=value + scaleLG4X_EIGN
            model.set_lg4x_rate(rate_number, value)
            eng.upload_model()
        elif which == LXWEIGHT_F:
            # optimizeModel.c:455: updateWeights + scaleLG4X_EIGN
            model.set_weight_exponent(rate_number, value)
            eng.upload_model()
        elif which == FREQ_F:
            w = self.freq_exponents[m]
            w[rate_number] = value
            ew = np.exp(w)
            model.frequencies[:] = ew / ew.sum()
            model.reinit()
            eng.upload_model()
        else:
            raise AssertionError(which)

    def _evaluate_change(self, rate_number, values, converged, which,
                         groups, valid):
        """evaluateChange (optimizeModel.c:464): apply parameter probes,
        mask converged groups and invalid partitions, full-traversal
        evaluate, return -sum(perPartitionLH) per group."""
        in_valid_group = set()
        pos = 0
        for gi, g in enumerate(groups):
            if not valid[gi]:
                continue
            for m in g:
                in_valid_group.add(m)
                if converged[pos]:
                    self.execute_model[m] = False
                else:
                    self._change_param(m, rate_number, values[pos], which)
            pos += 1
        for m in range(self.M):
            if m not in in_valid_group:
                self.execute_model[m] = False
        self.evaluate_generic(full=True)
        results = np.zeros(pos)
        pos = 0
        for gi, g in enumerate(groups):
            if not valid[gi]:
                continue
            results[pos] = -sum(self.per_partition_lnl[m] for m in g)
            pos += 1
        self.execute_model = [True] * self.M
        return results

    def _brak(self, param, ax, bx, cx, fa, fb, fc, lim_inf, lim_sup,
              rate_number, which, groups, valid):
        """brakGeneric (optimizeModel.c:800), numpy over the valid groups."""
        n = len(ax)
        converged = np.zeros(n, dtype=bool)
        state = np.zeros(n, dtype=int)
        end_state = np.zeros(n, dtype=int)
        u = np.zeros(n)
        ulim = np.zeros(n)
        fu_state = np.zeros(n)  # the reference's persistent fu[] array

        np.clip(ax, lim_inf, lim_sup, out=ax)
        param[:] = ax
        fa[:] = self._evaluate_change(rate_number, param, converged, which,
                                      groups, valid)
        np.clip(bx, lim_inf, lim_sup, out=bx)
        param[:] = bx
        fb[:] = self._evaluate_change(rate_number, param, converged, which,
                                      groups, valid)
        for i in range(n):
            if fb[i] > fa[i]:
                ax[i], bx[i] = bx[i], ax[i]
                fa[i], fb[i] = fb[i], fa[i]
            cx[i] = bx[i] + MNBRAK_GOLD * (bx[i] - ax[i])
            cx[i] = min(max(cx[i], lim_inf[i]), lim_sup[i])
            param[i] = cx[i]
        fc[:] = self._evaluate_change(rate_number, param, converged, which,
                                      groups, valid)

        while True:
            if converged.all():
                np.clip(ax, lim_inf, lim_sup, out=ax)
                np.clip(bx, lim_inf, lim_sup, out=bx)
                np.clip(cx, lim_inf, lim_sup, out=cx)
                return
            for i in range(n):
                if converged[i]:
                    continue
                if state[i] == 0:
                    end_state[i] = 0
                    if not (fb[i] > fc[i]):
                        converged[i] = True
                    else:
                        ax[i] = min(max(ax[i], lim_inf[i]), lim_sup[i])
                        bx[i] = min(max(bx[i], lim_inf[i]), lim_sup[i])
                        cx[i] = min(max(cx[i], lim_inf[i]), lim_sup[i])
                        r_ = (bx[i] - ax[i]) * (fb[i] - fc[i])
                        q_ = (bx[i] - cx[i]) * (fb[i] - fa[i])
                        u[i] = bx[i] - ((bx[i] - cx[i]) * q_ -
                                        (bx[i] - ax[i]) * r_) / \
                            (2.0 * _sign(max(abs(q_ - r_), MNBRAK_TINY),
                                         q_ - r_))
                        ulim[i] = bx[i] + MNBRAK_GLIMIT * (cx[i] - bx[i])
                        u[i] = min(max(u[i], lim_inf[i]), lim_sup[i])
                        ulim[i] = min(max(ulim[i], lim_inf[i]), lim_sup[i])
                        if (bx[i] - u[i]) * (u[i] - cx[i]) > 0.0:
                            u[i] = min(max(u[i], lim_inf[i]), lim_sup[i])
                            param[i] = u[i]
                            end_state[i] = 1
                        elif (cx[i] - u[i]) * (u[i] - ulim[i]) > 0.0:
                            u[i] = min(max(u[i], lim_inf[i]), lim_sup[i])
                            param[i] = u[i]
                            end_state[i] = 2
                        elif (u[i] - ulim[i]) * (ulim[i] - cx[i]) >= 0.0:
                            u[i] = ulim[i]
                            u[i] = min(max(u[i], lim_inf[i]), lim_sup[i])
                            param[i] = u[i]
                            end_state[i] = 0
                        else:
                            u[i] = cx[i] + MNBRAK_GOLD * (cx[i] - bx[i])
                            u[i] = min(max(u[i], lim_inf[i]), lim_sup[i])
                            param[i] = u[i]
                            end_state[i] = 0
                elif state[i] == 1:
                    end_state[i] = 0
                elif state[i] == 2:
                    end_state[i] = 3
                else:
                    raise AssertionError
            temp = self._evaluate_change(rate_number, param, converged,
                                         which, groups, valid)
            for i in range(n):
                if converged[i]:
                    continue
                if end_state[i] == 0:
                    fu = fu_state[i] = temp[i]
                    ax[i], bx[i], cx[i] = bx[i], cx[i], u[i]
                    fa[i], fb[i], fc[i] = fb[i], fc[i], fu
                    state[i] = 0
                elif end_state[i] == 1:
                    fu = fu_state[i] = temp[i]
                    if fu < fc[i]:
                        ax[i] = bx[i]
                        bx[i] = u[i]
                        fa[i] = fb[i]
                        fb[i] = fu
                        converged[i] = True
                    elif fu > fb[i]:
                        cx[i] = u[i]
                        fc[i] = fu
                        converged[i] = True
                    else:
                        u[i] = cx[i] + MNBRAK_GOLD * (cx[i] - bx[i])
                        u[i] = min(max(u[i], lim_inf[i]), lim_sup[i])
                        param[i] = u[i]
                        state[i] = 1
                elif end_state[i] == 2:
                    fu = fu_state[i] = temp[i]
                    if fu < fc[i]:
                        # the reference's SHFT(bx,cx,u, cx+GOLD*(cx-bx)) is
                        # SEQUENTIAL: the new u is computed from the ALREADY
                        # SHIFTED bx/cx (= old cx/u), no f-shifts, and param
                        # is left at the old u — the state-2 -> endState-3
                        # round re-evaluates that same point
                        # (optimizeModel.c:1046-1050)
                        bx[i] = cx[i]
                        cx[i] = u[i]
                        u[i] = cx[i] + MNBRAK_GOLD * (cx[i] - bx[i])
                        state[i] = 2
                    else:
                        state[i] = 0
                        ax[i], bx[i], cx[i] = bx[i], cx[i], u[i]
                        fa[i], fb[i], fc[i] = fb[i], fc[i], fu
                elif end_state[i] == 3:
                    # net effect of the reference's three sequential SHFTs
                    # (optimizeModel.c:1060-1063): fa=old fc, fb=old fu
                    # (the carried-over value, == temp here), fc=temp
                    fa[i] = fc[i]
                    fb[i] = fu_state[i]
                    fc[i] = temp[i]
                    fu_state[i] = temp[i]
                    ax[i], bx[i], cx[i] = bx[i], cx[i], u[i]
                    state[i] = 0
                else:
                    raise AssertionError

    def _brent(self, ax, bx, cx, fb, tol, rate_number, which, groups, valid,
               lim_inf, lim_sup):
        """brentGeneric (optimizeModel.c:582); returns (xmin, result)."""
        n = len(ax)
        a = np.minimum(ax, cx)
        b = np.maximum(ax, cx)
        x = bx.copy()
        w = bx.copy()
        v = bx.copy()
        fw = fb.copy()
        fv = fb.copy()
        fx = fb.copy()
        d = np.zeros(n)
        e = np.zeros(n)
        u = np.zeros(n)
        xmin = np.zeros(n)
        result = np.zeros(n)
        converged = np.zeros(n, dtype=bool)

        for _ in range(1, ITMAX + 1):
            if converged.all():
                return xmin, result
            for i in range(n):
                if converged[i]:
                    continue
                xm = 0.5 * (a[i] + b[i])
                tol1 = tol * abs(x[i]) + BRENT_ZEPS
                tol2 = 2.0 * tol1
                if abs(x[i] - xm) <= (tol2 - 0.5 * (b[i] - a[i])):
                    result[i] = -fx[i]
                    xmin[i] = x[i]
                    converged[i] = True
                    continue
                if abs(e[i]) > tol1:
                    r_ = (x[i] - w[i]) * (fx[i] - fv[i])
                    q_ = (x[i] - v[i]) * (fx[i] - fw[i])
                    p_ = (x[i] - v[i]) * q_ - (x[i] - w[i]) * r_
                    q_ = 2.0 * (q_ - r_)
                    if q_ > 0.0:
                        p_ = -p_
                    q_ = abs(q_)
                    etemp = e[i]
                    e[i] = d[i]
                    if (abs(p_) >= abs(0.5 * q_ * etemp)
                            or p_ <= q_ * (a[i] - x[i])
                            or p_ >= q_ * (b[i] - x[i])):
                        e[i] = a[i] - x[i] if x[i] >= xm else b[i] - x[i]
                        d[i] = BRENT_CGOLD * e[i]
                    else:
                        d[i] = p_ / q_
                        u[i] = x[i] + d[i]
                        if u[i] - a[i] < tol2 or b[i] - u[i] < tol2:
                            d[i] = _sign(tol1, xm - x[i])
                else:
                    e[i] = a[i] - x[i] if x[i] >= xm else b[i] - x[i]
                    d[i] = BRENT_CGOLD * e[i]
                u[i] = (x[i] + d[i]) if abs(d[i]) >= tol1 \
                    else (x[i] + _sign(tol1, d[i]))
            fu = self._evaluate_change(rate_number, u, converged, which,
                                       groups, valid)
            for i in range(n):
                if converged[i]:
                    continue
                if fu[i] <= fx[i]:
                    if u[i] >= x[i]:
                        a[i] = x[i]
                    else:
                        b[i] = x[i]
                    v[i], w[i], x[i] = w[i], x[i], u[i]
                    fv[i], fw[i], fx[i] = fw[i], fx[i], fu[i]
                else:
                    if u[i] < x[i]:
                        a[i] = u[i]
                    else:
                        b[i] = u[i]
                    if fu[i] <= fw[i] or w[i] == x[i]:
                        v[i] = w[i]
                        w[i] = u[i]
                        fv[i] = fw[i]
                        fw[i] = fu[i]
                    elif fu[i] <= fv[i] or v[i] == x[i] or v[i] == w[i]:
                        v[i] = u[i]
                        fv[i] = fu[i]
        raise AssertionError("Too many iterations in BRENT")

    def _opt_param_generic(self, groups, valid, rate_number, lim_inf_s,
                           lim_sup_s, which, model_epsilon):
        """optParamGeneric (optimizeModel.c:1283)."""
        self.evaluate_generic(full=True)
        vg = [g for gi, g in enumerate(groups) if valid[gi]]
        n = len(vg)
        if n == 0:
            return
        start_values = np.zeros(n)
        start_lh = np.zeros(n)
        lim_inf = np.zeros(n)
        lim_sup = np.zeros(n)
        for pos, g in enumerate(vg):
            for m in g:
                start_lh[pos] += self.per_partition_lnl[m]
                model = self.engines[m].model
                if which == ALPHA_F:
                    lim_inf[pos], lim_sup[pos] = lim_inf_s, lim_sup_s
                    start_values[pos] = model.alpha
                elif which == RATE_F:
                    lim_inf[pos], lim_sup[pos] = lim_inf_s, lim_sup_s
                    rates = (model.rates6 if model.states == 4
                             else model.rates190)
                    start_values[pos] = rates[rate_number]
                elif which == LXRATE_F:
                    lim_inf[pos], lim_sup[pos] = lim_inf_s, lim_sup_s
                    start_values[pos] = model.gammaRates[rate_number]
                elif which == LXWEIGHT_F:
                    lim_inf[pos], lim_sup[pos] = lim_inf_s, lim_sup_s
                    start_values[pos] = model.weightExponents[rate_number]
                elif which == FREQ_F:
                    lim_inf[pos] = self._min_freq(m, rate_number, lim_inf_s)
                    lim_sup[pos] = self._max_freq(m, rate_number, lim_sup_s)
                    start_values[pos] = self.freq_exponents[m][rate_number]
        a = np.clip(start_values + 0.1, lim_inf, lim_sup)
        b = np.clip(start_values - 0.1, lim_inf, lim_sup)
        c = np.zeros(n)
        fa = np.zeros(n)
        fb = np.zeros(n)
        fc = np.zeros(n)
        param = np.zeros(n)
        self._brak(param, a, b, c, fa, fb, fc, lim_inf, lim_sup, rate_number,
                   which, groups, valid)
        xmin, end_lh = self._brent(a, b, c, fb, model_epsilon, rate_number,
                                   which, groups, valid, lim_inf, lim_sup)
        for pos, g in enumerate(vg):
            val = start_values[pos] if start_lh[pos] > end_lh[pos] \
                else xmin[pos]
            for m in g:
                self._change_param(m, rate_number, val, which)

    def _min_freq(self, m, which_freq, absolute_min):
        """minFreq (optimizeModel.c:1222)."""
        w = self.freq_exponents[m]
        c = sum(math.exp(w[i]) for i in range(len(w)) if i != which_freq)
        return max(math.log(FREQ_MIN) + math.log(c) - math.log(1.0 -
                                                               FREQ_MIN),
                   absolute_min)

    def _max_freq(self, m, which_freq, absolute_max):
        """maxFreq (optimizeModel.c:1248): symmetric cap so one frequency
        cannot exceed 1 - (states-1)*FREQ_MIN."""
        w = self.freq_exponents[m]
        states = len(w)
        c = sum(math.exp(w[i]) for i in range(states) if i != which_freq)
        return min(math.log(1.0 - (states - 1) * FREQ_MIN) + math.log(c)
                   - math.log((states - 1) * FREQ_MIN), absolute_max)

    def opt_rates_generic(self, model_epsilon):
        """optRatesGeneric + optRates (optimizeModel.c:1634/1603):
        per-partition (unlinked) groups; DNA rate numbers 0..4."""
        groups = [[m] for m in range(self.M)]
        dna = [self.engines[m].model.states == 4 for m in range(self.M)]
        if any(dna):
            n_rates = (4 * 4 - 4) // 2 - 1
            for rn in range(n_rates):
                self._opt_param_generic(groups, dna, rn, RATE_MIN, RATE_MAX,
                                        RATE_F, model_epsilon)
        # AA GTR rate optimization only applies to protModels==GTR
        # partitions (AAisGTR, optimizeModel.c:1614) — LG et al. are fixed.

    def opt_alphas_generic(self, model_epsilon):
        """optAlphasGeneric (optimizeModel.c:1136): plain-alpha partitions
        via ALPHA_F, then LG4X partitions via optLG4X (LXRATE_F sweeps +
        weight optimization)."""
        groups = [[m] for m in range(self.M)]
        is_lg4x = [getattr(self.engines[m].model, "lg4x", False)
                   for m in range(self.M)]
        non_lg4x = [not b for b in is_lg4x]
        if any(non_lg4x):
            self._opt_param_generic(groups, non_lg4x, -1, ALPHA_MIN,
                                    ALPHA_MAX, ALPHA_F, model_epsilon)
        if any(is_lg4x):
            self._opt_lg4x(model_epsilon, groups, is_lg4x)

    def _opt_lg4x(self, model_epsilon, groups, valid):
        """optLG4X (optimizeModel.c:1116)."""
        for i in range(4):
            self._opt_param_generic(groups, valid, i, LG4X_RATE_MIN,
                                    LG4X_RATE_MAX, LXRATE_F, model_epsilon)
            self._optimize_weights(model_epsilon, groups, valid)

    def _optimize_weights(self, model_epsilon, groups, valid):
        """optimizeWeights (optimizeModel.c:389)."""
        self.evaluate_generic(full=True)
        initial = self.likelihood
        for i in range(4):
            self._opt_param_generic(groups, valid, i, -1000000.0, 200.0,
                                    LXWEIGHT_F, model_epsilon)
        self.evaluate_generic(full=True)
        assert self.likelihood >= initial - 1e-9


    def opt_base_freqs(self, model_epsilon):
        """optBaseFreqs + optFreqs (optimizeModel.c:1501/1594)."""
        groups = [[m] for m in range(self.M)]
        dna_valid = [self.engines[m].model.states == 4
                     and self.opt_freq_flags[m] for m in range(self.M)]
        if any(dna_valid):
            for rn in range(4):
                self._opt_param_generic(groups, dna_valid, rn, -1000000.0,
                                        200.0, FREQ_F, model_epsilon)
        aa_valid = [self.engines[m].model.states == 20
                    and self.opt_freq_flags[m] for m in range(self.M)]
        if any(aa_valid):
            for rn in range(20):
                self._opt_param_generic(groups, aa_valid, rn, -1000000.0,
                                        200.0, FREQ_F, model_epsilon)

    # -- AUTO protein model selection (optimizeModel.c:2606-2900) ----------

    def _aa_table(self):
        import os
        if not hasattr(self, "_aa_data"):
            self._aa_data = np.load(os.path.join(
                os.path.dirname(os.path.abspath(__file__)), "data",
                "aa_models.npz"))
        return self._aa_data

    def _set_auto_matrix(self, m, index, prot_freqs):
        """autoProtModels = index; protFreqs semantics for AUTO partitions:
        1 = the matrix's own frequencies, 0 = empirical
        (models.c:3528-3534)."""
        tab = self._aa_table()
        freqs = tab["frequencies"][index] if prot_freqs \
            else self.empirical_freqs[m]
        self.engines[m].model.set_matrix(tab["rates190"][index], freqs)
        self.engines[m].upload_model()
        self.auto_prot_models[m] = index
        self.prot_freqs[m] = prot_freqs

    def reset_branches(self):
        """resetBranches (optimizeModel.c:2509): every z to defaultz."""
        from .tree import DEFAULTZ
        for a, b in self.tree.edges():
            self.tree.set_z(a, b, DEFAULTZ)

    def _save_tree_z(self):
        return {e: self.tree.get_z(*e) for e in self.tree.edges()}

    def _restore_tree_z(self, saved):
        for (a, b), z in saved.items():
            self.tree.set_z(a, b, z)

    def _opt_model_pass(self, fixed_freqs):
        """optModel (optimizeModel.c:2606): try every candidate matrix on
        all AUTO partitions simultaneously; returns (bestIndex, bestScores)
        per partition."""
        best_index = [-1] * self.M
        best_scores = [UNLIKELY] * self.M
        for i in range(19):  # AUTO = 19 candidate matrices (axml.h:261)
            for m in range(self.M):
                if self.auto_flags[m]:
                    self._set_auto_matrix(m, i, 1 if fixed_freqs else 0)
            self.reset_branches()
            self.evaluate_generic(full=True)
            self.tree_evaluate(0.5)
            for m in range(self.M):
                if self.auto_flags[m] and \
                        self.per_partition_lnl[m] > best_scores[m]:
                    best_scores[m] = self.per_partition_lnl[m]
                    best_index[m] = i
        return best_index, best_scores

    def auto_protein(self, log=None):
        """autoProtein (optimizeModel.c:2669), ML criterion (the default,
        axml.c:982)."""
        if not any(self.auto_flags):
            return
        saved_z = self._save_tree_z()
        old_index = list(self.auto_prot_models)
        old_freqs = list(self.prot_freqs)
        self.evaluate_generic(full=True)
        start_lh = self.likelihood
        bi_fixed, bs_fixed = self._opt_model_pass(fixed_freqs=True)
        bi_emp, bs_emp = self._opt_model_pass(fixed_freqs=False)
        for m in range(self.M):
            if not self.auto_flags[m]:
                continue
            if bs_fixed[m] > bs_emp[m]:  # AUTO_ML
                self._set_auto_matrix(m, bi_fixed[m], 1)
            else:
                self._set_auto_matrix(m, bi_emp[m], 0)
            if log:
                tab = self._aa_table()
                log(f"AUTO partition {m}: "
                    f"{tab['names'][self.auto_prot_models[m]]} "
                    f"({'fixed' if self.prot_freqs[m] else 'empirical'} "
                    f"freqs)")
        self.reset_branches()
        self.evaluate_generic(full=True)
        self.tree_evaluate(2.0)
        if self.likelihood < start_lh:
            for m in range(self.M):
                if self.auto_flags[m]:
                    self._set_auto_matrix(m, old_index[m], old_freqs[m])
            self._restore_tree_z(saved_z)
            self.evaluate_generic(full=True)
        assert self.likelihood >= start_lh - 1e-6

    # -- CAT per-site rate optimization (optimizeModel.c:2403) -------------

    def _evaluate_partial(self, m, site, ki):
        """evaluatePartialGeneric (evaluatePartialGenericSpecial.c:259) via
        the host C implementation over the last full traversal."""
        import ctypes
        from . import lib
        eng = self.engines[m]
        entries, p, q, zv = self._last_full
        z = float(zv[m]) if self.NB > 1 else float(zv[0])
        # cache keyed by IDENTITY of the live entries list (a strong ref is
        # kept so the id cannot be recycled) + partition for -M
        mkey = m if self.NB > 1 else 0
        if (not hasattr(self, "_epg_ops") or self._epg_ops[1] is not entries
                or self._epg_ops[2] != mkey):
            src = entries if self.NB == 1 \
                else self._per_partition_entries(entries, m)
            arr = (TravEntry * len(src))(*src)
            self._epg_ops = (arr, entries, mkey)
        arr = self._epg_ops[0]
        fn = (lib().examl_host_evaluate_partial_dna_cat
              if eng.model.states == 4
              else lib().examl_host_evaluate_partial_prot_cat)
        return fn(
            ctypes.cast(arr, ctypes.c_void_p), len(entries),
            ctypes.c_int(p), ctypes.c_int(q), ctypes.c_double(z),
            ctypes.c_long(site), ctypes.c_double(ki),
            ctypes.c_int(int(eng.host_wgt[site])),
            eng.model.EIGN.ctypes.data_as(ctypes.c_void_p),
            eng.model.EI.ctypes.data_as(ctypes.c_void_p),
            eng.model.EV.ctypes.data_as(ctypes.c_void_p),
            eng.model.tipVector.ctypes.data_as(ctypes.c_void_p),
            eng.host_tips.ctypes.data_as(ctypes.c_void_p),
            ctypes.c_long(eng.width), ctypes.c_int(self.tree.ntips))

    def optimize_rate_categories(self, max_categories, log=None):
        """optimizeRateCategories (optimizeModel.c:2403): per-site rate
        search (optRateCatPthreads :1798), clustering
        (categorizeTheRates :2171), mean-1 rescale (updatePerSiteRates
        :2060) and restore-on-regression."""
        if max_categories == 1:
            return
        initial_lh = self.likelihood
        self.evaluate_generic(full=True)
        inv = self.rate_cat_invocations
        if inv == 1:
            lower_spacing, upper_spacing = 0.5 / inv, 1.0 / inv
        else:
            lower_spacing, upper_spacing = 0.05 / inv, 0.1 / inv
        lower_spacing = max(lower_spacing, 0.001)
        upper_spacing = max(upper_spacing, 0.001)
        self.rate_cat_invocations += 1

        backup = [(self.cat_patrat[m].copy(),
                   self.engines[m].per_site_rates.copy(),
                   self.engines[m].cptr.copy(),
                   self.engines[m].num_cats) for m in range(self.M)]

        eps = 0.00001
        for m, eng in enumerate(self.engines):
            patrat = self.cat_patrat[m]
            lhs = self.cat_lhs[m]
            for i in range(eng.width):
                r0 = patrat[i]
                l0 = self._evaluate_partial(m, i, r0)
                left_lh = right_lh = l0
                left_rate = right_rate = r0
                k = 1
                while True:
                    if not (r0 - k * lower_spacing > 0.0001):
                        break
                    v = self._evaluate_partial(m, i, r0 - k * lower_spacing)
                    if not (v > left_lh and abs(left_lh - v) > eps):
                        break
                    left_lh = v
                    left_rate = r0 - k * lower_spacing
                    k += 1
                k = 1
                while True:
                    v = self._evaluate_partial(m, i, r0 + k * upper_spacing)
                    if not (v > right_lh and abs(right_lh - v) > eps):
                        break
                    right_lh = v
                    right_rate = r0 + k * upper_spacing
                    k += 1
                if right_lh > l0 or left_lh > l0:
                    if right_lh > left_lh:
                        patrat[i] = right_rate
                        lhs[i] = right_lh
                    else:
                        patrat[i] = left_rate
                        lhs[i] = left_lh
                else:
                    lhs[i] = l0

        # categorizeTheRates (:2171) per partition
        for m, eng in enumerate(self.engines):
            patrat = self.cat_patrat[m]
            lhs = self.cat_lhs[m]
            rates = [patrat[0]]
            acc = [lhs[0]]
            for i in range(1, eng.width):
                temp = patrat[i]
                found = False
                for k in range(len(rates)):
                    if temp == rates[k] or abs(temp - rates[k]) < 0.001:
                        acc[k] += lhs[i]
                        found = True
                        break
                if not found:
                    rates.append(temp)
                    acc.append(lhs[i])
            order = sorted(range(len(rates)), key=lambda k: acc[k])
            rc_rates = [rates[k] for k in order]
            num = min(len(rc_rates), max_categories)
            # categorizePartition (:1733): exact/0.001 match among the kept
            # categories, else nearest
            new_cptr = np.zeros(eng.width, dtype=np.int32)
            for i in range(eng.width):
                temp = patrat[i]
                found = False
                for k in range(num):
                    if temp == rc_rates[k] or abs(temp - rc_rates[k]) < 0.001:
                        new_cptr[i] = k
                        found = True
                        break
                if not found:
                    best, bmin = 0, abs(temp - rc_rates[0])
                    for k in range(1, num):
                        d = abs(temp - rc_rates[k])
                        if d < bmin:
                            bmin, best = d, k
                    new_cptr[i] = best
            eng.set_site_rates(new_cptr, np.array(rc_rates[:num]))

        # updatePerSiteRates (:2060): per-partition mean-1 rescale under
        # -M (numBranches > 1), one global scaler otherwise
        if self.NB > 1:
            for eng in self.engines:
                w = eng.host_wgt
                scaler = float(w.sum()) / \
                    float((w * eng.per_site_rates[eng.cptr]).sum())
                eng.set_site_rates(eng.cptr, eng.per_site_rates * scaler)
        else:
            wsum = rsum = 0.0
            for m, eng in enumerate(self.engines):
                w = eng.host_wgt
                rsum += float((w * eng.per_site_rates[eng.cptr]).sum())
                wsum += float(w.sum())
            scaler = 1.0 / (rsum / wsum)
            for eng in self.engines:
                eng.set_site_rates(eng.cptr, eng.per_site_rates * scaler)

        self.evaluate_generic(full=True)
        if self.likelihood < initial_lh:
            for m, eng in enumerate(self.engines):
                pat, rates_b, cptr_b, _ = backup[m]
                self.cat_patrat[m][:] = pat
                eng.set_site_rates(cptr_b, rates_b)
            self.evaluate_generic(full=True)
            assert abs(self.likelihood - initial_lh) < 1e-6
        if log:
            log(f"rate categories: lnL {initial_lh:.6f} -> "
                f"{self.likelihood:.6f} "
                f"({[e.num_cats for e in self.engines]})")

    def mod_opt(self, likelihood_epsilon=0.1, model_epsilon=0.0001,
                log=None):
        """modOpt (optimizeModel.c:2963) for the GAMMA and CAT models."""
        cat_opt = 0
        self.evaluate_generic(full=True)
        while True:
            current = self.likelihood
            self.opt_rates_generic(model_epsilon)
            self.evaluate_generic(full=True)
            self.auto_protein(log=log)
            self.tree_evaluate(0.0625)
            self.evaluate_generic(full=True)
            self.opt_base_freqs(model_epsilon)
            self.evaluate_generic(full=True)
            self.tree_evaluate(0.0625)
            if self.rate_het == "GAMMA":
                self.opt_alphas_generic(model_epsilon)
                self.evaluate_generic(full=True)
                self.tree_evaluate(0.1)
            elif self.rate_het == "CAT":
                if cat_opt < 3:
                    self.evaluate_generic(full=True)
                    self.optimize_rate_categories(self.max_categories,
                                                  log=log)
                    cat_opt += 1
            else:
                raise AssertionError(self.rate_het)
            if log:
                log(f"modOpt pass: {current:.6f} -> {self.likelihood:.6f}")
            if abs(current - self.likelihood) <= likelihood_epsilon:
                break
        return self.likelihood

    def tree_evaluation_mode(self, log=None, epsilon=0.1):
        """The -f E (slow TREE_EVALUATION) flow for one tree
        (axml.c:2316-2331): evaluate, treeEvaluate(1), modOpt with
        adef->likelihoodEpsilon (-e, default 0.1)."""
        self.evaluate_generic(full=True)
        if log:
            log(f"initial lnL = {self.likelihood:.6f}")
        self.tree_evaluate(1.0)
        if log:
            log(f"after treeEvaluate = {self.likelihood:.6f}")
        return self.mod_opt(epsilon, log=log)


def evaluate_trees(trees, engines, fast=False, log=None, epsilon=0.1,
                   **search_kwargs):
    """optimizeTrees (axml.c:2721) for the -f E (slow) / -f e (fast)
    multi-tree input: tree 0 gets the full treeEvaluate(1)+modOpt(0.1),
    later trees get resetBranches and -- in fast mode -- only
    treeEvaluate(2); model parameters persist across trees (the
    reference keeps them in partitionData).  Returns the per-tree lnLs."""
    out = []
    for i, tree in enumerate(trees):
        ts = TreeSearch(tree, engines, **search_kwargs)
        if i > 0:
            ts.reset_branches()
        ts.evaluate_generic(full=True)
        if fast and i > 0:
            ts.tree_evaluate(2.0)
        else:
            ts.tree_evaluate(1.0)
            ts.mod_opt(epsilon, log=log)
        out.append(ts.likelihood)
    return out
