"""Shared test helpers: synthetic data and the ORACLE-side executor that
replays a product traversal on the CPU restatement (the checker for GPU
parity and host-logic tests)."""

import math

import numpy as np

import oracle as O
from examl_amd import INNER_INNER, TIP_INNER, TIP_TIP


def make_synthetic(ntips, width, seed=42):
    """Seeded synthetic alignment (delegates to the product generator)."""
    from examl_amd.synthetic import make_alignment
    return make_alignment(ntips, width, seed=seed)


def _model_arrays(model):
    """Aligned copies of a DnaGtrModel's vectors for the oracle's AVX-path
    reference calls."""
    def al(a):
        out = O.aligned(a.shape)
        out[:] = a
        return out
    return (al(model.EIGN), al(model.EV), al(model.EI), al(model.tipVector),
            al(model.gammaRates))


def oracle_full_lnl(entries, root, tree, model, tips, wgt,
                    return_state=False):
    """Replay the product's traversal entries through the CPU oracle
    (newview per entry + recursive scalers + evaluate at the root),
    restating evaluateGeneric end to end.  model.states picks DNA/protein."""
    EIGN, EV, EI, tipVector, g = _model_arrays(model)
    st = model.states
    nv = O.newview_dna_gamma if st == 4 else O.newview_prot_gamma
    ev = O.evaluate_dna_gamma if st == 4 else O.evaluate_prot_gamma
    width = tips.shape[1]
    ntips = tips.shape[0] - 1
    clv = {}
    scalers = np.zeros(2 * ntips, dtype=np.int64)
    wgt = np.ascontiguousarray(wgt, dtype=np.int32)

    for e in entries:
        qz = math.log(e.qz) if e.qz > O.ZMIN else math.log(O.ZMIN)
        rz = math.log(e.rz) if e.rz > O.ZMIN else math.log(O.ZMIN)
        left, right = O.make_p(qz, rz, g, EI, EIGN, 4, st)
        if e.tipCase == TIP_TIP:
            x3, inc = nv(
                TIP_TIP, None, None, EV, tipVector,
                np.ascontiguousarray(tips[e.x1Slot]),
                np.ascontiguousarray(tips[e.x2Slot]), width, left, right, wgt)
        elif e.tipCase == TIP_INNER:
            x3, inc = nv(
                TIP_INNER, None, clv[e.x2Slot], EV, tipVector,
                np.ascontiguousarray(tips[e.x1Slot]), None, width, left,
                right, wgt)
        else:
            x3, inc = nv(
                INNER_INNER, clv[e.x1Slot], clv[e.x2Slot], EV, tipVector,
                None, None, width, left, right, wgt)
        clv[e.x3Slot] = x3
        scalers[e.pNumber] = scalers[e.qNumber] + scalers[e.rNumber] + inc

    p, q, z = root
    diag = O.calc_diagptable(z, st, 4, g, EIGN)
    p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
    if q_tip:
        lnl = ev(wgt, None, clv[tree.clv_slot(p)],
                 tipVector, np.ascontiguousarray(tips[q]), width, diag)
    elif p_tip:
        lnl = ev(wgt, None, clv[tree.clv_slot(q)],
                 tipVector, np.ascontiguousarray(tips[p]), width, diag)
    else:
        lnl = ev(wgt, clv[tree.clv_slot(p)],
                 clv[tree.clv_slot(q)], tipVector, None, width, diag)
    lnl += float(scalers[p] + scalers[q]) * math.log(O.MINLIKELIHOOD)
    if return_state:
        return lnl, clv, scalers
    return lnl


def make_synthetic_aa(ntips, width, seed=42):
    """Seeded synthetic protein alignment (codes 1..22)."""
    rng = np.random.default_rng(seed)
    tips = np.zeros((ntips + 1, width), dtype=np.uint8)
    base = rng.integers(1, 21, width).astype(np.uint8)
    for t in range(1, ntips + 1):
        row = base.copy()
        mut = rng.random(width) < 0.15
        row[mut] = rng.integers(1, 21, int(mut.sum())).astype(np.uint8)
        amb = rng.random(width) < 0.01
        row[amb] = rng.integers(1, 23, int(amb.sum())).astype(np.uint8)
        tips[t] = row
    wgt = np.ones(width, dtype=np.int32)
    return tips, wgt


class OracleEngine:
    """CPU engine with the SAME interface as DnaGammaEngine, executing the
    oracle kernels — lets examl_amd.search.TreeSearch run end to end on the
    CPU restatement (the checker for the GPU search path, and the vehicle
    for the testData/49 -f E parity anchor)."""

    def __init__(self, tips, wgt, model):
        self.model = model
        self.ntips = tips.shape[0] - 1
        self.width = tips.shape[1]
        self.tips = tips
        self.wgt = np.ascontiguousarray(wgt, dtype=np.int32)
        self.host_tips = np.ascontiguousarray(tips)
        self.host_wgt = self.wgt
        self.clv = {}
        self.scalers = np.zeros(2 * self.ntips, dtype=np.int64)
        self.sumtable = None
        self._nv = O.newview_dna_gamma if model.states == 4 \
            else O.newview_prot_gamma
        self._ev = O.evaluate_dna_gamma if model.states == 4 \
            else O.evaluate_prot_gamma
        self._sum = O.sum_dna_gamma if model.states == 4 else O.sum_prot_gamma
        self._core = O.core_dna_gamma if model.states == 4 \
            else O.core_prot_gamma

    def upload_model(self):
        pass  # oracle reads model arrays directly per call

    def _arrays(self):
        return _model_arrays(self.model)

    def newview_traversal(self, entries):
        EIGN, EV, EI, tipVector, g = self._arrays()
        st = self.model.states
        for e in entries:
            qz = math.log(e.qz) if e.qz > O.ZMIN else math.log(O.ZMIN)
            rz = math.log(e.rz) if e.rz > O.ZMIN else math.log(O.ZMIN)
            left, right = O.make_p(qz, rz, g, EI, EIGN, 4, st)
            if e.tipCase == TIP_TIP:
                x3, inc = self._nv(
                    TIP_TIP, None, None, EV, tipVector,
                    np.ascontiguousarray(self.tips[e.x1Slot]),
                    np.ascontiguousarray(self.tips[e.x2Slot]), self.width,
                    left, right, self.wgt)
            elif e.tipCase == TIP_INNER:
                x3, inc = self._nv(
                    TIP_INNER, None, self.clv[e.x2Slot], EV, tipVector,
                    np.ascontiguousarray(self.tips[e.x1Slot]), None,
                    self.width, left, right, self.wgt)
            else:
                x3, inc = self._nv(
                    INNER_INNER, self.clv[e.x1Slot], self.clv[e.x2Slot], EV,
                    tipVector, None, None, self.width, left, right, self.wgt)
            self.clv[e.x3Slot] = x3
            self.scalers[e.pNumber] = (self.scalers[e.qNumber] +
                                       self.scalers[e.rNumber] + inc)

    def evaluate_root(self, tree, p, q, z):
        EIGN, EV, EI, tipVector, g = self._arrays()
        st = self.model.states
        diag = O.calc_diagptable(z, st, 4, g, EIGN)
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if q_tip:
            lnl = self._ev(self.wgt, None, self.clv[tree.clv_slot(p)],
                           tipVector, np.ascontiguousarray(self.tips[q]),
                           self.width, diag)
        elif p_tip:
            lnl = self._ev(self.wgt, None, self.clv[tree.clv_slot(q)],
                           tipVector, np.ascontiguousarray(self.tips[p]),
                           self.width, diag)
        else:
            lnl = self._ev(self.wgt, self.clv[tree.clv_slot(p)],
                           self.clv[tree.clv_slot(q)], tipVector, None,
                           self.width, diag)
        lnl += float(self.scalers[p] + self.scalers[q]) * \
            math.log(O.MINLIKELIHOOD)
        return lnl

    def sum_root(self, tree, p, q):
        EIGN, EV, EI, tipVector, g = self._arrays()
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if p_tip and q_tip:
            self.sumtable = self._sum(
                TIP_TIP, None, None, tipVector,
                np.ascontiguousarray(self.tips[p]),
                np.ascontiguousarray(self.tips[q]), self.width)
        elif q_tip:
            self.sumtable = self._sum(
                TIP_INNER, None, self.clv[tree.clv_slot(p)], tipVector,
                np.ascontiguousarray(self.tips[q]), None, self.width)
        elif p_tip:
            self.sumtable = self._sum(
                TIP_INNER, None, self.clv[tree.clv_slot(q)], tipVector,
                np.ascontiguousarray(self.tips[p]), None, self.width)
        else:
            self.sumtable = self._sum(
                INNER_INNER, self.clv[tree.clv_slot(p)],
                self.clv[tree.clv_slot(q)], tipVector, None, None,
                self.width)

    def core_derivs(self, lz):
        EIGN, EV, EI, tipVector, g = self._arrays()
        return self._core(self.width, self.sumtable, EIGN, g, lz, self.wgt)

    def core_derivs_async(self, lz):
        return self.core_derivs(lz)


class OracleCatEngine(OracleEngine):
    """CPU CAT (PSR) engine with the DnaCatEngine interface — lets
    TreeSearch(rate_het="CAT") run fully on the oracle kernels (the checker
    for the GPU PSR path and the vehicle for the 49 -f E -m PSR anchor)."""

    def __init__(self, tips, wgt, model, cptr, per_site_rates):
        super().__init__(tips, wgt, model)
        self.cptr = np.ascontiguousarray(cptr, dtype=np.int32)
        self.per_site_rates = np.ascontiguousarray(per_site_rates,
                                                   dtype=np.float64)
        self.num_cats = len(self.per_site_rates)

    def set_site_rates(self, cptr, per_site_rates):
        self.cptr = np.ascontiguousarray(cptr, dtype=np.int32)
        self.per_site_rates = np.ascontiguousarray(per_site_rates,
                                                   dtype=np.float64)
        self.num_cats = len(self.per_site_rates)

    def _rptr(self):
        r = O.aligned(self.num_cats)
        r[:] = self.per_site_rates
        return r

    def newview_traversal(self, entries):
        EIGN, EV, EI, tipVector, _ = self._arrays()
        rptr = self._rptr()
        for e in entries:
            qz = math.log(e.qz) if e.qz > O.ZMIN else math.log(O.ZMIN)
            rz = math.log(e.rz) if e.rz > O.ZMIN else math.log(O.ZMIN)
            left, right = O.make_p(qz, rz, rptr, EI, EIGN, self.num_cats, 4)
            if e.tipCase == TIP_TIP:
                x3, inc = O.newview_dna_cat(
                    TIP_TIP, EV, self.cptr, None, None, tipVector,
                    np.ascontiguousarray(self.tips[e.x1Slot]),
                    np.ascontiguousarray(self.tips[e.x2Slot]), self.width,
                    left, right, self.wgt)
            elif e.tipCase == TIP_INNER:
                x3, inc = O.newview_dna_cat(
                    TIP_INNER, EV, self.cptr, None, self.clv[e.x2Slot],
                    tipVector, np.ascontiguousarray(self.tips[e.x1Slot]),
                    None, self.width, left, right, self.wgt)
            else:
                x3, inc = O.newview_dna_cat(
                    INNER_INNER, EV, self.cptr, self.clv[e.x1Slot],
                    self.clv[e.x2Slot], tipVector, None, None, self.width,
                    left, right, self.wgt)
            self.clv[e.x3Slot] = x3
            self.scalers[e.pNumber] = (self.scalers[e.qNumber] +
                                       self.scalers[e.rNumber] + inc)

    def evaluate_root(self, tree, p, q, z):
        EIGN, EV, EI, tipVector, _ = self._arrays()
        diag = O.calc_diagptable(z, 4, self.num_cats, self._rptr(), EIGN)
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if q_tip:
            lnl = O.evaluate_dna_cat(
                self.cptr, self.wgt, None, self.clv[tree.clv_slot(p)],
                tipVector, np.ascontiguousarray(self.tips[q]), self.width,
                diag)
        elif p_tip:
            lnl = O.evaluate_dna_cat(
                self.cptr, self.wgt, None, self.clv[tree.clv_slot(q)],
                tipVector, np.ascontiguousarray(self.tips[p]), self.width,
                diag)
        else:
            lnl = O.evaluate_dna_cat(
                self.cptr, self.wgt, self.clv[tree.clv_slot(p)],
                self.clv[tree.clv_slot(q)], tipVector, None, self.width,
                diag)
        lnl += float(self.scalers[p] + self.scalers[q]) *             math.log(O.MINLIKELIHOOD)
        return lnl

    def sum_root(self, tree, p, q):
        EIGN, EV, EI, tipVector, _ = self._arrays()
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if p_tip and q_tip:
            self.sumtable = O.sum_dna_cat(
                TIP_TIP, None, None, tipVector,
                np.ascontiguousarray(self.tips[p]),
                np.ascontiguousarray(self.tips[q]), self.width)
        elif q_tip:
            self.sumtable = O.sum_dna_cat(
                TIP_INNER, None, self.clv[tree.clv_slot(p)], tipVector,
                np.ascontiguousarray(self.tips[q]), None, self.width)
        elif p_tip:
            self.sumtable = O.sum_dna_cat(
                TIP_INNER, None, self.clv[tree.clv_slot(q)], tipVector,
                np.ascontiguousarray(self.tips[p]), None, self.width)
        else:
            self.sumtable = O.sum_dna_cat(
                INNER_INNER, self.clv[tree.clv_slot(p)],
                self.clv[tree.clv_slot(q)], tipVector, None, None,
                self.width)

    def core_derivs(self, lz):
        EIGN, EV, EI, tipVector, _ = self._arrays()
        return O.core_dna_cat(self.width, self.num_cats, self.sumtable,
                              self.wgt, self._rptr(), EIGN, self.cptr, lz)


class OracleProtCatEngine(OracleCatEngine):
    """CPU protein CAT (PSR) engine over the oracle prot-CAT kernels."""

    def newview_traversal(self, entries):
        EIGN, EV, EI, tipVector, _ = self._arrays()
        rptr = self._rptr()
        for e in entries:
            qz = math.log(e.qz) if e.qz > O.ZMIN else math.log(O.ZMIN)
            rz = math.log(e.rz) if e.rz > O.ZMIN else math.log(O.ZMIN)
            left, right = O.make_p(qz, rz, rptr, EI, EIGN, self.num_cats, 20)
            if e.tipCase == TIP_TIP:
                x3, inc = O.newview_prot_cat(
                    TIP_TIP, EV, self.cptr, None, None, tipVector,
                    np.ascontiguousarray(self.tips[e.x1Slot]),
                    np.ascontiguousarray(self.tips[e.x2Slot]), self.width,
                    left, right, self.wgt)
            elif e.tipCase == TIP_INNER:
                x3, inc = O.newview_prot_cat(
                    TIP_INNER, EV, self.cptr, None, self.clv[e.x2Slot],
                    tipVector, np.ascontiguousarray(self.tips[e.x1Slot]),
                    None, self.width, left, right, self.wgt)
            else:
                x3, inc = O.newview_prot_cat(
                    INNER_INNER, EV, self.cptr, self.clv[e.x1Slot],
                    self.clv[e.x2Slot], tipVector, None, None, self.width,
                    left, right, self.wgt)
            self.clv[e.x3Slot] = x3
            self.scalers[e.pNumber] = (self.scalers[e.qNumber] +
                                       self.scalers[e.rNumber] + inc)

    def evaluate_root(self, tree, p, q, z):
        EIGN, EV, EI, tipVector, _ = self._arrays()
        diag = O.calc_diagptable(z, 20, self.num_cats, self._rptr(), EIGN)
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if q_tip:
            lnl = O.evaluate_prot_cat(
                self.cptr, self.wgt, None, self.clv[tree.clv_slot(p)],
                tipVector, np.ascontiguousarray(self.tips[q]), self.width,
                diag)
        elif p_tip:
            lnl = O.evaluate_prot_cat(
                self.cptr, self.wgt, None, self.clv[tree.clv_slot(q)],
                tipVector, np.ascontiguousarray(self.tips[p]), self.width,
                diag)
        else:
            lnl = O.evaluate_prot_cat(
                self.cptr, self.wgt, self.clv[tree.clv_slot(p)],
                self.clv[tree.clv_slot(q)], tipVector, None, self.width,
                diag)
        lnl += float(self.scalers[p] + self.scalers[q]) * \
            math.log(O.MINLIKELIHOOD)
        return lnl

    def sum_root(self, tree, p, q):
        EIGN, EV, EI, tipVector, _ = self._arrays()
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if p_tip and q_tip:
            self.sumtable = O.sum_prot_cat(
                TIP_TIP, None, None, tipVector,
                np.ascontiguousarray(self.tips[p]),
                np.ascontiguousarray(self.tips[q]), self.width)
        elif q_tip:
            self.sumtable = O.sum_prot_cat(
                TIP_INNER, None, self.clv[tree.clv_slot(p)], tipVector,
                np.ascontiguousarray(self.tips[q]), None, self.width)
        elif p_tip:
            self.sumtable = O.sum_prot_cat(
                TIP_INNER, None, self.clv[tree.clv_slot(q)], tipVector,
                np.ascontiguousarray(self.tips[p]), None, self.width)
        else:
            self.sumtable = O.sum_prot_cat(
                INNER_INNER, self.clv[tree.clv_slot(p)],
                self.clv[tree.clv_slot(q)], tipVector, None, None,
                self.width)

    def core_derivs(self, lz):
        EIGN, EV, EI, tipVector, _ = self._arrays()
        return O.core_prot_cat(self.width, self.num_cats, self.sumtable,
                               self.wgt, self._rptr(), EIGN, self.cptr, lz)


def oracle_makenewz(entries, root, tree, model, tips, wgt, z0, maxiter=64):
    """CPU restatement of topLevelMakenewz (numBranches=1) over the oracle
    sum/core kernels — the checker for DnaGammaEngine.makenewz."""
    EIGN, EV, EI, tipVector, g = _model_arrays(model)
    states = model.states
    sum_fn = O.sum_dna_gamma if states == 4 else O.sum_prot_gamma
    core_fn = O.core_dna_gamma if states == 4 else O.core_prot_gamma
    width = tips.shape[1]
    _, clv, _ = oracle_full_lnl(entries, root, tree, model, tips, wgt,
                                return_state=True)
    p, q, _ = root
    p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
    wgt = np.ascontiguousarray(wgt, dtype=np.int32)
    if p_tip and q_tip:
        st = sum_fn(TIP_TIP, None, None, tipVector,
                    np.ascontiguousarray(tips[p]),
                    np.ascontiguousarray(tips[q]), width)
    elif q_tip:
        st = sum_fn(TIP_INNER, None, clv[tree.clv_slot(p)],
                    tipVector, np.ascontiguousarray(tips[q]), None, width)
    elif p_tip:
        st = sum_fn(TIP_INNER, None, clv[tree.clv_slot(q)],
                    tipVector, np.ascontiguousarray(tips[p]), None, width)
    else:
        st = sum_fn(INNER_INNER, clv[tree.clv_slot(p)],
                    clv[tree.clv_slot(q)], tipVector, None, None, width)

    z = float(z0)
    zprev, zstep = z, (1.0 - O.ZMAX) * z + O.ZMIN
    curvat_ok, outer_converged, it = True, False, maxiter
    while not outer_converged:
        if curvat_ok:
            curvat_ok = False
            zprev = z
            zstep = (1.0 - O.ZMAX) * z + O.ZMIN
        z = min(max(z, O.ZMIN), O.ZMAX)
        lz = math.log(z)
        dlnL, d2lnL = core_fn(width, st, EIGN, g, lz, wgt)
        if (d2lnL >= 0.0) and (z < O.ZMAX):
            zprev = z = 0.37 * z + 0.63
            continue
        curvat_ok = True
        if d2lnL < 0.0:
            tantmp = -dlnL / d2lnL
            if tantmp < 100:
                z *= math.exp(tantmp)
                z = max(z, O.ZMIN)
                z = min(z, 0.25 * zprev + 0.75)
            else:
                z = 0.25 * zprev + 0.75
        z = min(z, O.ZMAX)
        it -= 1
        if abs(z - zprev) > zstep:
            if it < -20:
                z = float(z0)
                outer_converged = True
        else:
            outer_converged = True
    return z


def oracle_cat_full_lnl(entries, root, tree, model, tips, wgt, cptr,
                        per_site_rates, return_state=False):
    """CAT replay of a product traversal on the oracle kernels (the
    checker for DnaCatEngine)."""
    EIGN, EV, EI, tipVector, _ = _model_arrays(model)
    rptr = O.aligned(len(per_site_rates))
    rptr[:] = per_site_rates
    num_cats = len(per_site_rates)
    width = tips.shape[1]
    ntips = tips.shape[0] - 1
    cptr = np.ascontiguousarray(cptr, dtype=np.int32)
    wgt = np.ascontiguousarray(wgt, dtype=np.int32)
    clv = {}
    scalers = np.zeros(2 * ntips, dtype=np.int64)
    for e in entries:
        qz = math.log(e.qz) if e.qz > O.ZMIN else math.log(O.ZMIN)
        rz = math.log(e.rz) if e.rz > O.ZMIN else math.log(O.ZMIN)
        left, right = O.make_p(qz, rz, rptr, EI, EIGN, num_cats, 4)
        if e.tipCase == TIP_TIP:
            x3, inc = O.newview_dna_cat(
                TIP_TIP, EV, cptr, None, None, tipVector,
                np.ascontiguousarray(tips[e.x1Slot]),
                np.ascontiguousarray(tips[e.x2Slot]), width, left, right,
                wgt)
        elif e.tipCase == TIP_INNER:
            x3, inc = O.newview_dna_cat(
                TIP_INNER, EV, cptr, None, clv[e.x2Slot], tipVector,
                np.ascontiguousarray(tips[e.x1Slot]), None, width, left,
                right, wgt)
        else:
            x3, inc = O.newview_dna_cat(
                INNER_INNER, EV, cptr, clv[e.x1Slot], clv[e.x2Slot],
                tipVector, None, None, width, left, right, wgt)
        clv[e.x3Slot] = x3
        scalers[e.pNumber] = scalers[e.qNumber] + scalers[e.rNumber] + inc
    p, q, z = root
    diag = O.calc_diagptable(z, 4, num_cats, rptr, EIGN)
    p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
    if q_tip:
        lnl = O.evaluate_dna_cat(cptr, wgt, None, clv[tree.clv_slot(p)],
                                 tipVector, np.ascontiguousarray(tips[q]),
                                 width, diag)
    elif p_tip:
        lnl = O.evaluate_dna_cat(cptr, wgt, None, clv[tree.clv_slot(q)],
                                 tipVector, np.ascontiguousarray(tips[p]),
                                 width, diag)
    else:
        lnl = O.evaluate_dna_cat(cptr, wgt, clv[tree.clv_slot(p)],
                                 clv[tree.clv_slot(q)], tipVector, None,
                                 width, diag)
    lnl += float(scalers[p] + scalers[q]) * math.log(O.MINLIKELIHOOD)
    if return_state:
        return lnl, clv, scalers
    return lnl


class OracleLg4Engine(OracleEngine):
    """CPU LG4 (LG4M/LG4X) engine — per-category matrices through the
    oracle LG4 kernels (the checker for the GPU LG4 path)."""

    def __init__(self, tips, wgt, model):
        super().__init__(tips, wgt, model)

    def newview_traversal(self, entries):
        m = self.model
        for e in entries:
            qz = math.log(e.qz) if e.qz > O.ZMIN else math.log(O.ZMIN)
            rz = math.log(e.rz) if e.rz > O.ZMIN else math.log(O.ZMIN)
            left, right = O.make_p_lg4(qz, rz, m.gammaRates, m.EI4, m.EIGN4)
            if e.tipCase == TIP_TIP:
                x3, inc = O.newview_prot_lg4(
                    TIP_TIP, None, None, m.EV4, m.tipVector4,
                    np.ascontiguousarray(self.tips[e.x1Slot]),
                    np.ascontiguousarray(self.tips[e.x2Slot]), self.width,
                    left, right, self.wgt)
            elif e.tipCase == TIP_INNER:
                x3, inc = O.newview_prot_lg4(
                    TIP_INNER, None, self.clv[e.x2Slot], m.EV4, m.tipVector4,
                    np.ascontiguousarray(self.tips[e.x1Slot]), None,
                    self.width, left, right, self.wgt)
            else:
                x3, inc = O.newview_prot_lg4(
                    INNER_INNER, self.clv[e.x1Slot], self.clv[e.x2Slot],
                    m.EV4, m.tipVector4, None, None, self.width, left,
                    right, self.wgt)
            self.clv[e.x3Slot] = x3
            self.scalers[e.pNumber] = (self.scalers[e.qNumber] +
                                       self.scalers[e.rNumber] + inc)

    def evaluate_root(self, tree, p, q, z):
        m = self.model
        diag = O.calc_diagptable_lg4(z, m.gammaRates, m.EIGN4)
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if q_tip:
            lnl = O.evaluate_prot_lg4(
                self.wgt, None, self.clv[tree.clv_slot(p)], m.tipVector4,
                np.ascontiguousarray(self.tips[q]), self.width, diag,
                m.weights)
        elif p_tip:
            lnl = O.evaluate_prot_lg4(
                self.wgt, None, self.clv[tree.clv_slot(q)], m.tipVector4,
                np.ascontiguousarray(self.tips[p]), self.width, diag,
                m.weights)
        else:
            lnl = O.evaluate_prot_lg4(
                self.wgt, self.clv[tree.clv_slot(p)],
                self.clv[tree.clv_slot(q)], m.tipVector4, None, self.width,
                diag, m.weights)
        lnl += float(self.scalers[p] + self.scalers[q]) * \
            math.log(O.MINLIKELIHOOD)
        return lnl

    def sum_root(self, tree, p, q):
        m = self.model
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if p_tip and q_tip:
            self.sumtable = O.sum_prot_lg4(
                TIP_TIP, None, None, m.tipVector4,
                np.ascontiguousarray(self.tips[p]),
                np.ascontiguousarray(self.tips[q]), self.width)
        elif q_tip:
            self.sumtable = O.sum_prot_lg4(
                TIP_INNER, None, self.clv[tree.clv_slot(p)], m.tipVector4,
                np.ascontiguousarray(self.tips[q]), None, self.width)
        elif p_tip:
            self.sumtable = O.sum_prot_lg4(
                TIP_INNER, None, self.clv[tree.clv_slot(q)], m.tipVector4,
                np.ascontiguousarray(self.tips[p]), None, self.width)
        else:
            self.sumtable = O.sum_prot_lg4(
                INNER_INNER, self.clv[tree.clv_slot(p)],
                self.clv[tree.clv_slot(q)], m.tipVector4, None, None,
                self.width)

    def core_derivs(self, lz):
        m = self.model
        return O.core_prot_lg4(self.width, self.sumtable, m.EIGN4,
                               m.gammaRates, m.weights, lz, self.wgt)


class OracleSaveEngine(OracleEngine):
    """CPU -S (saveMemory) DNA engine over the oracle SAVE kernels: gap
    vectors, gap columns and compacted CLV slabs (the checker for
    SaveDnaEngine)."""

    def __init__(self, tips, wgt, model):
        super().__init__(tips, wgt, model)
        assert model.states == 4
        self.gvl = self.width // 32 + 1
        nn = 2 * self.ntips
        self.gap = np.zeros((nn, self.gvl), dtype=np.uint32)
        for t in range(1, self.ntips + 1):
            idx = np.nonzero(tips[t] == 15)[0]
            np.bitwise_or.at(self.gap[t], idx // 32,
                             (np.uint32(1) << (idx % 32).astype(np.uint32)))
        self.gapcol = {}

    def _gapcol_tip(self):
        _, _, _, tipVector, _ = self._arrays()
        return np.ascontiguousarray(tipVector[15 * 4:16 * 4])

    def newview_traversal(self, entries):
        EIGN, EV, EI, tipVector, g = self._arrays()
        import ctypes as C
        dp = lambda a: (a.ctypes.data_as(C.POINTER(C.c_double))
                        if a is not None else C.cast(None,
                                                     C.POINTER(C.c_double)))
        up = lambda a: (a.ctypes.data_as(C.POINTER(C.c_uint))
                        if a is not None else C.cast(None,
                                                     C.POINTER(C.c_uint)))
        u8 = lambda a: (a.ctypes.data_as(C.POINTER(C.c_ubyte))
                        if a is not None else C.cast(None,
                                                     C.POINTER(C.c_ubyte)))
        for e in entries:
            qz = math.log(e.qz) if e.qz > O.ZMIN else math.log(O.ZMIN)
            rz = math.log(e.rz) if e.rz > O.ZMIN else math.log(O.ZMIN)
            left, right = O.make_p(qz, rz, g, EI, EIGN, 4, 4)
            p, q, r = e.pNumber, e.qNumber, e.rNumber
            self.gap[p] = self.gap[q] & self.gap[r]
            nz = self.width - int(
                np.unpackbits(self.gap[p].view(np.uint8),
                              bitorder="little")[:self.width].sum())
            x3 = O.aligned(max(nz, 1) * 16)
            gcol3 = O.aligned(16)
            q_tip = e.tipCase != INNER_INNER
            r_tip = e.tipCase == TIP_TIP
            gc1 = self._gapcol_tip() if q_tip else self.gapcol[e.x1Slot]
            gc2 = self._gapcol_tip() if r_tip else self.gapcol[e.x2Slot]
            inc = C.c_int(0)
            O._orc.oracle_newview_dna_gamma_save(
                C.c_int(e.tipCase),
                dp(None if q_tip else self.clv[e.x1Slot]),
                dp(None if r_tip else self.clv[e.x2Slot]),
                dp(x3), dp(EV), dp(tipVector),
                u8(np.ascontiguousarray(self.tips[e.x1Slot])
                   if q_tip else None),
                u8(np.ascontiguousarray(self.tips[e.x2Slot])
                   if r_tip else None),
                C.c_int(self.width), dp(left), dp(right),
                self.wgt.ctypes.data_as(C.POINTER(C.c_int)),
                C.byref(inc), up(self.gap[q]), up(self.gap[r]),
                up(self.gap[p]), dp(gc1), dp(gc2), dp(gcol3))
            self.clv[e.x3Slot] = x3
            self.gapcol[e.x3Slot] = gcol3
            self.scalers[p] = self.scalers[q] + self.scalers[r] + inc.value

    def evaluate_root(self, tree, p, q, z):
        EIGN, EV, EI, tipVector, g = self._arrays()
        import ctypes as C
        diag = O.calc_diagptable(z, 4, 4, g, EIGN)
        dp = lambda a: (a.ctypes.data_as(C.POINTER(C.c_double))
                        if a is not None else C.cast(None,
                                                     C.POINTER(C.c_double)))
        up = lambda a: (a.ctypes.data_as(C.POINTER(C.c_uint))
                        if a is not None else C.cast(None,
                                                     C.POINTER(C.c_uint)))
        O._orc.oracle_evaluate_dna_gamma_save.restype = C.c_double
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if q_tip or p_tip:
            tip, inner = (q, p) if q_tip else (p, q)
            sl = tree.clv_slot(inner)
            lnl = O._orc.oracle_evaluate_dna_gamma_save(
                self.wgt.ctypes.data_as(C.POINTER(C.c_int)), dp(None),
                dp(self.clv[sl]), dp(tipVector),
                np.ascontiguousarray(self.tips[tip]).ctypes.data_as(
                    C.POINTER(C.c_ubyte)),
                C.c_int(self.width), dp(diag), dp(None),
                dp(self.gapcol[sl]), up(None), up(self.gap[inner]))
        else:
            s1, s2 = tree.clv_slot(p), tree.clv_slot(q)
            lnl = O._orc.oracle_evaluate_dna_gamma_save(
                self.wgt.ctypes.data_as(C.POINTER(C.c_int)),
                dp(self.clv[s1]), dp(self.clv[s2]), dp(tipVector),
                C.cast(None, C.POINTER(C.c_ubyte)), C.c_int(self.width),
                dp(diag), dp(self.gapcol[s1]), dp(self.gapcol[s2]),
                up(self.gap[p]), up(self.gap[q]))
        lnl += float(self.scalers[p] + self.scalers[q]) * \
            math.log(O.MINLIKELIHOOD)
        return lnl

    def sum_root(self, tree, p, q):
        EIGN, EV, EI, tipVector, g = self._arrays()
        import ctypes as C
        dp = lambda a: (a.ctypes.data_as(C.POINTER(C.c_double))
                        if a is not None else C.cast(None,
                                                     C.POINTER(C.c_double)))
        up = lambda a: (a.ctypes.data_as(C.POINTER(C.c_uint))
                        if a is not None else C.cast(None,
                                                     C.POINTER(C.c_uint)))
        u8 = lambda a: (a.ctypes.data_as(C.POINTER(C.c_ubyte))
                        if a is not None else C.cast(None,
                                                     C.POINTER(C.c_ubyte)))
        self.sumtable = O.aligned(self.width * 16)
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if p_tip and q_tip:
            O._orc.oracle_sum_dna_gamma_save(
                0, dp(self.sumtable), dp(None), dp(None), dp(tipVector),
                u8(np.ascontiguousarray(self.tips[p])),
                u8(np.ascontiguousarray(self.tips[q])),
                C.c_int(self.width), dp(None), dp(None), up(None), up(None))
        elif q_tip or p_tip:
            tip, inner = (q, p) if q_tip else (p, q)
            sl = tree.clv_slot(inner)
            O._orc.oracle_sum_dna_gamma_save(
                1, dp(self.sumtable), dp(None), dp(self.clv[sl]),
                dp(tipVector), u8(np.ascontiguousarray(self.tips[tip])),
                u8(None), C.c_int(self.width), dp(None),
                dp(self.gapcol[sl]), up(None), up(self.gap[inner]))
        else:
            s1, s2 = tree.clv_slot(p), tree.clv_slot(q)
            O._orc.oracle_sum_dna_gamma_save(
                2, dp(self.sumtable), dp(self.clv[s1]), dp(self.clv[s2]),
                dp(tipVector), u8(None), u8(None), C.c_int(self.width),
                dp(self.gapcol[s1]), dp(self.gapcol[s2]), up(self.gap[p]),
                up(self.gap[q]))

    def clv_bytes(self):
        return sum(v.nbytes for v in self.clv.values())
