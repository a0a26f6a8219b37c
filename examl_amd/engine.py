"""Device engine for one DNA GTRGAMMA partition slice: owns the HBM-resident
state (CLVs, tips, weights, model vectors, scaler counts) and drives the HIP
kernels through the C-ABI (include/examl_hip.h).

This is the GPU replacement for the per-partition bodies of
newviewIterative / evaluateIterative / makenewzIterative+execCore
(SURVEY.md §8a rows a3-a5).  All buffers are torch CUDA tensors (device
memory + stream plumbing only); kernels run on torch's current stream so
torch.distributed (RCCL) collectives compose in stream order.
"""

import ctypes
import math

import numpy as np
import torch

from . import INNER_INNER, TIP_INNER, TIP_TIP, TravEntry, ZMIN, ZMAX, check, lib


def _vp(t):
    return ctypes.c_void_p(t.data_ptr())


def _np_vp(a):
    return a.ctypes.data_as(ctypes.c_void_p)


class DnaGammaEngine:
    """Engine for one GTRGAMMA partition slice; `model.states` selects the
    DNA (4-state) or protein (20-state) kernel family."""

    def __init__(self, tips, wgt, model, device="cuda", max_ops=None):
        """tips: uint8 [ntips+1, width] (row 0 unused; ambiguity codes —
        1..15 DNA / 1..22 AA — the yVector of examl/axml.h:599);
        wgt: int32 [width]; model: DnaGtrModel or ProtGtrModel."""
        assert tips.dtype == np.uint8 and tips.ndim == 2
        self.states = model.states
        self.SPAN = 4 * self.states
        self._sfx = "dna" if self.states == 4 else "prot"
        self.ntips = tips.shape[0] - 1
        self.width = tips.shape[1]
        self.ninner = self.ntips - 2
        self.model = model
        self.device = torch.device(device)
        if self.device.type == "cuda" and not torch.cuda.is_available():
            raise RuntimeError("examl_amd: CUDA/HIP device not available")
        n_ops = max_ops or (self.ninner + 8)

        # host copies kept for evaluatePartialGeneric (CAT per-site rate
        # search, evaluatePartialGenericSpecial.c:259) and updatePerSiteRates
        self.host_tips = np.ascontiguousarray(tips)
        self.host_wgt = np.ascontiguousarray(wgt, dtype=np.int32)

        dev = self.device
        self.d_tips = torch.from_numpy(np.ascontiguousarray(tips)).to(dev)
        self.d_wgt = torch.from_numpy(
            np.ascontiguousarray(wgt, dtype=np.int32)).to(dev)
        self.d_EV = torch.from_numpy(model.EV).to(dev)
        self.d_tipVector = torch.from_numpy(model.tipVector).to(dev)
        # CLV pool: one slot per inner node (xVector, device-resident for the
        # life of the run — the GPU answer to the lazy allocation of
        # newviewGenericSpecial.c:1200-1215)
        self.d_clv = torch.empty((self.ninner, self.width * self.SPAN),
                                 dtype=torch.float64, device=dev)
        # per-node scaler counts (globalScaler, axml.h:601); tips stay 0
        self.d_scalers = torch.zeros(2 * self.ntips, dtype=torch.int32,
                                     device=dev)
        self.d_pbuf = torch.empty(n_ops * 8 * self.states * self.states,
                                  dtype=torch.float64, device=dev)
        self.d_inc = torch.empty(n_ops, dtype=torch.int32, device=dev)
        self.d_diag = torch.empty(4 * self.states, dtype=torch.float64,
                                  device=dev)
        self.d_dtab = torch.empty(12 * self.states, dtype=torch.float64,
                                  device=dev)
        self.d_partials = torch.empty(2 * 8192, dtype=torch.float64,
                                      device=dev)
        self.d_lnl = torch.zeros(1, dtype=torch.float64, device=dev)
        self.d_out2 = torch.zeros(2, dtype=torch.float64, device=dev)
        self.d_sum = None  # sumBuffer (axml.h:558), allocated on first use
        self._max_ops = n_ops

    # -- plumbing -----------------------------------------------------------

    def _fn(self, name):
        return getattr(lib(), f"examl_hip_{name}_{self._sfx}_gamma")

    def _stream(self):
        if self.device.type == "cuda":
            return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
        return ctypes.c_void_p(0)

    def sync(self):
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)

    def upload_model(self):
        """Re-upload EV/tipVector after a model-parameter change
        (changeModelParameters -> initReversibleGTR,
        optimizeModel.c:419-449).  EIGN/EI/gammaRates are host-side inputs
        of each call and need no upload."""
        self.d_EV.copy_(torch.from_numpy(self.model.EV))
        self.d_tipVector.copy_(torch.from_numpy(self.model.tipVector))

    # -- newviewIterative ---------------------------------------------------

    def newview_traversal(self, entries):
        """Run the post-order CLV updates (newviewIterative,
        newviewGenericSpecial.c:917)."""
        if not entries:
            return
        assert len(entries) <= self._max_ops, "grow max_ops"
        arr = (TravEntry * len(entries))(*entries)
        m = self.model
        check(self._fn("newview_traversal")(
            ctypes.cast(arr, ctypes.c_void_p), len(entries),
            _np_vp(m.EIGN), _np_vp(m.EI), _np_vp(m.gammaRates),
            _vp(self.d_EV), _vp(self.d_tipVector), _vp(self.d_clv),
            ctypes.c_long(self.width * self.SPAN), _vp(self.d_tips),
            ctypes.c_long(self.width), _vp(self.d_wgt),
            ctypes.c_long(self.width), _vp(self.d_scalers), _vp(self.d_inc),
            _vp(self.d_pbuf), self._stream()), "newview_traversal")

    # -- evaluateIterative --------------------------------------------------

    def _root_case(self, tree, p, q):
        """Resolve the root branch operands like evaluateIterative
        (evaluateGenericSpecial.c:612-668): tip goes to the tip slot, the
        other side's CLV to x2."""
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        assert not (p_tip and q_tip)
        if q_tip:
            return TIP_INNER, -1, tree.clv_slot(p), q, -1, p, q
        if p_tip:
            return TIP_INNER, -1, tree.clv_slot(q), p, -1, p, q
        return (INNER_INNER, tree.clv_slot(p), tree.clv_slot(q), -1, -1, p, q)

    def evaluate_root(self, tree, p, q, z, all_reduce=False):
        """lnL at the branch p--q (evaluateIterative's per-partition body +
        the C1 all-reduce, evaluateGenericSpecial.c:403/969).  Returns the
        device scalar tensor (call .item() to sync)."""
        tc, x1s, x2s, tslot, _, pn, qn = self._root_case(tree, p, q)
        self.d_lnl.zero_()
        m = self.model
        check(self._fn("evaluate_root")(
            tc, pn, qn, x1s, x2s, tslot, ctypes.c_double(z),
            _np_vp(m.EIGN), _np_vp(m.gammaRates), _vp(self.d_tipVector),
            _vp(self.d_clv), ctypes.c_long(self.width * self.SPAN),
            _vp(self.d_tips), ctypes.c_long(self.width), _vp(self.d_wgt),
            ctypes.c_long(self.width), _vp(self.d_scalers), _vp(self.d_diag),
            _vp(self.d_partials), _vp(self.d_lnl), self._stream()),
            "evaluate_root")
        if all_reduce and torch.distributed.is_initialized():
            torch.distributed.all_reduce(self.d_lnl)
        return self.d_lnl

    def full_lnl(self, tree, root_edge=None, all_reduce=False):
        """Full-traversal evaluateGeneric (evaluateGenericSpecial.c:897)."""
        entries, (p, q, z) = tree.full_traversal(root_edge)
        self.newview_traversal(entries)
        return self.evaluate_root(tree, p, q, z, all_reduce=all_reduce)

    # -- makenewz (NR branch-length optimization) ---------------------------

    def _ensure_sum(self):
        if self.d_sum is None:
            self.d_sum = torch.empty(self.width * self.SPAN,
                                     dtype=torch.float64, device=self.device)

    def sum_root(self, tree, p, q):
        """sumBuffer precompute at branch p--q (makenewzIterative,
        makenewzGenericSpecial.c:628; CLVs must already face the branch)."""
        self._ensure_sum()
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if p_tip and q_tip:
            tc, x1s, x2s, t1, t2 = TIP_TIP, -1, -1, p, q
        elif q_tip:
            tc, x1s, x2s, t1, t2 = TIP_INNER, -1, tree.clv_slot(p), q, -1
        elif p_tip:
            tc, x1s, x2s, t1, t2 = TIP_INNER, -1, tree.clv_slot(q), p, -1
        else:
            tc, x1s, x2s, t1, t2 = (INNER_INNER, tree.clv_slot(p),
                                    tree.clv_slot(q), -1, -1)
        check(self._fn("sum_root")(
            tc, x1s, x2s, t1, t2, _vp(self.d_tipVector), _vp(self.d_clv),
            ctypes.c_long(self.width * self.SPAN), _vp(self.d_tips),
            ctypes.c_long(self.width), _vp(self.d_sum),
            ctypes.c_long(self.width), self._stream()), "sum_root")

    def core_derivs(self, lz, all_reduce=False):
        """execCore (makenewzGenericSpecial.c:849) + the C2 all-reduce
        (:1244).  Returns (dlnLdlz, d2lnLdlz2) as host floats."""
        m = self.model
        self.d_out2.zero_()
        check(self._fn("core_root")(
            ctypes.c_long(self.width), _vp(self.d_sum), _np_vp(m.EIGN),
            _np_vp(m.gammaRates), ctypes.c_double(lz), _vp(self.d_wgt),
            _vp(self.d_dtab), _vp(self.d_partials), _vp(self.d_out2),
            self._stream()), "core_root")
        if all_reduce and torch.distributed.is_initialized():
            torch.distributed.all_reduce(self.d_out2)
        v = self.d_out2.cpu()
        return float(v[0]), float(v[1])

    def core_derivs_async(self, lz):
        """Launch execCore without reading back (the search layer batches
        one sync over all partitions)."""
        m = self.model
        self.d_out2.zero_()
        check(self._fn("core_root")(
            ctypes.c_long(self.width), _vp(self.d_sum), _np_vp(m.EIGN),
            _np_vp(m.gammaRates), ctypes.c_double(lz), _vp(self.d_wgt),
            _vp(self.d_dtab), _vp(self.d_partials), _vp(self.d_out2),
            self._stream()), "core_root")
        return self.d_out2

    def makenewz(self, tree, p, q, z0, maxiter=64, all_reduce=False):
        """Newton-Raphson branch length at p--q, restating topLevelMakenewz
        (makenewzGenericSpecial.c:1133) for the joint-branch-length case
        (numBranches=1).  CLVs must already face the branch (the caller runs
        a traversal to p--q first, as makenewzGeneric:1385 does)."""
        self.sum_root(tree, p, q)
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
            dlnL, d2lnL = self.core_derivs(lz, all_reduce=all_reduce)
            if (d2lnL >= 0.0) and (z < ZMAX):
                zprev = z = 0.37 * z + 0.63  # bad curvature, shorten branch
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


class DnaCatEngine(DnaGammaEngine):
    """CAT (PSR, -m PSR) variant: span 4, per-site rate category cptr and
    perSiteRates (pInfo.rateCategory / perSiteRates, axml.h:591-592) in
    place of the 4 discrete gamma rates."""

    def __init__(self, tips, wgt, model, cptr, per_site_rates, device="cuda",
                 max_ops=None):
        self.cptr = np.ascontiguousarray(cptr, dtype=np.int32)
        self.per_site_rates = np.ascontiguousarray(per_site_rates,
                                                   dtype=np.float64)
        super().__init__(tips, wgt, model, device=device, max_ops=max_ops)
        self.SPAN = 4
        self.num_cats = len(self.per_site_rates)
        dev = self.device
        self.d_cptr = torch.from_numpy(self.cptr).to(dev)
        # re-size CAT-specific buffers (base class sized them for GAMMA);
        # scratch is sized for maxCategories=25 up front so
        # optimizeRateCategories can grow numberOfCategories without
        # reallocating (reference allocates for tr->maxCategories,
        # axml.c:1936)
        MAXC = 25
        self.d_clv = torch.empty((self.ninner, self.width * 4),
                                 dtype=torch.float64, device=dev)
        self.d_pbuf = torch.empty(self._max_ops * MAXC * 32,
                                  dtype=torch.float64, device=dev)
        self.d_diag = torch.empty(MAXC * 4, dtype=torch.float64,
                                  device=dev)
        self.d_dtab = torch.empty(MAXC * 4 + 8 + MAXC, dtype=torch.float64,
                                  device=dev)

    def set_site_rates(self, cptr, per_site_rates):
        """Install a new rate categorization (the device-side half of
        optimizeRateCategories' writeback, optimizeModel.c:2465-2470)."""
        assert len(per_site_rates) <= 25
        self.cptr = np.ascontiguousarray(cptr, dtype=np.int32)
        self.per_site_rates = np.ascontiguousarray(per_site_rates,
                                                   dtype=np.float64)
        self.num_cats = len(self.per_site_rates)
        self.d_cptr.copy_(torch.from_numpy(self.cptr))

    def newview_traversal(self, entries):
        if not entries:
            return
        assert len(entries) <= self._max_ops, "grow max_ops"
        arr = (TravEntry * len(entries))(*entries)
        m = self.model
        check(lib().examl_hip_newview_traversal_dna_cat(
            ctypes.cast(arr, ctypes.c_void_p), len(entries),
            _np_vp(m.EIGN), _np_vp(m.EI), _np_vp(self.per_site_rates),
            self.num_cats, _vp(self.d_EV), _vp(self.d_tipVector),
            _vp(self.d_cptr), _vp(self.d_clv),
            ctypes.c_long(self.width * 4), _vp(self.d_tips),
            ctypes.c_long(self.width), _vp(self.d_wgt),
            ctypes.c_long(self.width), _vp(self.d_scalers), _vp(self.d_inc),
            _vp(self.d_pbuf), self._stream()), "newview_traversal_cat")

    def evaluate_root(self, tree, p, q, z, all_reduce=False):
        tc, x1s, x2s, tslot, _, pn, qn = self._root_case(tree, p, q)
        self.d_lnl.zero_()
        m = self.model
        check(lib().examl_hip_evaluate_root_dna_cat_x(
            tc, pn, qn, x1s, x2s, tslot, ctypes.c_double(z),
            _np_vp(m.EIGN), _np_vp(self.per_site_rates), self.num_cats,
            _vp(self.d_tipVector), _vp(self.d_cptr), _vp(self.d_clv),
            ctypes.c_long(self.width * 4), _vp(self.d_tips),
            ctypes.c_long(self.width), _vp(self.d_wgt),
            ctypes.c_long(self.width), _vp(self.d_scalers), _vp(self.d_diag),
            _vp(self.d_partials), _vp(self.d_lnl), self._stream()),
            "evaluate_root_cat")
        if all_reduce and torch.distributed.is_initialized():
            torch.distributed.all_reduce(self.d_lnl)
        return self.d_lnl

    def sum_root(self, tree, p, q):
        if self.d_sum is None:
            self.d_sum = torch.empty(self.width * 4, dtype=torch.float64,
                                     device=self.device)
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if p_tip and q_tip:
            tc, x1s, x2s, t1, t2 = TIP_TIP, -1, -1, p, q
        elif q_tip:
            tc, x1s, x2s, t1, t2 = TIP_INNER, -1, tree.clv_slot(p), q, -1
        elif p_tip:
            tc, x1s, x2s, t1, t2 = TIP_INNER, -1, tree.clv_slot(q), p, -1
        else:
            tc, x1s, x2s, t1, t2 = (INNER_INNER, tree.clv_slot(p),
                                    tree.clv_slot(q), -1, -1)
        check(lib().examl_hip_sum_root_dna_cat(
            tc, x1s, x2s, t1, t2, _vp(self.d_tipVector), _vp(self.d_clv),
            ctypes.c_long(self.width * 4), _vp(self.d_tips),
            ctypes.c_long(self.width), _vp(self.d_sum),
            ctypes.c_long(self.width), self._stream()), "sum_root_cat")

    def core_derivs_async(self, lz):
        m = self.model
        self.d_out2.zero_()
        check(lib().examl_hip_core_root_dna_cat(
            ctypes.c_long(self.width), _vp(self.d_sum), _np_vp(m.EIGN),
            _np_vp(self.per_site_rates), self.num_cats, ctypes.c_double(lz),
            _vp(self.d_wgt), _vp(self.d_cptr), _vp(self.d_dtab),
            _vp(self.d_partials), _vp(self.d_out2), self._stream()),
            "core_root_cat")
        return self.d_out2

    def core_derivs(self, lz, all_reduce=False):
        self.core_derivs_async(lz)
        if all_reduce and torch.distributed.is_initialized():
            torch.distributed.all_reduce(self.d_out2)
        v = self.d_out2.cpu()
        return float(v[0]), float(v[1])


class Lg4Engine(DnaGammaEngine):
    """LG4M/LG4X variant: per-gamma-category matrices (EV4/tipVector4
    device buffers, LG4 kernels; the LG4 branches of newviewIterative /
    evaluateIterative / makenewzIterative)."""

    def __init__(self, tips, wgt, model, device="cuda", max_ops=None):
        super().__init__(tips, wgt, model, device=device, max_ops=max_ops)
        dev = self.device
        # per-category model buffers replace the base d_EV/d_tipVector
        self.d_EV4 = torch.from_numpy(model.EV4).to(dev)
        self.d_tipVector4 = torch.from_numpy(model.tipVector4).to(dev)
        self.d_diag = torch.empty(84, dtype=torch.float64, device=dev)
        self.d_dtab = torch.empty(244, dtype=torch.float64, device=dev)

    def upload_model(self):
        m = self.model
        self.d_EV4.copy_(torch.from_numpy(m.EV4))
        self.d_tipVector4.copy_(torch.from_numpy(m.tipVector4))

    def newview_traversal(self, entries):
        if not entries:
            return
        assert len(entries) <= self._max_ops, "grow max_ops"
        arr = (TravEntry * len(entries))(*entries)
        m = self.model
        check(lib().examl_hip_newview_traversal_prot_lg4(
            ctypes.cast(arr, ctypes.c_void_p), len(entries),
            _np_vp(m.EIGN4), _np_vp(m.EI4), _np_vp(m.gammaRates),
            _vp(self.d_EV4), _vp(self.d_tipVector4), _vp(self.d_clv),
            ctypes.c_long(self.width * 80), _vp(self.d_tips),
            ctypes.c_long(self.width), _vp(self.d_wgt),
            ctypes.c_long(self.width), _vp(self.d_scalers), _vp(self.d_inc),
            _vp(self.d_pbuf), self._stream()), "newview_traversal_lg4")

    def evaluate_root(self, tree, p, q, z, all_reduce=False):
        tc, x1s, x2s, tslot, _, pn, qn = self._root_case(tree, p, q)
        self.d_lnl.zero_()
        m = self.model
        check(lib().examl_hip_evaluate_root_prot_lg4(
            tc, pn, qn, x1s, x2s, tslot, ctypes.c_double(z),
            _np_vp(m.EIGN4), _np_vp(m.gammaRates), _np_vp(m.weights),
            _vp(self.d_tipVector4), _vp(self.d_clv),
            ctypes.c_long(self.width * 80), _vp(self.d_tips),
            ctypes.c_long(self.width), _vp(self.d_wgt),
            ctypes.c_long(self.width), _vp(self.d_scalers), _vp(self.d_diag),
            _vp(self.d_partials), _vp(self.d_lnl), self._stream()),
            "evaluate_root_lg4")
        if all_reduce and torch.distributed.is_initialized():
            torch.distributed.all_reduce(self.d_lnl)
        return self.d_lnl

    def sum_root(self, tree, p, q):
        if self.d_sum is None:
            self.d_sum = torch.empty(self.width * 80, dtype=torch.float64,
                                     device=self.device)
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if p_tip and q_tip:
            tc, x1s, x2s, t1, t2 = TIP_TIP, -1, -1, p, q
        elif q_tip:
            tc, x1s, x2s, t1, t2 = TIP_INNER, -1, tree.clv_slot(p), q, -1
        elif p_tip:
            tc, x1s, x2s, t1, t2 = TIP_INNER, -1, tree.clv_slot(q), p, -1
        else:
            tc, x1s, x2s, t1, t2 = (INNER_INNER, tree.clv_slot(p),
                                    tree.clv_slot(q), -1, -1)
        check(lib().examl_hip_sum_root_prot_lg4(
            tc, x1s, x2s, t1, t2, _vp(self.d_tipVector4), _vp(self.d_clv),
            ctypes.c_long(self.width * 80), _vp(self.d_tips),
            ctypes.c_long(self.width), _vp(self.d_sum),
            ctypes.c_long(self.width), self._stream()), "sum_root_lg4")

    def core_derivs_async(self, lz):
        m = self.model
        self.d_out2.zero_()
        check(lib().examl_hip_core_root_prot_lg4(
            ctypes.c_long(self.width), _vp(self.d_sum), _np_vp(m.EIGN4),
            _np_vp(m.gammaRates), _np_vp(m.weights), ctypes.c_double(lz),
            _vp(self.d_wgt), _vp(self.d_dtab), _vp(self.d_partials),
            _vp(self.d_out2), self._stream()), "core_root_lg4")
        return self.d_out2


class SaveDnaEngine(DnaGammaEngine):
    """-S (saveMemory, SEV) DNA GTRGAMMA engine: per-node gap bit vectors,
    gap-column CLVs and COMPACTED per-node CLV slabs (allocated at each
    node's non-gap length, like the reference's xSpaceVector realloc,
    newviewGenericSpecial.c:1200-1218).  The reference's sequential
    compaction pointers become per-node non-gap prefix arrays for O(1)
    per-thread indexing on the GPU."""

    def __init__(self, tips, wgt, model, device="cuda", max_ops=None):
        assert model.states == 4
        super().__init__(tips, wgt, model, device=device, max_ops=max_ops)
        dev = self.device
        self.d_clv = None  # base dense pool unused
        self.gvl = self.width // 32 + 1
        nn = 2 * self.ntips
        # gap vectors: tip rows = (code == 15) (axml.c:2168)
        host_gap = np.zeros((nn, self.gvl), dtype=np.uint32)
        for t in range(1, self.ntips + 1):
            idx = np.nonzero(tips[t] == 15)[0]
            np.bitwise_or.at(host_gap[t], idx // 32,
                             (np.uint32(1) << (idx % 32).astype(np.uint32)))
        self.d_gap = torch.from_numpy(host_gap.view(np.int32)).to(dev)
        # non-gap prefixes (exclusive, per 32-site word; [gvl] = total)
        host_pre = np.zeros((nn, self.gvl + 1), dtype=np.int32)
        for t in range(1, self.ntips + 1):
            cnt = np.zeros(self.gvl, dtype=np.int64)
            for w in range(self.gvl):
                lo, hi = w * 32, min((w + 1) * 32, self.width)
                bits = int(host_gap[t, w])
                cnt[w] = (hi - lo) - bin(bits & ((1 << (hi - lo)) - 1)
                                         ).count("1")
            host_pre[t, 1:] = np.cumsum(cnt)
        self.d_pre = torch.from_numpy(host_pre).to(dev)
        self.d_gapcol = torch.zeros(self.ninner * 16, dtype=torch.float64,
                                    device=dev)
        self.d_scalegap = torch.zeros(1, dtype=torch.int32, device=dev)
        self.clv_slots = {}  # slot -> compacted tensor (lazy realloc)
        self._tipvec_gapcol = None  # tipVector[15*4:] device view

    def _gap_row(self, node):
        return _vp(self.d_gap[node])

    def _pre_row(self, node):
        return _vp(self.d_pre[node])

    def _gapcol_of(self, node_or_slot, is_tip):
        if is_tip:
            # undetermined tipVector row (gapOffset, newviewGeneric:1229)
            return ctypes.c_void_p(self.d_tipVector.data_ptr() + 15 * 4 * 8)
        return ctypes.c_void_p(self.d_gapcol.data_ptr() +
                               node_or_slot * 16 * 8)

    def _slot_clv(self, slot, required):
        t = self.clv_slots.get(slot)
        if t is None or t.numel() != required:
            t = torch.empty(required, dtype=torch.float64,
                            device=self.device)
            self.clv_slots[slot] = t
        return t

    def newview_traversal(self, entries):
        if not entries:
            return
        m = self.model
        L = lib()
        s = self._stream()
        self.d_inc.zero_()  # the per-op scaler accumulator starts at 0
        for e in entries:
            qz = math.log(e.qz) if e.qz > ZMIN else math.log(ZMIN)
            rz = math.log(e.rz) if e.rz > ZMIN else math.log(ZMIN)
            hostP = np.empty(128)
            L.examl_host_make_p(ctypes.c_double(qz), ctypes.c_double(rz),
                                _np_vp(m.gammaRates), _np_vp(m.EI),
                                _np_vp(m.EIGN), 4, _np_vp(hostP),
                                ctypes.c_void_p(hostP.ctypes.data + 64 * 8),
                                4)
            d_P = torch.from_numpy(hostP).to(self.device)
            p, q, r = e.pNumber, e.qNumber, e.rNumber
            check(L.examl_hip_gap_and_prefix(
                self._gap_row(q), self._gap_row(r), self._gap_row(p),
                self._pre_row(p), self.gvl, ctypes.c_long(self.width), s),
                "gap_and_prefix")
            # compacted x3 slab: width - setBits sites
            total = int(self.d_pre[p, self.gvl].item())
            x3 = self._slot_clv(e.x3Slot, max(total, 1) * 16)
            q_tip = e.tipCase != INNER_INNER
            r_tip = e.tipCase == TIP_TIP
            x1 = None if q_tip else self.clv_slots[e.x1Slot]
            x2 = None if r_tip else self.clv_slots[e.x2Slot]
            t1 = (ctypes.c_void_p(self.d_tips.data_ptr() +
                                  e.x1Slot * self.width)
                  if q_tip else None)
            t2 = (ctypes.c_void_p(self.d_tips.data_ptr() +
                                  e.x2Slot * self.width)
                  if r_tip else None)
            check(L.examl_hip_newview_dna_save(
                e.tipCase,
                _vp(x1) if x1 is not None else None,
                _vp(x2) if x2 is not None else None,
                _vp(x3), _vp(d_P), _vp(self.d_EV), _vp(self.d_tipVector),
                t1, t2, _vp(self.d_wgt), ctypes.c_long(self.width),
                _vp(self.d_inc), self._gap_row(q), self._gap_row(r),
                self._gap_row(p), self._pre_row(q), self._pre_row(r),
                self._pre_row(p),
                self._gapcol_of(q if q_tip else self.tree_slot(e.x1Slot),
                                q_tip),
                self._gapcol_of(r if r_tip else self.tree_slot(e.x2Slot),
                                r_tip),
                self._gapcol_of(e.x3Slot, False), _vp(self.d_scalegap), s),
                "newview_dna_save")
            self._finalize_scaler_one(p, q, r)

    def tree_slot(self, slot):
        # gap-column indices are CLV-slot based for inner nodes
        return slot

    def _finalize_scaler_one(self, p, q, r):
        # gs[p] = gs[q] + gs[r] + inc (newviewGenericSpecial.c:1503);
        # single-op version using torch ops on the stream
        inc = self.d_inc[0]
        self.d_scalers[p] = self.d_scalers[q] + self.d_scalers[r] + inc
        self.d_inc.zero_()

    def evaluate_root(self, tree, p, q, z, all_reduce=False):
        tc, x1s, x2s, tslot, _, pn, qn = self._root_case(tree, p, q)
        self.d_lnl.zero_()
        m = self.model
        hostDiag = np.empty(16)
        lib().examl_host_calc_diagptable(ctypes.c_double(z), 4, 4,
                                         _np_vp(m.gammaRates),
                                         _np_vp(m.EIGN), _np_vp(hostDiag))
        d_diag = torch.from_numpy(hostDiag).to(self.device)
        s = self._stream()
        grid_partials = self.d_partials
        if tc == TIP_INNER:
            x2 = self.clv_slots[x2s]
            inner_node = p if tree.is_tip(q) else q
            tip_node = q if tree.is_tip(q) else p
            check(lib().examl_hip_evaluate_dna_save(
                TIP_INNER, None, _vp(x2), _vp(self.d_tipVector),
                ctypes.c_void_p(self.d_tips.data_ptr() +
                                tslot * self.width),
                _vp(self.d_wgt), _vp(d_diag), ctypes.c_long(self.width),
                None, self._gap_row(inner_node), None,
                self._pre_row(inner_node),
                None, self._gapcol_of(x2s, False), pn, qn,
                _vp(self.d_scalers), _vp(grid_partials), _vp(self.d_lnl),
                s), "evaluate_dna_save")
        else:
            x1 = self.clv_slots[x1s]
            x2 = self.clv_slots[x2s]
            check(lib().examl_hip_evaluate_dna_save(
                INNER_INNER, _vp(x1), _vp(x2), _vp(self.d_tipVector), None,
                _vp(self.d_wgt), _vp(d_diag), ctypes.c_long(self.width),
                self._gap_row(p), self._gap_row(q), self._pre_row(p),
                self._pre_row(q), self._gapcol_of(x1s, False),
                self._gapcol_of(x2s, False), pn, qn, _vp(self.d_scalers),
                _vp(grid_partials), _vp(self.d_lnl), s),
                "evaluate_dna_save")
        if all_reduce and torch.distributed.is_initialized():
            torch.distributed.all_reduce(self.d_lnl)
        return self.d_lnl

    def sum_root(self, tree, p, q):
        if self.d_sum is None:
            self.d_sum = torch.empty(self.width * 16, dtype=torch.float64,
                                     device=self.device)
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        s = self._stream()
        if p_tip and q_tip:
            check(lib().examl_hip_sum_dna_save(
                TIP_TIP, _vp(self.d_sum), None, None,
                _vp(self.d_tipVector),
                ctypes.c_void_p(self.d_tips.data_ptr() + p * self.width),
                ctypes.c_void_p(self.d_tips.data_ptr() + q * self.width),
                ctypes.c_long(self.width), None, None, None, None, None,
                None, s), "sum_dna_save")
            return
        if q_tip or p_tip:
            tip, inner = (q, p) if q_tip else (p, q)
            sl = tree.clv_slot(inner)
            check(lib().examl_hip_sum_dna_save(
                TIP_INNER, _vp(self.d_sum), None,
                _vp(self.clv_slots[sl]), _vp(self.d_tipVector),
                ctypes.c_void_p(self.d_tips.data_ptr() + tip * self.width),
                None, ctypes.c_long(self.width), None,
                self._gap_row(inner), None, self._pre_row(inner), None,
                self._gapcol_of(sl, False), s), "sum_dna_save")
            return
        s1, s2 = tree.clv_slot(p), tree.clv_slot(q)
        check(lib().examl_hip_sum_dna_save(
            INNER_INNER, _vp(self.d_sum), _vp(self.clv_slots[s1]),
            _vp(self.clv_slots[s2]), _vp(self.d_tipVector), None, None,
            ctypes.c_long(self.width), self._gap_row(p), self._gap_row(q),
            self._pre_row(p), self._pre_row(q), self._gapcol_of(s1, False),
            self._gapcol_of(s2, False), s), "sum_dna_save")

    def clv_bytes(self):
        """actual CLV memory footprint (the -S saving)"""
        return sum(t.numel() * 8 for t in self.clv_slots.values())


class ProtCatEngine(DnaGammaEngine):
    """Protein CAT (PSR) engine: span 20, per-site rate categories
    (newviewGTRCATPROT_AVX family).  Mirrors DnaCatEngine with the
    20-state kernels; scratch sized for maxCategories=25 up front."""

    def __init__(self, tips, wgt, model, cptr, per_site_rates, device="cuda",
                 max_ops=None):
        assert model.states == 20
        self.cptr = np.ascontiguousarray(cptr, dtype=np.int32)
        self.per_site_rates = np.ascontiguousarray(per_site_rates,
                                                   dtype=np.float64)
        super().__init__(tips, wgt, model, device=device, max_ops=max_ops)
        self.SPAN = 20
        self.num_cats = len(self.per_site_rates)
        dev = self.device
        self.d_cptr = torch.from_numpy(self.cptr).to(dev)
        MAXC = 25
        self.d_clv = torch.empty((self.ninner, self.width * 20),
                                 dtype=torch.float64, device=dev)
        self.d_pbuf = torch.empty(self._max_ops * MAXC * 800,
                                  dtype=torch.float64, device=dev)
        self.d_diag = torch.empty(MAXC * 20, dtype=torch.float64, device=dev)
        self.d_dtab = torch.empty(MAXC * 20 + 40 + MAXC,
                                  dtype=torch.float64, device=dev)

    def set_site_rates(self, cptr, per_site_rates):
        assert len(per_site_rates) <= 25
        self.cptr = np.ascontiguousarray(cptr, dtype=np.int32)
        self.per_site_rates = np.ascontiguousarray(per_site_rates,
                                                   dtype=np.float64)
        self.num_cats = len(self.per_site_rates)
        self.d_cptr.copy_(torch.from_numpy(self.cptr))

    def newview_traversal(self, entries):
        if not entries:
            return
        assert len(entries) <= self._max_ops, "grow max_ops"
        arr = (TravEntry * len(entries))(*entries)
        m = self.model
        check(lib().examl_hip_newview_traversal_prot_cat(
            ctypes.cast(arr, ctypes.c_void_p), len(entries),
            _np_vp(m.EIGN), _np_vp(m.EI), _np_vp(self.per_site_rates),
            self.num_cats, _vp(self.d_EV), _vp(self.d_tipVector),
            _vp(self.d_cptr), _vp(self.d_clv),
            ctypes.c_long(self.width * 20), _vp(self.d_tips),
            ctypes.c_long(self.width), _vp(self.d_wgt),
            ctypes.c_long(self.width), _vp(self.d_scalers), _vp(self.d_inc),
            _vp(self.d_pbuf), self._stream()), "newview_traversal_prot_cat")

    def evaluate_root(self, tree, p, q, z, all_reduce=False):
        tc, x1s, x2s, tslot, _, pn, qn = self._root_case(tree, p, q)
        self.d_lnl.zero_()
        m = self.model
        check(lib().examl_hip_evaluate_root_prot_cat(
            tc, pn, qn, x1s, x2s, tslot, ctypes.c_double(z),
            _np_vp(m.EIGN), _np_vp(self.per_site_rates), self.num_cats,
            _vp(self.d_tipVector), _vp(self.d_cptr), _vp(self.d_clv),
            ctypes.c_long(self.width * 20), _vp(self.d_tips),
            ctypes.c_long(self.width), _vp(self.d_wgt),
            ctypes.c_long(self.width), _vp(self.d_scalers), _vp(self.d_diag),
            _vp(self.d_partials), _vp(self.d_lnl), self._stream()),
            "evaluate_root_prot_cat")
        if all_reduce and torch.distributed.is_initialized():
            torch.distributed.all_reduce(self.d_lnl)
        return self.d_lnl

    def sum_root(self, tree, p, q):
        if self.d_sum is None:
            self.d_sum = torch.empty(self.width * 20, dtype=torch.float64,
                                     device=self.device)
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if p_tip and q_tip:
            tc, x1s, x2s, t1, t2 = TIP_TIP, -1, -1, p, q
        elif q_tip:
            tc, x1s, x2s, t1, t2 = TIP_INNER, -1, tree.clv_slot(p), q, -1
        elif p_tip:
            tc, x1s, x2s, t1, t2 = TIP_INNER, -1, tree.clv_slot(q), p, -1
        else:
            tc, x1s, x2s, t1, t2 = (INNER_INNER, tree.clv_slot(p),
                                    tree.clv_slot(q), -1, -1)
        check(lib().examl_hip_sum_root_prot_cat(
            tc, x1s, x2s, t1, t2, _vp(self.d_tipVector), _vp(self.d_clv),
            ctypes.c_long(self.width * 20), _vp(self.d_tips),
            ctypes.c_long(self.width), _vp(self.d_sum),
            ctypes.c_long(self.width), self._stream()), "sum_root_prot_cat")

    def core_derivs_async(self, lz):
        m = self.model
        self.d_out2.zero_()
        check(lib().examl_hip_core_root_prot_cat(
            ctypes.c_long(self.width), _vp(self.d_sum), _np_vp(m.EIGN),
            _np_vp(self.per_site_rates), self.num_cats, ctypes.c_double(lz),
            _vp(self.d_wgt), _vp(self.d_cptr), _vp(self.d_dtab),
            _vp(self.d_partials), _vp(self.d_out2), self._stream()),
            "core_root_prot_cat")
        return self.d_out2


class MultiEngine:
    """Fused multi-partition GTRGAMMA engine (DNA or protein): drives ALL partitions'
    per-op work through one kernel launch per (traversal level, tipCase)
    via the mseg executors — the GPU answer to the per-partition dispatch
    loops of newviewIterative (newviewGenericSpecial.c:1064) and execCore
    (makenewzGenericSpecial.c:885) that made partitioned shapes (config 3,
    140.model) launch-bound.  Wraps per-partition DnaGammaEngine buffers;
    P matrices are computed on device, so results agree with the
    per-partition engines to <=1e-11 relative (device exp vs libm), not
    bit-exactly."""

    def __init__(self, engines):
        assert engines and len({e.states for e in engines}) == 1
        self.states = engines[0].states
        assert self.states in (4, 20)
        self.engines = engines
        self.device = engines[0].device
        NP = self.NP = len(engines)
        self.max_ops = max(e._max_ops for e in engines)
        self.d_lnl = torch.zeros(NP, dtype=torch.float64,
                                 device=self.device)
        self.d_out2 = torch.zeros(2 * NP, dtype=torch.float64,
                                  device=self.device)

        def parr(vals):
            return (ctypes.c_void_p * NP)(*vals)

        span = 4 * self.states
        h = ctypes.c_void_p()
        check(lib().examl_hip_multi_create(
            self.states, NP,
            (ctypes.c_long * NP)(*[e.width for e in engines]),
            parr([e.d_clv.data_ptr() for e in engines]),
            (ctypes.c_long * NP)(*[e.width * span for e in engines]),
            parr([e.d_tips.data_ptr() for e in engines]),
            (ctypes.c_long * NP)(*[e.width for e in engines]),
            parr([e.d_wgt.data_ptr() for e in engines]),
            parr([e.d_scalers.data_ptr() for e in engines]),
            parr([e.d_EV.data_ptr() for e in engines]),
            parr([e.d_tipVector.data_ptr() for e in engines]),
            self.max_ops, ctypes.byref(h)), "multi_create")
        self.h = h

    def __del__(self):
        try:
            if getattr(self, "h", None):
                lib().examl_hip_multi_destroy(self.h)
        except Exception:
            pass

    def _stream(self):
        if self.device.type == "cuda":
            return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
        return ctypes.c_void_p(0)

    def sync(self):
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)

    def _models(self, attr):
        return (ctypes.c_void_p * self.NP)(
            *[getattr(e.model, attr).ctypes.data for e in self.engines])

    def _active(self, active):
        if active is None:
            return None
        return (ctypes.c_ubyte * self.NP)(*[1 if a else 0 for a in active])

    def newview_traversal(self, entries, active=None, qz_ov=None,
                          rz_ov=None):
        if not entries:
            return
        assert len(entries) <= self.max_ops
        arr = (TravEntry * len(entries))(*entries)

        def dvec(v):
            if v is None:
                return None
            a = np.ascontiguousarray(v, dtype=np.float64)
            assert a.size == len(entries) * self.NP
            return a

        qz_a, rz_a = dvec(qz_ov), dvec(rz_ov)
        check(lib().examl_hip_newview_traversal_multi(
            self.h, ctypes.cast(arr, ctypes.c_void_p), len(entries),
            self._models("EIGN"), self._models("EI"),
            self._models("gammaRates"), self._active(active),
            _np_vp(qz_a) if qz_a is not None else None,
            _np_vp(rz_a) if rz_a is not None else None,
            self._stream()), "newview_traversal_multi")

    def evaluate_root(self, tree, p, q, z, active=None, all_reduce=False):
        """z: scalar (joint) or per-partition sequence (-M).  Returns the
        per-partition lnL device vector (sum for the total)."""
        e0 = self.engines[0]
        tc, x1s, x2s, tslot, _, pn, qn = e0._root_case(tree, p, q)
        self.d_lnl.zero_()
        zv = np.atleast_1d(np.asarray(z, dtype=np.float64))
        per_part = 1 if zv.size > 1 else 0
        check(lib().examl_hip_evaluate_root_multi(
            self.h, tc, pn, qn, x1s, x2s, tslot, _np_vp(zv), per_part,
            self._models("EIGN"), self._models("gammaRates"),
            self._active(active), _vp(self.d_lnl), self._stream()),
            "evaluate_root_multi")
        if all_reduce and torch.distributed.is_initialized():
            torch.distributed.all_reduce(self.d_lnl)
        return self.d_lnl

    def full_lnl(self, tree, root_edge=None, all_reduce=False):
        entries, (p, q, z) = tree.full_traversal(root_edge)
        self.newview_traversal(entries)
        return self.evaluate_root(tree, p, q, z, all_reduce=all_reduce)

    def sum_root(self, tree, p, q, active=None):
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        if p_tip and q_tip:
            tc, x1s, x2s, t1, t2 = TIP_TIP, -1, -1, p, q
        elif q_tip:
            tc, x1s, x2s, t1, t2 = TIP_INNER, -1, tree.clv_slot(p), q, -1
        elif p_tip:
            tc, x1s, x2s, t1, t2 = TIP_INNER, -1, tree.clv_slot(q), p, -1
        else:
            tc, x1s, x2s, t1, t2 = (INNER_INNER, tree.clv_slot(p),
                                    tree.clv_slot(q), -1, -1)
        check(lib().examl_hip_sum_root_multi(
            self.h, tc, x1s, x2s, t1, t2, self._active(active),
            self._stream()), "sum_root_multi")

    def core_derivs_vec(self, lz, active=None):
        """Per-partition {dlnL, d2lnL} (device vector 2*NP); lz scalar or
        per-partition."""
        self.d_out2.zero_()
        lzv = np.atleast_1d(np.asarray(lz, dtype=np.float64))
        check(lib().examl_hip_core_root_multi(
            self.h, _np_vp(lzv), 1 if lzv.size > 1 else 0,
            self._models("EIGN"), self._models("gammaRates"),
            self._active(active), _vp(self.d_out2), self._stream()),
            "core_root_multi")
        return self.d_out2

    def core_derivs(self, lz, all_reduce=False):
        v = self.core_derivs_vec(lz)
        if all_reduce and torch.distributed.is_initialized():
            torch.distributed.all_reduce(v)
        host = v.cpu().numpy().reshape(self.NP, 2)
        return float(host[:, 0].sum()), float(host[:, 1].sum())

    def makenewz(self, tree, p, q, z0, maxiter=64, all_reduce=False):
        """Joint-branch NR (topLevelMakenewz, makenewzGenericSpecial.c:1133)
        over all partitions with fused sum/core launches."""
        self.sum_root(tree, p, q)
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
            dlnL, d2lnL = self.core_derivs(math.log(z),
                                           all_reduce=all_reduce)
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


class SaveProtEngine(SaveDnaEngine):
    """-S protein GTRGAMMA engine (span 80): the GPU side of the
    *_GAPPED_SAVE protein family (newviewGTRGAMMAPROT_AVX_GAPPED_SAVE,
    avxLikelihood.c:3125; evaluate :1291; sum makenewzGenericSpecial.c:1896)
    with the same per-node gap-vector + prefix-compaction design as
    SaveDnaEngine; undetermined AA code 22 (gapOffset 440)."""

    def __init__(self, tips, wgt, model, device="cuda", max_ops=None):
        assert model.states == 20
        # SaveDnaEngine.__init__ asserts states == 4; replicate its setup
        DnaGammaEngine.__init__(self, tips, wgt, model, device=device,
                                max_ops=max_ops)
        dev = self.device
        self.d_clv = None
        self.gvl = self.width // 32 + 1
        nn = 2 * self.ntips
        host_gap = np.zeros((nn, self.gvl), dtype=np.uint32)
        for t in range(1, self.ntips + 1):
            idx = np.nonzero(tips[t] == 22)[0]  # undetermined AA
            np.bitwise_or.at(host_gap[t], idx // 32,
                             (np.uint32(1) << (idx % 32).astype(np.uint32)))
        self.d_gap = torch.from_numpy(host_gap.view(np.int32)).to(dev)
        host_pre = np.zeros((nn, self.gvl + 1), dtype=np.int32)
        for t in range(1, self.ntips + 1):
            cnt = np.zeros(self.gvl, dtype=np.int64)
            for w in range(self.gvl):
                lo, hi = w * 32, min((w + 1) * 32, self.width)
                bits = int(host_gap[t, w])
                cnt[w] = (hi - lo) - bin(bits & ((1 << (hi - lo)) - 1)
                                         ).count("1")
            host_pre[t, 1:] = np.cumsum(cnt)
        self.d_pre = torch.from_numpy(host_pre).to(dev)
        self.d_gapcol = torch.zeros(self.ninner * 80, dtype=torch.float64,
                                    device=dev)
        self.d_scalegap = torch.zeros(1, dtype=torch.int32, device=dev)
        self.clv_slots = {}

    def _gapcol_of(self, node_or_slot, is_tip):
        if is_tip:
            return ctypes.c_void_p(self.d_tipVector.data_ptr() +
                                   22 * 20 * 8)
        return ctypes.c_void_p(self.d_gapcol.data_ptr() +
                               node_or_slot * 80 * 8)

    def newview_traversal(self, entries):
        if not entries:
            return
        m = self.model
        L = lib()
        s = self._stream()
        self.d_inc.zero_()
        for e in entries:
            qz = math.log(e.qz) if e.qz > ZMIN else math.log(ZMIN)
            rz = math.log(e.rz) if e.rz > ZMIN else math.log(ZMIN)
            hostP = np.empty(3200)
            L.examl_host_make_p(ctypes.c_double(qz), ctypes.c_double(rz),
                                _np_vp(m.gammaRates), _np_vp(m.EI),
                                _np_vp(m.EIGN), 4, _np_vp(hostP),
                                ctypes.c_void_p(hostP.ctypes.data + 1600*8),
                                20)
            d_P = torch.from_numpy(hostP).to(self.device)
            p, q, r = e.pNumber, e.qNumber, e.rNumber
            check(L.examl_hip_gap_and_prefix(
                self._gap_row(q), self._gap_row(r), self._gap_row(p),
                self._pre_row(p), self.gvl, ctypes.c_long(self.width), s),
                "gap_and_prefix")
            total = int(self.d_pre[p, self.gvl].item())
            x3 = self._slot_clv(e.x3Slot, max(total, 1) * 80)
            q_tip = e.tipCase != INNER_INNER
            r_tip = e.tipCase == TIP_TIP
            x1 = None if q_tip else self.clv_slots[e.x1Slot]
            x2 = None if r_tip else self.clv_slots[e.x2Slot]
            t1 = (ctypes.c_void_p(self.d_tips.data_ptr() +
                                  e.x1Slot * self.width) if q_tip else None)
            t2 = (ctypes.c_void_p(self.d_tips.data_ptr() +
                                  e.x2Slot * self.width) if r_tip else None)
            check(L.examl_hip_newview_prot_save(
                e.tipCase,
                _vp(x1) if x1 is not None else None,
                _vp(x2) if x2 is not None else None,
                _vp(x3), _vp(d_P), _vp(self.d_EV), _vp(self.d_tipVector),
                t1, t2, _vp(self.d_wgt), ctypes.c_long(self.width),
                _vp(self.d_inc), self._gap_row(q), self._gap_row(r),
                self._gap_row(p), self._pre_row(q), self._pre_row(r),
                self._pre_row(p),
                self._gapcol_of(q if q_tip else self.tree_slot(e.x1Slot),
                                q_tip),
                self._gapcol_of(r if r_tip else self.tree_slot(e.x2Slot),
                                r_tip),
                self._gapcol_of(e.x3Slot, False), _vp(self.d_scalegap), s),
                "newview_prot_save")
            self._finalize_scaler_one(p, q, r)

    def evaluate_root(self, tree, p, q, z, all_reduce=False):
        tc, x1s, x2s, tslot, _, pn, qn = self._root_case(tree, p, q)
        self.d_lnl.zero_()
        m = self.model
        hostDiag = np.empty(80)
        lib().examl_host_calc_diagptable(ctypes.c_double(z), 20, 4,
                                         _np_vp(m.gammaRates),
                                         _np_vp(m.EIGN), _np_vp(hostDiag))
        d_diag = torch.from_numpy(hostDiag).to(self.device)
        s = self._stream()
        if tc == TIP_INNER:
            x2 = self.clv_slots[x2s]
            inner_node = p if tree.is_tip(q) else q
            check(lib().examl_hip_evaluate_prot_save(
                TIP_INNER, None, _vp(x2), _vp(self.d_tipVector),
                ctypes.c_void_p(self.d_tips.data_ptr() + tslot * self.width),
                _vp(self.d_wgt), _vp(d_diag), ctypes.c_long(self.width),
                None, self._gap_row(inner_node), None,
                self._pre_row(inner_node), None,
                self._gapcol_of(x2s, False), pn, qn, _vp(self.d_scalers),
                _vp(self.d_partials), _vp(self.d_lnl), s),
                "evaluate_prot_save")
        else:
            x1 = self.clv_slots[x1s]
            x2 = self.clv_slots[x2s]
            check(lib().examl_hip_evaluate_prot_save(
                INNER_INNER, _vp(x1), _vp(x2), _vp(self.d_tipVector), None,
                _vp(self.d_wgt), _vp(d_diag), ctypes.c_long(self.width),
                self._gap_row(p), self._gap_row(q), self._pre_row(p),
                self._pre_row(q), self._gapcol_of(x1s, False),
                self._gapcol_of(x2s, False), pn, qn, _vp(self.d_scalers),
                _vp(self.d_partials), _vp(self.d_lnl), s),
                "evaluate_prot_save")
        if all_reduce and torch.distributed.is_initialized():
            torch.distributed.all_reduce(self.d_lnl)
        return self.d_lnl

    def sum_root(self, tree, p, q):
        if self.d_sum is None:
            self.d_sum = torch.empty(self.width * 80, dtype=torch.float64,
                                     device=self.device)
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        s = self._stream()
        if p_tip and q_tip:
            check(lib().examl_hip_sum_prot_save(
                TIP_TIP, _vp(self.d_sum), None, None, _vp(self.d_tipVector),
                ctypes.c_void_p(self.d_tips.data_ptr() + p * self.width),
                ctypes.c_void_p(self.d_tips.data_ptr() + q * self.width),
                ctypes.c_long(self.width), None, None, None, None, None,
                None, s), "sum_prot_save")
            return
        if q_tip or p_tip:
            tip, inner = (q, p) if q_tip else (p, q)
            sl = tree.clv_slot(inner)
            check(lib().examl_hip_sum_prot_save(
                TIP_INNER, _vp(self.d_sum), None, _vp(self.clv_slots[sl]),
                _vp(self.d_tipVector),
                ctypes.c_void_p(self.d_tips.data_ptr() + tip * self.width),
                None, ctypes.c_long(self.width), None, self._gap_row(inner),
                None, self._pre_row(inner), None,
                self._gapcol_of(sl, False), s), "sum_prot_save")
            return
        s1, s2 = tree.clv_slot(p), tree.clv_slot(q)
        check(lib().examl_hip_sum_prot_save(
            INNER_INNER, _vp(self.d_sum), _vp(self.clv_slots[s1]),
            _vp(self.clv_slots[s2]), _vp(self.d_tipVector), None, None,
            ctypes.c_long(self.width), self._gap_row(p), self._gap_row(q),
            self._pre_row(p), self._pre_row(q), self._gapcol_of(s1, False),
            self._gapcol_of(s2, False), s), "sum_prot_save")


class SaveCatEngine(DnaCatEngine):
    """-S CAT (PSR) engine for DNA (span 4) and protein (span 20): the GPU
    side of newviewGTRCAT_AVX_GAPPED_SAVE (avxLikelihood.c:2306) /
    newviewGTRCATPROT_AVX_GAPPED_SAVE (:2607) and their evaluate/sum
    twins, with the saveMem rate-1.0 P pair at slot maxCats (makeP's
    saveMem branch, newviewGenericSpecial.c:140-165).  Per-node gap
    vectors + prefix compaction as in SaveDnaEngine; CAT CLVs are span
    `states` per site."""

    MAXC = 25

    def __init__(self, tips, wgt, model, cptr, per_site_rates,
                 device="cuda", max_ops=None):
        super().__init__(tips, wgt, model, cptr, per_site_rates,
                         device=device, max_ops=max_ops)
        st = self.states
        self.SPAN = st
        dev = self.device
        self.d_clv = None
        if st == 20:  # DnaCatEngine sized these for DNA CAT
            MAXC = self.MAXC
            self.d_diag = torch.empty(MAXC * 20, dtype=torch.float64,
                                      device=dev)
            self.d_dtab = torch.empty(MAXC * 20 + 40 + MAXC,
                                      dtype=torch.float64, device=dev)
        undet = 15 if st == 4 else 22
        self._undet = undet
        self.gvl = self.width // 32 + 1
        nn = 2 * self.ntips
        host_gap = np.zeros((nn, self.gvl), dtype=np.uint32)
        for t in range(1, self.ntips + 1):
            idx = np.nonzero(tips[t] == undet)[0]
            np.bitwise_or.at(host_gap[t], idx // 32,
                             (np.uint32(1) << (idx % 32).astype(np.uint32)))
        self.d_gap = torch.from_numpy(host_gap.view(np.int32)).to(dev)
        host_pre = np.zeros((nn, self.gvl + 1), dtype=np.int32)
        for t in range(1, self.ntips + 1):
            cnt = np.zeros(self.gvl, dtype=np.int64)
            for w in range(self.gvl):
                lo, hi = w * 32, min((w + 1) * 32, self.width)
                bits = int(host_gap[t, w])
                cnt[w] = (hi - lo) - bin(bits & ((1 << (hi - lo)) - 1)
                                         ).count("1")
            host_pre[t, 1:] = np.cumsum(cnt)
        self.d_pre = torch.from_numpy(host_pre).to(dev)
        self.d_gapcol = torch.zeros(self.ninner * st, dtype=torch.float64,
                                    device=dev)
        self.d_scalegap = torch.zeros(1, dtype=torch.int32, device=dev)
        self.clv_slots = {}

    # gap helpers (same layout as SaveDnaEngine)
    def _gap_row(self, node):
        return _vp(self.d_gap[node])

    def _pre_row(self, node):
        return _vp(self.d_pre[node])

    def _gapcol_of(self, node_or_slot, is_tip):
        if is_tip:
            return ctypes.c_void_p(self.d_tipVector.data_ptr() +
                                   self._undet * self.states * 8)
        return ctypes.c_void_p(self.d_gapcol.data_ptr() +
                               node_or_slot * self.states * 8)

    def _slot_clv(self, slot, required):
        t = self.clv_slots.get(slot)
        if t is None or t.numel() != required:
            t = torch.empty(required, dtype=torch.float64,
                            device=self.device)
            self.clv_slots[slot] = t
        return t

    def tree_slot(self, slot):
        return slot

    def _finalize_scaler_one(self, p, q, r):
        inc = self.d_inc[0]
        self.d_scalers[p] = self.d_scalers[q] + self.d_scalers[r] + inc
        self.d_inc.zero_()

    def newview_traversal(self, entries):
        if not entries:
            return
        m = self.model
        st = self.states
        sq = st * st
        L = lib()
        s = self._stream()
        self.d_inc.zero_()
        for e in entries:
            qz = math.log(e.qz) if e.qz > ZMIN else math.log(ZMIN)
            rz = math.log(e.rz) if e.rz > ZMIN else math.log(ZMIN)
            hostP = np.zeros(2 * (self.MAXC + 1) * sq)
            L.examl_host_make_p_save(
                ctypes.c_double(qz), ctypes.c_double(rz),
                _np_vp(self.per_site_rates), _np_vp(m.EI), _np_vp(m.EIGN),
                self.num_cats, _np_vp(hostP),
                ctypes.c_void_p(hostP.ctypes.data + (self.MAXC + 1)*sq*8),
                self.MAXC, st)
            d_P = torch.from_numpy(hostP).to(self.device)
            p, q, r = e.pNumber, e.qNumber, e.rNumber
            check(L.examl_hip_gap_and_prefix(
                self._gap_row(q), self._gap_row(r), self._gap_row(p),
                self._pre_row(p), self.gvl, ctypes.c_long(self.width), s),
                "gap_and_prefix")
            total = int(self.d_pre[p, self.gvl].item())
            x3 = self._slot_clv(e.x3Slot, max(total, 1) * st)
            q_tip = e.tipCase != INNER_INNER
            r_tip = e.tipCase == TIP_TIP
            x1 = None if q_tip else self.clv_slots[e.x1Slot]
            x2 = None if r_tip else self.clv_slots[e.x2Slot]
            t1 = (ctypes.c_void_p(self.d_tips.data_ptr() +
                                  e.x1Slot * self.width) if q_tip else None)
            t2 = (ctypes.c_void_p(self.d_tips.data_ptr() +
                                  e.x2Slot * self.width) if r_tip else None)
            check(L.examl_hip_newview_cat_save(
                st, e.tipCase, _vp(self.d_EV), _vp(self.d_cptr),
                _vp(x1) if x1 is not None else None,
                _vp(x2) if x2 is not None else None,
                _vp(x3), _vp(self.d_tipVector), t1, t2, _vp(self.d_wgt),
                ctypes.c_long(self.width), _vp(d_P), self.MAXC,
                _vp(self.d_inc), self._gap_row(q), self._gap_row(r),
                self._gap_row(p), self._pre_row(q), self._pre_row(r),
                self._pre_row(p),
                self._gapcol_of(q if q_tip else self.tree_slot(e.x1Slot),
                                q_tip),
                self._gapcol_of(r if r_tip else self.tree_slot(e.x2Slot),
                                r_tip),
                self._gapcol_of(e.x3Slot, False), _vp(self.d_scalegap), s),
                "newview_cat_save")
            self._finalize_scaler_one(p, q, r)

    def evaluate_root(self, tree, p, q, z, all_reduce=False):
        tc, x1s, x2s, tslot, _, pn, qn = self._root_case(tree, p, q)
        self.d_lnl.zero_()
        m = self.model
        st = self.states
        hostDiag = np.empty(self.num_cats * st)
        lib().examl_host_calc_diagptable(
            ctypes.c_double(z), st, self.num_cats,
            _np_vp(self.per_site_rates), _np_vp(m.EIGN), _np_vp(hostDiag))
        d_diag = torch.from_numpy(hostDiag).to(self.device)
        s = self._stream()
        if tc == TIP_INNER:
            x2 = self.clv_slots[x2s]
            inner_node = p if tree.is_tip(q) else q
            check(lib().examl_hip_evaluate_cat_save(
                st, _vp(self.d_cptr), _vp(self.d_wgt), None, _vp(x2),
                _vp(self.d_tipVector),
                ctypes.c_void_p(self.d_tips.data_ptr() + tslot * self.width),
                ctypes.c_long(self.width), _vp(d_diag), None,
                self._gap_row(inner_node), None, self._pre_row(inner_node),
                None, self._gapcol_of(x2s, False), pn, qn,
                _vp(self.d_scalers), _vp(self.d_partials), _vp(self.d_lnl),
                s), "evaluate_cat_save")
        else:
            x1 = self.clv_slots[x1s]
            x2 = self.clv_slots[x2s]
            check(lib().examl_hip_evaluate_cat_save(
                st, _vp(self.d_cptr), _vp(self.d_wgt), _vp(x1), _vp(x2),
                _vp(self.d_tipVector), None, ctypes.c_long(self.width),
                _vp(d_diag), self._gap_row(p), self._gap_row(q),
                self._pre_row(p), self._pre_row(q),
                self._gapcol_of(x1s, False), self._gapcol_of(x2s, False),
                pn, qn, _vp(self.d_scalers), _vp(self.d_partials),
                _vp(self.d_lnl), s), "evaluate_cat_save")
        if all_reduce and torch.distributed.is_initialized():
            torch.distributed.all_reduce(self.d_lnl)
        return self.d_lnl

    def sum_root(self, tree, p, q):
        st = self.states
        if self.d_sum is None:
            self.d_sum = torch.empty(self.width * st, dtype=torch.float64,
                                     device=self.device)
        p_tip, q_tip = tree.is_tip(p), tree.is_tip(q)
        s = self._stream()
        if p_tip and q_tip:
            check(lib().examl_hip_sum_cat_save(
                st, TIP_TIP, _vp(self.d_sum), None, None,
                _vp(self.d_tipVector),
                ctypes.c_void_p(self.d_tips.data_ptr() + p * self.width),
                ctypes.c_void_p(self.d_tips.data_ptr() + q * self.width),
                ctypes.c_long(self.width), None, None, None, None, None,
                None, s), "sum_cat_save")
            return
        if q_tip or p_tip:
            tip, inner = (q, p) if q_tip else (p, q)
            sl = tree.clv_slot(inner)
            check(lib().examl_hip_sum_cat_save(
                st, TIP_INNER, _vp(self.d_sum), None,
                _vp(self.clv_slots[sl]), _vp(self.d_tipVector),
                ctypes.c_void_p(self.d_tips.data_ptr() + tip * self.width),
                None, ctypes.c_long(self.width), None,
                self._gap_row(inner), None, self._pre_row(inner), None,
                self._gapcol_of(sl, False), s), "sum_cat_save")
            return
        s1, s2 = tree.clv_slot(p), tree.clv_slot(q)
        check(lib().examl_hip_sum_cat_save(
            st, INNER_INNER, _vp(self.d_sum), _vp(self.clv_slots[s1]),
            _vp(self.clv_slots[s2]), _vp(self.d_tipVector), None, None,
            ctypes.c_long(self.width), self._gap_row(p), self._gap_row(q),
            self._pre_row(p), self._pre_row(q), self._gapcol_of(s1, False),
            self._gapcol_of(s2, False), s), "sum_cat_save")

    def clv_bytes(self):
        return sum(t.numel() * 8 for t in self.clv_slots.values())

    def core_derivs_async(self, lz):
        """NR derivatives on the dense sumBuffer: the dense CAT core
        kernels apply unchanged (sum_cat_save writes densely)."""
        if self.states == 4:
            return super().core_derivs_async(lz)
        m = self.model
        self.d_out2.zero_()
        check(lib().examl_hip_core_root_prot_cat(
            ctypes.c_long(self.width), _vp(self.d_sum), _np_vp(m.EIGN),
            _np_vp(self.per_site_rates), self.num_cats, ctypes.c_double(lz),
            _vp(self.d_wgt), _vp(self.d_cptr), _vp(self.d_dtab),
            _vp(self.d_partials), _vp(self.d_out2), self._stream()),
            "core_root_prot_cat")
        return self.d_out2


MultiDnaEngine = MultiEngine  # backward-compatible alias
