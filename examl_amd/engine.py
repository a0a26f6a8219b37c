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
