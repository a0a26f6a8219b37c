"""-S GPU engines for the remaining kernel families:
protein GTRGAMMA SAVE and DNA/protein CAT SAVE.

- Protein GAMMA: SaveProtEngine must be bit-transparent against the dense
  prot engine on the same gappy inputs (the same property the DNA SAVE
  engine test pins; gap-site columns are mathematically identical to the
  per-site dense computation).
- CAT (PSR): dense CAT and SAVE CAT genuinely differ at gap sites (the
  reference computes the shared gap column with the saveMem rate-1.0 P
  pair, newviewGenericSpecial.c:140-165), so the GPU kernels are pinned
  BIT-EXACT against the oracle CAT SAVE kernels (which are bit-exact vs
  the reference's *_GAPPED_SAVE family, tests/test_prot_cat.py)."""

import ctypes as C
import math
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import oracle as O  # noqa: E402

import examl_amd as ea  # noqa: E402
from tests.helpers import make_synthetic, _model_arrays  # noqa: E402

pytestmark = pytest.mark.gpu


def _gappy_aa(ntips, width, frac, seed):
    rng = np.random.default_rng(seed)
    tips = np.zeros((ntips + 1, width), dtype=np.uint8)
    for t in range(1, ntips + 1):
        tips[t] = rng.integers(1, 23, width).astype(np.uint8)
        tips[t][rng.random(width) < frac] = 22
    wgt = rng.integers(1, 4, width).astype(np.int32)
    return tips, wgt


def _gappy_dna(ntips, width, frac, seed):
    tips, wgt = make_synthetic(ntips, width, seed=seed)
    rng = np.random.default_rng(seed + 1)
    for t in range(1, ntips + 1):
        tips[t][rng.random(width) < frac] = 15
    return tips, wgt


def _aa_model():
    aa = np.load(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examl_amd", "data", "aa_models.npz"))
    return ea.ProtGtrModel(aa["frequencies"][4], aa["rates190"][4], 0.9)


def test_prot_save_engine_transparent_gpu():
    """SaveProtEngine == dense prot engine (lnL, NR branch length, and a
    smaller CLV footprint) on 30%-gappy AA data."""
    import torch

    from examl_amd.search import TreeSearch
    assert torch.cuda.is_available()
    tips, wgt = _gappy_aa(14, 3000, 0.30, 7)
    t1 = ea.PhyloTree.random(14, seed=4, rng_z=True)
    t2 = ea.PhyloTree.random(14, seed=4, rng_z=True)
    m = _aa_model()
    e1 = ea.DnaGammaEngine(tips, wgt, m, device="cuda:0")
    e2 = ea.SaveProtEngine(tips, wgt, m, device="cuda:0")
    ts1 = TreeSearch(t1, [e1])
    ts2 = TreeSearch(t2, [e2])
    l1 = ts1.evaluate_generic(full=True)
    l2 = ts2.evaluate_generic(full=True)
    assert l1 == l2, (l1, l2)
    p, q = 1, next(iter(t1.adj[1]))
    z1 = ts1.makenewz_generic(p, q, t1.get_z(p, q), 64)
    z2 = ts2.makenewz_generic(p, q, t2.get_z(p, q), 64)
    assert z1 == z2
    a1 = ts1.tree_evaluate(1.0)
    a2 = ts2.tree_evaluate(1.0)
    assert a1 == a2, (a1, a2)
    assert e2.clv_bytes() < e1.d_clv.numel() * 8


@pytest.mark.parametrize("states", [4, 20])
def test_cat_save_kernels_gpu_vs_oracle(states):
    """GPU CAT SAVE kernels (newview all tipCases + evaluate + sum) are
    bit-exact vs the oracle CAT SAVE restatement on gappy data with a
    random rate categorization."""
    import torch
    assert torch.cuda.is_available()
    dev = "cuda:0"
    rng = np.random.default_rng(100 + states)
    n = 700
    MAXC = 25
    num_cats = 7
    undet = 15 if states == 4 else 22
    span = states
    sq = states * states
    if states == 4:
        m = ea.DnaGtrModel([0.3, 0.2, 0.26, 0.24],
                           [1.2, 2.4, 0.7, 0.9, 3.1, 1.0], 0.8)
    else:
        m = _aa_model()
    EIGN, EV, EI, tipVector, _ = _model_arrays(m)
    rates = np.sort(rng.uniform(0.05, 4.0, num_cats))
    cptr = rng.integers(0, num_cats, n).astype(np.int32)
    wgt = rng.integers(1, 4, n).astype(np.int32)
    hi = 16 if states == 4 else 23
    t1c = rng.integers(1, hi, n).astype(np.uint8)
    t2c = rng.integers(1, hi, n).astype(np.uint8)
    t1c[rng.random(n) < 0.3] = undet
    t2c[rng.random(n) < 0.3] = undet
    gvl = n // 32 + 1

    def gap_of(codes):
        gv = np.zeros(gvl, dtype=np.uint32)
        idx = np.nonzero(codes == undet)[0]
        np.bitwise_or.at(gv, idx // 32,
                         (np.uint32(1) << (idx % 32).astype(np.uint32)))
        return gv

    def pre_of(gv):
        pre = np.zeros(gvl + 1, dtype=np.int32)
        c = 0
        for w in range(gvl):
            pre[w] = c
            lo, hi_ = w * 32, min((w + 1) * 32, n)
            bits = int(gv[w])
            c += (hi_ - lo) - bin(bits & ((1 << (hi_ - lo)) - 1)).count("1")
        pre[gvl] = c
        return pre

    g1, g2 = gap_of(t1c), gap_of(t2c)
    g3 = g1 & g2
    pre1, pre2, pre3 = pre_of(g1), pre_of(g2), pre_of(g3)
    nz1, nz2, nz3 = int(pre1[gvl]), int(pre2[gvl]), int(pre3[gvl])

    # P with the rate-1.0 saveMem pair
    qz, rz = math.log(0.43), math.log(0.81)
    left = O.aligned((MAXC + 1) * sq)
    right = O.aligned((MAXC + 1) * sq)
    O._orc.oracle_make_p_save(
        C.c_double(qz), C.c_double(rz),
        rates.ctypes.data_as(C.POINTER(C.c_double)),
        EI.ctypes.data_as(C.POINTER(C.c_double)),
        EIGN.ctypes.data_as(C.POINTER(C.c_double)), C.c_int(num_cats),
        left.ctypes.data_as(C.POINTER(C.c_double)),
        right.ctypes.data_as(C.POINTER(C.c_double)), C.c_int(MAXC),
        C.c_int(states))

    # child CLVs (compacted) + gap columns
    x1 = O.aligned(max(nz1, 1) * span)
    x1[:] = rng.uniform(0.05, 1.0, x1.size)
    x2 = O.aligned(max(nz2, 1) * span)
    x2[:] = rng.uniform(0.05, 1.0, x2.size)
    gc1 = np.ascontiguousarray(rng.uniform(0.05, 1.0, span))
    gc2 = np.ascontiguousarray(rng.uniform(0.05, 1.0, span))

    import torch
    d = lambda a: torch.from_numpy(np.ascontiguousarray(a)).to(dev)
    dP = d(np.concatenate([left, right]))
    dEV, dTV = d(EV), d(tipVector)
    d_cptr, d_wgt = d(cptr), d(wgt)
    d_t1, d_t2 = d(t1c), d(t2c)
    dg1, dg2, dg3 = d(g1.view(np.int32)), d(g2.view(np.int32)), \
        d(g3.view(np.int32))
    dp1, dp2, dp3 = d(pre1), d(pre2), d(pre3)
    dx1, dx2 = d(x1), d(x2)
    dgc1, dgc2 = d(gc1), d(gc2)
    vp = lambda t: C.c_void_p(t.data_ptr())
    L = ea.lib()

    orc_nv = getattr(O._orc, f"oracle_newview_"
                             f"{'dna' if states == 4 else 'prot'}_cat_save")
    for tc in (ea.TIP_TIP, ea.TIP_INNER, ea.INNER_INNER):
        ref_x3 = O.aligned(max(nz3, 1) * span)
        ref_gc3 = np.zeros(span)
        ref_inc = C.c_int(0)
        dpd = lambda a: a.ctypes.data_as(C.POINTER(C.c_double))
        upd = lambda a: a.ctypes.data_as(C.POINTER(C.c_uint))
        u8d = lambda a: a.ctypes.data_as(C.POINTER(C.c_ubyte))
        orc_nv(C.c_int(tc), dpd(EV),
               cptr.ctypes.data_as(C.POINTER(C.c_int)),
               dpd(tipVector if tc == ea.TIP_TIP else
                   (tipVector if tc == ea.TIP_INNER else x1)),
               dpd(x2), dpd(ref_x3), dpd(tipVector), u8d(t1c), u8d(t2c),
               C.c_int(n), dpd(left), dpd(right),
               wgt.ctypes.data_as(C.POINTER(C.c_int)), C.byref(ref_inc),
               upd(g1), upd(g2), upd(g3), dpd(gc1), dpd(gc2), dpd(ref_gc3),
               C.c_int(MAXC))
        # oracle signature: (tipCase, EV, cptr, x1, x2, x3, tipVector, ...)
        # x1 operand is the compacted CLV only for INNER_INNER
        gpu_x3 = torch.zeros(max(nz3, 1) * span, dtype=torch.float64,
                             device=dev)
        gpu_gc3 = torch.zeros(span, dtype=torch.float64, device=dev)
        d_inc = torch.zeros(1, dtype=torch.int32, device=dev)
        d_sg = torch.zeros(1, dtype=torch.int32, device=dev)
        ea.check(L.examl_hip_newview_cat_save(
            states, tc, vp(dEV), vp(d_cptr),
            vp(dx1) if tc == ea.INNER_INNER else None,
            vp(dx2) if tc != ea.TIP_TIP else None,
            vp(gpu_x3), vp(dTV),
            vp(d_t1) if tc != ea.INNER_INNER else None,
            vp(d_t2) if tc == ea.TIP_TIP else None,
            vp(d_wgt), C.c_long(n), vp(dP), MAXC, vp(d_inc), vp(dg1),
            vp(dg2), vp(dg3), vp(dp1), vp(dp2), vp(dp3), vp(dgc1),
            vp(dgc2), vp(gpu_gc3), vp(d_sg), None), "newview_cat_save")
        torch.cuda.synchronize()
        got = gpu_x3.cpu().numpy()
        assert np.array_equal(got, ref_x3), (states, tc)
        assert np.array_equal(gpu_gc3.cpu().numpy(), ref_gc3), (states, tc)
        assert int(d_inc.cpu()) == ref_inc.value, (states, tc)

    # evaluate (TIP_INNER root) + sum (INNER_INNER): bit-exact sum; lnL to
    # 1e-12 (different but fixed reduction order)
    diag = np.empty(num_cats * span)
    ea.lib().examl_host_calc_diagptable(
        C.c_double(0.61), states, num_cats,
        rates.ctypes.data_as(C.c_void_p),
        EIGN.ctypes.data_as(C.c_void_p),
        diag.ctypes.data_as(C.c_void_p))
    orc_ev = getattr(O._orc, f"oracle_evaluate_"
                             f"{'dna' if states == 4 else 'prot'}_cat_save")
    orc_ev.restype = C.c_double
    ref_lnl = orc_ev(
        cptr.ctypes.data_as(C.POINTER(C.c_int)),
        wgt.ctypes.data_as(C.POINTER(C.c_int)), None,
        x2.ctypes.data_as(C.POINTER(C.c_double)),
        tipVector.ctypes.data_as(C.POINTER(C.c_double)),
        t1c.ctypes.data_as(C.POINTER(C.c_ubyte)), C.c_int(n),
        diag.ctypes.data_as(C.POINTER(C.c_double)),
        C.cast(None, C.POINTER(C.c_double)),
        gc2.ctypes.data_as(C.POINTER(C.c_double)),
        C.cast(None, C.POINTER(C.c_uint)),
        g2.ctypes.data_as(C.POINTER(C.c_uint)))
    d_diag = d(diag)
    d_partials = torch.zeros(2 * 8192, dtype=torch.float64, device=dev)
    d_lnl = torch.zeros(1, dtype=torch.float64, device=dev)
    ea.check(L.examl_hip_evaluate_cat_save(
        states, vp(d_cptr), vp(d_wgt), None, vp(dx2), vp(dTV), vp(d_t1),
        C.c_long(n), vp(d_diag), None, vp(dg2), None, vp(dp2), None,
        vp(dgc2), 0, 0, None, vp(d_partials), vp(d_lnl), None),
        "evaluate_cat_save")
    torch.cuda.synchronize()
    got_lnl = float(d_lnl.cpu())
    assert abs(got_lnl - ref_lnl) <= 1e-12 * max(1.0, abs(ref_lnl)), \
        (states, got_lnl, ref_lnl)

    ref_sum = O.aligned(n * span)
    orc_sum = getattr(O._orc, f"oracle_sum_"
                              f"{'dna' if states == 4 else 'prot'}_cat_save")
    orc_sum(C.c_int(ea.INNER_INNER),
            ref_sum.ctypes.data_as(C.POINTER(C.c_double)),
            x1.ctypes.data_as(C.POINTER(C.c_double)),
            x2.ctypes.data_as(C.POINTER(C.c_double)),
            tipVector.ctypes.data_as(C.POINTER(C.c_double)),
            C.cast(None, C.POINTER(C.c_ubyte)),
            C.cast(None, C.POINTER(C.c_ubyte)), C.c_int(n),
            gc1.ctypes.data_as(C.POINTER(C.c_double)),
            gc2.ctypes.data_as(C.POINTER(C.c_double)),
            g1.ctypes.data_as(C.POINTER(C.c_uint)),
            g2.ctypes.data_as(C.POINTER(C.c_uint)))
    d_sum = torch.zeros(n * span, dtype=torch.float64, device=dev)
    ea.check(L.examl_hip_sum_cat_save(
        states, ea.INNER_INNER, vp(d_sum), vp(dx1), vp(dx2), vp(dTV), None,
        None, C.c_long(n), vp(dg1), vp(dg2), vp(dp1), vp(dp2), vp(dgc1),
        vp(dgc2), None), "sum_cat_save")
    torch.cuda.synchronize()
    assert np.array_equal(d_sum.cpu().numpy(), ref_sum), states
