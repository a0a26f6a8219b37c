"""CAT (PSR, -m PSR) kernel parity — SURVEY §8f row 1.  CPU: the oracle
restatement vs the reference golden vectors (bit-exact).  GPU: the HIP CAT
kernels vs the same goldens."""

import ctypes
import os

import numpy as np
import pytest

import examl_amd as ea
import oracle as O


@pytest.fixture(scope="module")
def kern(golden_dir):
    return np.load(os.path.join(golden_dir, "kernels_dna_cat.npz"))


def _al(a):
    out = O.aligned(a.shape, a.dtype)
    out[:] = a
    return out


@pytest.fixture(scope="module")
def model(golden_dir):
    d = np.load(os.path.join(golden_dir, "model_dna.npz"))
    return ea.DnaGtrModel(d["m1_freqs"], d["m1_rates6"], float(d["m1_alpha"]))


# ---------------- CPU: oracle vs golden -----------------------------------

@pytest.mark.parametrize("tag", ["norm", "tiny"])
@pytest.mark.parametrize("tc", [O.TIP_TIP, O.TIP_INNER, O.INNER_INNER])
def test_oracle_newview_cat(kern, model, tag, tc):
    EV = _al(model.EV)
    tipVector = _al(model.tipVector)
    left = _al(kern["left"])
    right = _al(kern["right"])
    cptr = np.ascontiguousarray(kern["cptr"])
    x1 = _al(kern[f"{tag}_x1"])
    x2 = _al(kern[f"{tag}_x2"])
    wgt = np.ascontiguousarray(kern[f"{tag}_wgt"])
    t1 = np.ascontiguousarray(kern[f"{tag}_tipX1"])
    t2 = np.ascontiguousarray(kern[f"{tag}_tipX2"])
    n = len(wgt)
    args = {
        O.TIP_TIP: (None, None, t1, t2),
        O.TIP_INNER: (None, x2, t1, None),
        O.INNER_INNER: (x1, x2, None, None),
    }[tc]
    x3, inc = O.newview_dna_cat(tc, EV, cptr, args[0], args[1], tipVector,
                                args[2], args[3], n, left, right, wgt)
    assert inc == int(kern[f"{tag}_newview_tc{tc}_inc"])
    assert np.array_equal(x3, kern[f"{tag}_newview_tc{tc}_x3"])


def test_oracle_evaluate_sum_core_cat(kern, model):
    tipVector = _al(model.tipVector)
    EIGN = _al(model.EIGN)
    rptr = _al(kern["rptr"])
    num_cats = int(kern["num_cats"])
    cptr = np.ascontiguousarray(kern["cptr"])
    x1 = _al(kern["norm_x1"])
    x2 = _al(kern["norm_x2"])
    wgt = np.ascontiguousarray(kern["norm_wgt"])
    t1 = np.ascontiguousarray(kern["norm_tipX1"])
    t2 = np.ascontiguousarray(kern["norm_tipX2"])
    diag = _al(kern["diag"])
    n = len(wgt)
    assert O.evaluate_dna_cat(cptr, wgt, x1, x2, tipVector, None, n,
                              diag) == float(kern["eval_II"])
    assert O.evaluate_dna_cat(cptr, wgt, None, x2, tipVector, t1, n,
                              diag) == float(kern["eval_TIP"])
    # makeP with numCats != 4 must also match
    left, right = O.make_p(np.log(float(kern["z_q"])),
                           np.log(float(kern["z_r"])), rptr, _al(model.EI),
                           EIGN, num_cats, 4)
    assert np.array_equal(left, kern["left"])
    assert np.array_equal(right, kern["right"])
    for tc, a1, a2, u1, u2 in [
        (O.TIP_TIP, None, None, t1, t2),
        (O.TIP_INNER, None, x2, t1, None),
        (O.INNER_INNER, x1, x2, None, None),
    ]:
        st = O.sum_dna_cat(tc, a1, a2, tipVector, u1, u2, n)
        assert np.array_equal(st, kern[f"sum_tc{tc}"])
        d1, d2 = O.core_dna_cat(n, num_cats, st, wgt, rptr, EIGN, cptr,
                                float(kern["lz_core"]))
        assert d1 == float(kern[f"core_tc{tc}_d1"])
        assert d2 == float(kern[f"core_tc{tc}_d2"])


# ---------------- GPU: HIP kernels vs golden -------------------------------

def vp(t):
    return ctypes.c_void_p(t.data_ptr())


@pytest.mark.gpu
@pytest.mark.parametrize("tag", ["norm", "tiny"])
@pytest.mark.parametrize("tc", [ea.TIP_TIP, ea.TIP_INNER, ea.INNER_INNER])
def test_gpu_newview_cat_bit_exact(kern, model, tag, tc):
    import torch
    dev = torch.device("cuda:0")

    def td(a):
        return torch.from_numpy(np.ascontiguousarray(a)).to(dev)

    n = len(kern[f"{tag}_wgt"])
    num_cats = int(kern["num_cats"])
    d_x1 = td(kern[f"{tag}_x1"])
    d_x2 = td(kern[f"{tag}_x2"])
    d_x3 = torch.zeros(n * 4, dtype=torch.float64, device=dev)
    d_P = td(np.concatenate([kern["left"], kern["right"]]))
    d_EV = td(model.EV)
    d_tv = td(model.tipVector)
    d_cptr = td(kern["cptr"])
    d_t1 = td(kern[f"{tag}_tipX1"])
    d_t2 = td(kern[f"{tag}_tipX2"])
    d_wgt = td(kern[f"{tag}_wgt"])
    d_inc = torch.zeros(1, dtype=torch.int32, device=dev)
    null = ctypes.c_void_p(0)
    ea.check(ea.lib().examl_hip_newview_dna_cat(
        tc, vp(d_EV), vp(d_cptr),
        vp(d_x1) if tc == ea.INNER_INNER else null,
        vp(d_x2) if tc != ea.TIP_TIP else null,
        vp(d_x3), vp(d_tv),
        vp(d_t1) if tc != ea.INNER_INNER else null,
        vp(d_t2) if tc == ea.TIP_TIP else null,
        ctypes.c_long(n), vp(d_P), num_cats, vp(d_wgt), vp(d_inc),
        ctypes.c_void_p(0)), "newview_cat")
    torch.cuda.synchronize()
    assert int(d_inc.item()) == int(kern[f"{tag}_newview_tc{tc}_inc"])
    assert np.array_equal(d_x3.cpu().numpy(), kern[f"{tag}_newview_tc{tc}_x3"])


@pytest.mark.gpu
@pytest.mark.parametrize("tc", [ea.TIP_TIP, ea.TIP_INNER, ea.INNER_INNER])
def test_gpu_evaluate_sum_core_cat(kern, model, tc):
    import torch
    dev = torch.device("cuda:0")

    def td(a):
        return torch.from_numpy(np.ascontiguousarray(a)).to(dev)

    n = len(kern["norm_wgt"])
    num_cats = int(kern["num_cats"])
    d_x1 = td(kern["norm_x1"])
    d_x2 = td(kern["norm_x2"])
    d_tv = td(model.tipVector)
    d_cptr = td(kern["cptr"])
    d_t1 = td(kern["norm_tipX1"])
    d_t2 = td(kern["norm_tipX2"])
    d_wgt = td(kern["norm_wgt"])
    d_diag = td(kern["diag"])
    d_lnl = torch.zeros(1, dtype=torch.float64, device=dev)
    d_part = torch.zeros(2 * 8192, dtype=torch.float64, device=dev)
    null = ctypes.c_void_p(0)
    if tc == ea.INNER_INNER:  # evaluate only has tip/inner bodies
        ea.check(ea.lib().examl_hip_evaluate_dna_cat(
            vp(d_cptr), vp(d_wgt), vp(d_x1), vp(d_x2), vp(d_tv), null,
            ctypes.c_long(n), vp(d_diag), num_cats, null, null,
            ctypes.c_double(0.0), vp(d_part), vp(d_lnl), ctypes.c_void_p(0)),
            "evaluate_cat")
        torch.cuda.synchronize()
        assert np.isclose(d_lnl.item(), float(kern["eval_II"]), rtol=1e-12)
    elif tc == ea.TIP_INNER:
        ea.check(ea.lib().examl_hip_evaluate_dna_cat(
            vp(d_cptr), vp(d_wgt), null, vp(d_x2), vp(d_tv), vp(d_t1),
            ctypes.c_long(n), vp(d_diag), num_cats, null, null,
            ctypes.c_double(0.0), vp(d_part), vp(d_lnl), ctypes.c_void_p(0)),
            "evaluate_cat")
        torch.cuda.synchronize()
        assert np.isclose(d_lnl.item(), float(kern["eval_TIP"]), rtol=1e-12)

    d_sum = torch.zeros(n * 4, dtype=torch.float64, device=dev)
    ea.check(ea.lib().examl_hip_sum_dna_cat(
        tc, vp(d_sum),
        vp(d_x1) if tc == ea.INNER_INNER else null,
        vp(d_x2) if tc != ea.TIP_TIP else null,
        vp(d_tv),
        vp(d_t1) if tc != ea.INNER_INNER else null,
        vp(d_t2) if tc == ea.TIP_TIP else null,
        ctypes.c_long(n), ctypes.c_void_p(0)), "sum_cat")
    torch.cuda.synchronize()
    assert np.array_equal(d_sum.cpu().numpy(), kern[f"sum_tc{tc}"])

    EIGN = np.ascontiguousarray(model.EIGN)
    rptr = np.ascontiguousarray(kern["rptr"])
    d_dtab = torch.zeros(num_cats * 4 + 8 + num_cats, dtype=torch.float64,
                         device=dev)
    d_out2 = torch.zeros(2, dtype=torch.float64, device=dev)
    ea.check(ea.lib().examl_hip_core_root_dna_cat(
        ctypes.c_long(n), vp(d_sum),
        EIGN.ctypes.data_as(ctypes.c_void_p),
        rptr.ctypes.data_as(ctypes.c_void_p), num_cats,
        ctypes.c_double(float(kern["lz_core"])), vp(d_wgt), vp(d_cptr),
        vp(d_dtab), vp(d_part), vp(d_out2), ctypes.c_void_p(0)), "core_cat")
    torch.cuda.synchronize()
    out = d_out2.cpu().numpy()
    assert np.isclose(out[0], float(kern[f"core_tc{tc}_d1"]), rtol=1e-11)
    assert np.isclose(out[1], float(kern[f"core_tc{tc}_d2"]), rtol=1e-11)


@pytest.mark.gpu
def test_gpu_cat_full_pipeline_vs_oracle(model):
    """DnaCatEngine end to end (traversal executor + evaluate + scalers)
    against the oracle CAT replay, plus makenewz agreement."""
    import math
    import torch
    from tests.helpers import make_synthetic, oracle_cat_full_lnl
    dev = torch.device("cuda:0")
    ntips, width, num_cats = 22, 6000, 11
    rng = np.random.default_rng(71)
    tips, wgt = make_synthetic(ntips, width, seed=61)
    cptr = rng.integers(0, num_cats, width).astype(np.int32)
    rates = rng.uniform(0.05, 4.0, num_cats)
    tree = ea.PhyloTree.random(ntips, seed=19, rng_z=True)
    eng = ea.DnaCatEngine(tips, wgt, model, cptr, rates, device=dev)
    entries, root = tree.full_traversal()
    lnl = eng.full_lnl(tree).item()
    ref, clv_ref, scalers_ref = oracle_cat_full_lnl(
        entries, root, tree, model, tips, wgt, cptr, rates,
        return_state=True)
    assert math.isfinite(lnl) and lnl < 0
    assert abs(lnl - ref) / abs(ref) < 1e-11
    clv = eng.d_clv.cpu().numpy()
    for slot, x in clv_ref.items():
        assert np.array_equal(clv[slot], x), f"CAT CLV slot {slot} differs"
    sc = eng.d_scalers.cpu().numpy()
    for node in range(ntips + 1, 2 * ntips - 1):
        assert sc[node] == scalers_ref[node]
    # NR derivatives through the CAT sum/core path
    p, q, z0 = root
    eng.sum_root(tree, p, q)
    d_gpu = eng.core_derivs(float(np.log(z0)))
    import oracle as O
    from tests.helpers import _model_arrays
    EIGN, EV, EI, tipVector, _ = _model_arrays(model)
    rptr = O.aligned(num_cats); rptr[:] = rates
    p_tip = tree.is_tip(p)
    x1 = None if p_tip else clv_ref[tree.clv_slot(p)]
    st = O.sum_dna_cat(ea.TIP_INNER if p_tip else ea.INNER_INNER,
                       x1, clv_ref[tree.clv_slot(q)] if not tree.is_tip(q)
                       else None, tipVector,
                       np.ascontiguousarray(tips[p]) if p_tip else None,
                       None, width)
    d_ref = O.core_dna_cat(width, num_cats, st,
                           np.ascontiguousarray(wgt, np.int32), rptr, EIGN,
                           np.ascontiguousarray(cptr, np.int32),
                           float(np.log(z0)))
    assert np.isclose(d_gpu[0], d_ref[0], rtol=1e-10)
    assert np.isclose(d_gpu[1], d_ref[1], rtol=1e-10)


# ---------------------------------------------------------------------------
# evaluatePartialGeneric (the CAT per-site rate probe) — product host fn vs
# the reference's own evaluatePartialGTRCAT
# ---------------------------------------------------------------------------

@pytest.mark.skipif(not O.have_ref(), reason="reference libref.so not built")
def test_evaluate_partial_bit_exact_vs_reference():
    """examl_host_evaluate_partial_dna_cat (model_prep.cpp, restating
    evaluatePartialGenericSpecial.c:50-257) is BIT-EXACT against the
    reference's evaluatePartialGTRCAT over a 12-taxon traversal, 5 sites x
    3 probe rates."""
    from examl_amd.search import TreeSearch
    from tests.helpers import make_synthetic, OracleEngine, _model_arrays

    NUMB = 256  # the reference's NUM_BRANCHES (axml.h:126)

    class RefTI(ctypes.Structure):
        _fields_ = [("tipCase", ctypes.c_int), ("pNumber", ctypes.c_int),
                    ("qNumber", ctypes.c_int), ("rNumber", ctypes.c_int),
                    ("qz", ctypes.c_double * NUMB),
                    ("rz", ctypes.c_double * NUMB)]

    ntips, width = 12, 200
    tips, wgt = make_synthetic(ntips, width, seed=77)
    model = ea.DnaGtrModel([0.3, 0.2, 0.26, 0.24],
                           [1.2, 2.4, 0.7, 0.9, 3.1, 1.0], 1.0)
    tree = ea.PhyloTree.random(ntips, seed=4, rng_z=True)
    ts = TreeSearch(tree, [OracleEngine(tips, wgt, model)])
    ts.evaluate_generic(full=True)
    p = 1
    q = next(iter(tree.adj[1]))
    entries = []
    ts.oriented.clear()
    ts._collect(p, q, False, entries)
    ts._collect(q, p, False, entries)
    root_z = tree.get_z(p, q)

    # reference td[0]: ti[0] = root record {pNumber, qNumber, qz[0]}
    # (evaluatePartialGenericSpecial.c:264-270), ti[1..] = newview ops
    n_ti = len(entries) + 1
    arr = (RefTI * n_ti)()
    arr[0].tipCase = 0
    arr[0].pNumber = p
    arr[0].qNumber = q
    arr[0].qz[0] = root_z
    for k, e in enumerate(entries):
        arr[k + 1].tipCase = e.tipCase
        arr[k + 1].pNumber = e.pNumber
        arr[k + 1].qNumber = e.qNumber
        arr[k + 1].rNumber = e.rNumber
        arr[k + 1].qz[0] = e.qz
        arr[k + 1].rz[0] = e.rz
    rows = (ctypes.POINTER(ctypes.c_ubyte) * (ntips + 1))()
    tipsC = np.ascontiguousarray(tips)
    for t in range(1, ntips + 1):
        rows[t] = (ctypes.c_ubyte * width).from_buffer(tipsC[t])
    EIGN, EV, EI, tipVector, _ = _model_arrays(model)
    ref = O._ref
    ref.evaluatePartialGTRCAT.restype = ctypes.c_double
    L = ea.lib()

    def dp(a):
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_double))

    ops_arr = (ea.TravEntry * len(entries))(*entries)
    for site in [0, 3, 17, 100, 199]:
        for ki in [0.2, 1.0, 2.7]:
            r = ref.evaluatePartialGTRCAT(
                ctypes.c_int(site), ctypes.c_double(ki), ctypes.c_int(n_ti),
                arr, ctypes.c_double(root_z), ctypes.c_int(int(wgt[site])),
                dp(EIGN), dp(EI), dp(EV), dp(tipVector), rows,
                ctypes.c_int(0), ctypes.c_int(ntips))
            m = L.examl_host_evaluate_partial_dna_cat(
                ctypes.cast(ops_arr, ctypes.c_void_p), len(entries),
                ctypes.c_int(p), ctypes.c_int(q), ctypes.c_double(root_z),
                ctypes.c_long(site), ctypes.c_double(ki),
                ctypes.c_int(int(wgt[site])),
                model.EIGN.ctypes.data_as(ctypes.c_void_p),
                model.EI.ctypes.data_as(ctypes.c_void_p),
                model.EV.ctypes.data_as(ctypes.c_void_p),
                model.tipVector.ctypes.data_as(ctypes.c_void_p),
                tipsC.ctypes.data_as(ctypes.c_void_p),
                ctypes.c_long(width), ctypes.c_int(ntips))
            assert r == m, (site, ki, r, m)


def test_cat_optimizer_small_cpu():
    """optimizeRateCategories + the CAT modOpt branch on the oracle engines:
    lnL monotone through categorization, numberOfCategories grows, and the
    mean per-site rate is 1 after updatePerSiteRates."""
    from examl_amd.search import TreeSearch
    from tests.helpers import make_synthetic, OracleCatEngine

    tips, wgt = make_synthetic(ntips=12, width=120, seed=7)
    tree = ea.PhyloTree.random(12, seed=3)
    model = ea.DnaGtrModel(np.full(4, 0.25),
                           np.array([1.1, 2.2, 0.7, 1.3, 0.9, 1.0]), 1.0)
    eng = OracleCatEngine(tips, wgt, model, np.zeros(120, dtype=np.int32),
                          np.array([1.0]))
    ts = TreeSearch(tree, [eng], rate_het="CAT")
    l0 = ts.evaluate_generic(full=True)
    ts.optimize_rate_categories(25)
    assert ts.likelihood >= l0 - 1e-9
    assert eng.num_cats > 1
    # updatePerSiteRates (optimizeModel.c:2060): weighted mean rate == 1
    mean = float((eng.host_wgt * eng.per_site_rates[eng.cptr]).sum()
                 / eng.host_wgt.sum())
    assert abs(mean - 1.0) < 1e-10
    l1 = ts.mod_opt(0.5)
    assert l1 > l0
