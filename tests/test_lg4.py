"""LG4M / LG4X (one matrix per gamma category, Le/Dang/Gascuel 2012):

- oracle LG4 kernels bit-exact vs the reference's own
  (newviewGTRGAMMAPROT_AVX_LG4, evaluateGTRGAMMAPROT_LG4, sumGAMMAPROT_LG4,
  coreGTRGAMMAPROT_LG4, makeP_FlexLG4, calcDiagptableFlex_LG4)
- the -f E pipeline on testData/140 re-partitioned as LG4M + LG4X + WAG
  (tests/golden/140lg4.binary, built with the reference's own parser)
  against the reference's final lnL -120844.546570 and the per-pass modOpt
  trace; our CPU replay matches every pass boundary digit-for-digit and
  the final to 2.9e-8 absolute
- the same flow on the MI355X Lg4Engine (GPU).
"""

import ctypes
import os

import numpy as np
import pytest

import examl_amd as ea
import oracle as O
from examl_amd.examl_io import read_byte_file, read_newick_topology
from examl_amd.search import TreeSearch

GOLDEN_FINAL_LNL = -120844.546570
# per-pass modOpt boundaries from the reference (-D_DEBUG_MOD_OPT build)
GOLDEN_PASS_1 = -120846.391174
GOLDEN_PASS_2 = -120844.938667
GOLDEN_START = -121048.404643  # after treeEvaluate(1)
TOL = abs(GOLDEN_FINAL_LNL) * 1e-6

LG4M, LG4X = 20, 21


def _engines(parts, lg4_cls, wag_cls):
    aa = np.load(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examl_amd", "data", "aa_models.npz"))
    engines = []
    for p in parts:
        if p.protModels == LG4M:
            engines.append(lg4_cls(p.tips, p.wgt, ea.Lg4Model.lg4m()))
        elif p.protModels == LG4X:
            engines.append(lg4_cls(p.tips, p.wgt, ea.Lg4Model.lg4x()))
        else:
            freqs = aa["frequencies"][p.protModels] if p.protFreqs == 0 \
                else p.frequencies
            engines.append(wag_cls(
                p.tips, p.wgt,
                ea.ProtGtrModel(freqs, aa["rates190"][p.protModels], 1.0)))
    return engines


def _load(golden_dir):
    taxa, parts = read_byte_file(os.path.join(golden_dir, "140lg4.binary"))
    tree = read_newick_topology(os.path.join(golden_dir, "140.tree"), taxa)
    return parts, tree


@pytest.mark.skipif(not O.have_ref(), reason="reference libref.so not built")
def test_lg4_kernels_bit_exact_vs_reference():
    rng = np.random.default_rng(11)
    d = np.load(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examl_amd", "data", "lg4_models.npz"))
    mdl = ea.Lg4Model(d["lg4x_frequencies"], d["lg4x_rates190"], 1.0)

    class A:  # 32-byte-aligned copies (the reference kernels use AVX loads)
        pass

    m = A()
    for name, size in (("EI4", 1600), ("EV4", 1600), ("tipVector4", 1840)):
        buf = O.aligned(size)
        buf[:] = getattr(mdl, name)
        setattr(m, name, buf)
    rates = O.aligned(4)
    rates[:] = [0.2, 0.7, 1.3, 2.5]
    weights = O.aligned(4)
    weights[:] = [0.1, 0.3, 0.4, 0.2]
    EIGN4 = O.aligned(80)
    EIGN4[:] = mdl.EIGN4_raw / float((weights[:4] * rates[:4]).sum())
    n = 64
    tips1 = rng.integers(1, 23, n).astype(np.uint8)
    tips2 = rng.integers(1, 23, n).astype(np.uint8)
    wgt = np.ones(n, dtype=np.int32)
    ref = O._ref

    def dp(a):
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_double))

    def u8(a):
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_ubyte))

    def ip(a):
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int))

    def pa(buf, stride):
        t = (ctypes.POINTER(ctypes.c_double) * 4)()
        for k in range(4):
            t[k] = ctypes.cast(buf.ctypes.data + 8 * stride * k,
                               ctypes.POINTER(ctypes.c_double))
        return t

    z1, z2 = -0.11, -0.31
    ol, orr = O.make_p_lg4(z1, z2, rates, m.EI4, EIGN4)
    rl = O.aligned(1600)
    rr = O.aligned(1600)
    ref.makeP_FlexLG4(ctypes.c_double(z1), ctypes.c_double(z2), dp(rates),
                      pa(m.EI4, 400), pa(EIGN4, 20), ctypes.c_int(4),
                      dp(rl), dp(rr), ctypes.c_int(20))
    assert np.array_equal(ol, rl) and np.array_equal(orr, rr)

    # TIP_TIP -> TIP_INNER -> INNER_INNER chain
    x3o, inco = O.newview_prot_lg4(0, None, None, m.EV4, m.tipVector4,
                                   tips1, tips2, n, ol, orr, wgt)
    x3r = O.aligned(n * 80)
    incr = ctypes.c_int(0)
    ref.newviewGTRGAMMAPROT_AVX_LG4(
        0, None, None, dp(x3r), pa(m.EV4, 400), pa(m.tipVector4, 460), None,
        u8(tips1), u8(tips2), ctypes.c_int(n), dp(rl), dp(rr), ip(wgt),
        ctypes.byref(incr), ctypes.c_int(1))
    assert np.array_equal(x3o, x3r) and inco == incr.value
    x3o2, _ = O.newview_prot_lg4(1, None, x3o, m.EV4, m.tipVector4, tips1,
                                 None, n, ol, orr, wgt)
    x3r2 = O.aligned(n * 80)
    incr2 = ctypes.c_int(0)
    ref.newviewGTRGAMMAPROT_AVX_LG4(
        1, None, dp(x3r), dp(x3r2), pa(m.EV4, 400), pa(m.tipVector4, 460),
        None, u8(tips1), None, ctypes.c_int(n), dp(rl), dp(rr), ip(wgt),
        ctypes.byref(incr2), ctypes.c_int(1))
    assert np.array_equal(x3o2, x3r2)
    x3o3, _ = O.newview_prot_lg4(2, x3o, x3o2, m.EV4, m.tipVector4, None,
                                 None, n, ol, orr, wgt)
    x3r3 = O.aligned(n * 80)
    incr3 = ctypes.c_int(0)
    ref.newviewGTRGAMMAPROT_AVX_LG4(
        2, dp(x3r), dp(x3r2), dp(x3r3), pa(m.EV4, 400),
        pa(m.tipVector4, 460), None, None, None, ctypes.c_int(n), dp(rl),
        dp(rr), ip(wgt), ctypes.byref(incr3), ctypes.c_int(1))
    assert np.array_equal(x3o3, x3r3)

    z = 0.83
    diag_o = O.calc_diagptable_lg4(z, rates, EIGN4)
    diag_r = O.aligned(80)
    ref.calcDiagptableFlex_LG4(ctypes.c_double(z), ctypes.c_int(4),
                               dp(rates), pa(EIGN4, 20), dp(diag_r),
                               ctypes.c_int(20))
    assert np.array_equal(diag_o, diag_r)
    lo = O.evaluate_prot_lg4(wgt, None, x3o3, m.tipVector4, tips1, n,
                             diag_o, weights)
    ref.evaluateGTRGAMMAPROT_LG4.restype = ctypes.c_double
    lr = ref.evaluateGTRGAMMAPROT_LG4(
        None, None, ip(wgt), None, dp(x3r3), pa(m.tipVector4, 460),
        u8(tips1), ctypes.c_int(n), dp(diag_r), ctypes.c_int(1),
        dp(weights))
    assert lo == lr

    sto = O.sum_prot_lg4(1, None, x3o3, m.tipVector4, tips1, None, n)
    str_ = O.aligned(n * 80)
    ref.sumGAMMAPROT_LG4(1, dp(str_), None, dp(x3r3), pa(m.tipVector4, 460),
                         u8(tips1), None, ctypes.c_int(n))
    assert np.array_equal(sto, str_)
    do1, do2 = O.core_prot_lg4(n, sto, EIGN4, rates, weights, -0.17, wgt)
    r1 = ctypes.c_double()
    r2 = ctypes.c_double()
    ref.coreGTRGAMMAPROT_LG4(dp(rates), pa(EIGN4, 20), dp(str_),
                             ctypes.c_int(n), ip(wgt), ctypes.byref(r1),
                             ctypes.byref(r2), ctypes.c_double(-0.17),
                             dp(weights))
    assert do1 == r1.value and do2 == r2.value


def test_lg4_f_E_two_passes_cpu_oracle(golden_dir):
    """Bounded LG4 coverage: treeEvaluate(1) + the first two modOpt passes
    land digit-for-digit on the reference's own trace (~100 s); the full
    pipeline is the opt-in test below (and the GPU test runs it whole)."""
    from tests.helpers import OracleEngine, OracleLg4Engine
    parts, tree = _load(golden_dir)
    ts = TreeSearch(tree, _engines(parts, OracleLg4Engine, OracleEngine),
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts])
    ts.evaluate_generic(full=True)
    start = ts.tree_evaluate(1.0)
    assert abs(start - GOLDEN_START) < 5e-6
    for expect in (GOLDEN_PASS_1, GOLDEN_PASS_2):
        ts.opt_rates_generic(0.0001)
        ts.evaluate_generic(full=True)
        ts.tree_evaluate(0.0625)
        ts.evaluate_generic(full=True)
        ts.opt_base_freqs(0.0001)
        ts.evaluate_generic(full=True)
        ts.tree_evaluate(0.0625)
        ts.opt_alphas_generic(0.0001)
        ts.evaluate_generic(full=True)
        ts.tree_evaluate(0.1)
        assert abs(ts.likelihood - expect) < 5e-6, (ts.likelihood, expect)


@pytest.mark.skipif(not os.environ.get("EXAML_E2E_LG4"),
                    reason="full LG4 -f E on CPU oracle (~4 min): set "
                           "EXAML_E2E_LG4=1")
def test_full_lg4_f_E_pipeline_cpu_oracle(golden_dir):
    from tests.helpers import OracleEngine, OracleLg4Engine
    parts, tree = _load(golden_dir)
    ts = TreeSearch(tree, _engines(parts, OracleLg4Engine, OracleEngine),
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts])
    lnl = ts.tree_evaluation_mode()
    assert abs(lnl - GOLDEN_FINAL_LNL) < TOL, lnl


@pytest.mark.gpu
def test_full_lg4_f_E_pipeline_gpu(golden_dir):
    """The whole LG4M+LG4X+WAG -f E flow on the MI355X engines."""
    import torch
    assert torch.cuda.is_available()
    parts, tree = _load(golden_dir)
    ts = TreeSearch(
        tree,
        _engines(parts,
                 lambda t, w, m: ea.Lg4Engine(t, w, m, device="cuda:0"),
                 lambda t, w, m: ea.DnaGammaEngine(t, w, m,
                                                   device="cuda:0")),
        opt_freq_flags=[bool(p.optimizeBaseFrequencies) for p in parts])
    lnl = ts.tree_evaluation_mode()
    assert abs(lnl - GOLDEN_FINAL_LNL) < TOL, lnl
