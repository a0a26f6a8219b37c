"""-S (saveMemory, SEV — Izquierdo-Carrasco et al.) DNA GTRGAMMA:
gap-bit-compacted CLVs + per-node gap columns.

- oracle SAVE kernels bit-exact vs the reference's own
  (newviewGTRGAMMA_AVX_GAPPED_SAVE, evaluateGTRGAMMA_GAPPED_SAVE,
  sumGAMMA_GAPPED_SAVE) over all tip cases on gappy data
- the SAVE engine is numerically TRANSPARENT: identical lnL / branch
  lengths to the dense engine (the reference's -S -f E reproduces the
  non-SAVE golden -16205.671990 exactly), while the CLV footprint shrinks
- GPU: SaveDnaEngine vs DnaGammaEngine on the same inputs.
"""

import ctypes
import os

import numpy as np
import pytest

import examl_amd as ea
import oracle as O
from examl_amd.search import TreeSearch


def _gappy(ntips, width, frac, seed):
    from tests.helpers import make_synthetic
    rng = np.random.default_rng(seed)
    tips, wgt = make_synthetic(ntips, width, seed=seed)
    for t in range(1, ntips + 1):
        tips[t][rng.random(width) < frac] = 15
    return tips, wgt


@pytest.mark.skipif(not O.have_ref(), reason="reference libref.so not built")
def test_save_kernels_bit_exact_vs_reference():
    from tests.helpers import _model_arrays
    rng = np.random.default_rng(3)
    m = ea.DnaGtrModel([0.3, 0.2, 0.26, 0.24],
                       [1.2, 2.4, 0.7, 0.9, 3.1, 1.0], 0.8)
    EIGN, EV, EI, tipVector, g = _model_arrays(m)
    n = 200
    t1 = rng.integers(1, 16, n).astype(np.uint8)
    t1[rng.random(n) < 0.3] = 15
    t2 = rng.integers(1, 16, n).astype(np.uint8)
    t2[rng.random(n) < 0.3] = 15
    wgt = np.ones(n, dtype=np.int32)
    gvl = n // 32 + 1

    def gap_of(tips):
        gv = np.zeros(gvl, dtype=np.uint32)
        idx = np.nonzero(tips == 15)[0]
        np.bitwise_or.at(gv, idx // 32,
                         (np.uint32(1) << (idx % 32).astype(np.uint32)))
        return gv

    g1, g2 = gap_of(t1), gap_of(t2)
    left, right = O.make_p(-0.2, -0.5, g, EI, EIGN, 4, 4)

    def dp(a):
        return (a.ctypes.data_as(ctypes.POINTER(ctypes.c_double))
                if a is not None else
                ctypes.cast(None, ctypes.POINTER(ctypes.c_double)))

    def u8(a):
        return (a.ctypes.data_as(ctypes.POINTER(ctypes.c_ubyte))
                if a is not None else
                ctypes.cast(None, ctypes.POINTER(ctypes.c_ubyte)))

    def ip(a):
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int))

    def up(a):
        return (a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint))
                if a is not None else
                ctypes.cast(None, ctypes.POINTER(ctypes.c_uint)))

    ref = O._ref
    tv_gap = O.aligned(4)
    tv_gap[:] = tipVector[15 * 4:16 * 4]

    def run_pair(tc, x1r, x1o, x2r, x2o, ga, gb, gca_r, gca_o, gcb_r, gcb_o,
                 ta, tb):
        g3 = ga & gb
        nz = int(n - sum(bin(int(w)).count("1") for w in g3))
        x3r = O.aligned(nz * 16 + 16)
        x3o = O.aligned(nz * 16 + 16)
        gr = O.aligned(16)
        go = O.aligned(16)
        ir = ctypes.c_int(0)
        io = ctypes.c_int(0)
        ref.newviewGTRGAMMA_AVX_GAPPED_SAVE(
            tc, dp(x1r), dp(x2r), dp(x3r), dp(EV), dp(tipVector), None,
            u8(ta), u8(tb), ctypes.c_int(n), dp(left), dp(right), ip(wgt),
            ctypes.byref(ir), ctypes.c_int(1), up(ga), up(gb), up(g3),
            dp(gca_r), dp(gcb_r), dp(gr))
        O._orc.oracle_newview_dna_gamma_save(
            tc, dp(x1o), dp(x2o), dp(x3o), dp(EV), dp(tipVector), u8(ta),
            u8(tb), ctypes.c_int(n), dp(left), dp(right), ip(wgt),
            ctypes.byref(io), up(ga), up(gb), up(g3), dp(gca_o), dp(gcb_o),
            dp(go))
        assert np.array_equal(x3r[:nz * 16], x3o[:nz * 16])
        assert np.array_equal(gr, go)
        assert ir.value == io.value
        return g3, x3r, x3o, gr, go

    g3, x3r, x3o, gcr, gco = run_pair(0, None, None, None, None, g1, g2,
                                      tv_gap, tv_gap, tv_gap, tv_gap, t1, t2)
    g3b, x3r2, x3o2, gcr2, gco2 = run_pair(1, None, None, x3r, x3o, g1, g3,
                                           tv_gap, tv_gap, gcr, gco, t1,
                                           None)
    g3c, x3r3, x3o3, gcr3, gco3 = run_pair(2, x3r, x3o, x3r2, x3o2, g3, g3b,
                                           gcr, gco, gcr2, gco2, None, None)

    diag = O.calc_diagptable(0.7, 4, 4, g, EIGN)
    ref.evaluateGTRGAMMA_GAPPED_SAVE.restype = ctypes.c_double
    O._orc.oracle_evaluate_dna_gamma_save.restype = ctypes.c_double
    lr = ref.evaluateGTRGAMMA_GAPPED_SAVE(
        ip(wgt), dp(x3r), dp(x3r2), dp(tipVector), None, ctypes.c_int(n),
        dp(diag), dp(gcr), dp(gcr2), up(g3), up(g3b))
    lo = O._orc.oracle_evaluate_dna_gamma_save(
        ip(wgt), dp(x3o), dp(x3o2), dp(tipVector), None, ctypes.c_int(n),
        dp(diag), dp(gco), dp(gco2), up(g3), up(g3b))
    assert lr == lo

    sr = O.aligned(n * 16)
    so = O.aligned(n * 16)
    ref.sumGAMMA_GAPPED_SAVE(2, dp(sr), dp(x3r), dp(x3r2), dp(tipVector),
                             None, None, ctypes.c_int(n), dp(gcr), dp(gcr2),
                             up(g3), up(g3b))
    O._orc.oracle_sum_dna_gamma_save(2, dp(so), dp(x3o), dp(x3o2),
                                     dp(tipVector), None, None,
                                     ctypes.c_int(n), dp(gco), dp(gco2),
                                     up(g3), up(g3b))
    assert np.array_equal(sr, so)


def test_save_engine_transparent_cpu():
    """The CPU SAVE engine produces IDENTICAL results to the dense engine
    through evaluate / makenewz / treeEvaluate on heavily gapped data,
    with a smaller CLV footprint."""
    from tests.helpers import OracleEngine, OracleSaveEngine
    tips, wgt = _gappy(16, 300, 0.35, 5)
    t1 = ea.PhyloTree.random(16, seed=9, rng_z=True)
    t2 = ea.PhyloTree.random(16, seed=9, rng_z=True)
    m = ea.DnaGtrModel([0.3, 0.2, 0.26, 0.24],
                       [1.2, 2.4, 0.7, 0.9, 3.1, 1.0], 0.8)
    e1 = OracleEngine(tips, wgt, m)
    e2 = OracleSaveEngine(tips, wgt, m)
    ts1 = TreeSearch(t1, [e1])
    ts2 = TreeSearch(t2, [e2])
    assert ts1.evaluate_generic(full=True) == ts2.evaluate_generic(full=True)
    p, q = 1, next(iter(t1.adj[1]))
    assert ts1.makenewz_generic(p, q, t1.get_z(p, q), 64) == \
        ts2.makenewz_generic(p, q, t2.get_z(p, q), 64)
    assert ts1.tree_evaluate(1.0) == ts2.tree_evaluate(1.0)
    for a, b in t1.edges():
        assert t1.get_z(a, b) == t2.get_z(a, b)
    dense = sum(v.nbytes for v in e1.clv.values())
    assert e2.clv_bytes() < dense  # the actual -S saving


def test_save_engine_on_49_cpu(golden_dir):
    """Bounded real-data check: initial evaluation + treeEvaluate on
    testData/49 with the SAVE engine equals the dense engine exactly
    (the reference's own -S -f E reproduces the non-SAVE golden)."""
    from examl_amd.examl_io import read_byte_file, read_newick_topology
    from tests.helpers import OracleEngine, OracleSaveEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "49.binary"))
    tree1 = read_newick_topology(os.path.join(golden_dir, "49.tree"), taxa)
    tree2 = read_newick_topology(os.path.join(golden_dir, "49.tree"), taxa)
    mk = lambda: [ea.DnaGtrModel(p.frequencies, [1.0] * 6, 1.0)
                  for p in parts]
    ts1 = TreeSearch(tree1, [OracleEngine(p.tips, p.wgt, m)
                             for p, m in zip(parts, mk())])
    ts2 = TreeSearch(tree2, [OracleSaveEngine(p.tips, p.wgt, m)
                             for p, m in zip(parts, mk())])
    assert ts1.evaluate_generic(full=True) == ts2.evaluate_generic(full=True)
    assert ts1.tree_evaluate(1.0) == ts2.tree_evaluate(1.0)


@pytest.mark.gpu
def test_save_engine_transparent_gpu():
    """SaveDnaEngine (HIP SAVE kernels, prefix-indexed compaction) equals
    the dense DnaGammaEngine on the same gappy inputs."""
    import torch
    assert torch.cuda.is_available()
    tips, wgt = _gappy(16, 5000, 0.35, 5)
    t1 = ea.PhyloTree.random(16, seed=9, rng_z=True)
    t2 = ea.PhyloTree.random(16, seed=9, rng_z=True)
    m = ea.DnaGtrModel([0.3, 0.2, 0.26, 0.24],
                       [1.2, 2.4, 0.7, 0.9, 3.1, 1.0], 0.8)
    e1 = ea.DnaGammaEngine(tips, wgt, m, device="cuda:0")
    e2 = ea.SaveDnaEngine(tips, wgt, m, device="cuda:0")
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
    dense = e1.d_clv.numel() * 8
    assert e2.clv_bytes() < dense


@pytest.mark.skipif(not O.have_ref(), reason="reference libref.so not built")
def test_prot_save_kernels_bit_exact_vs_reference():
    """Protein (span-80) GAPPED_SAVE kernels: newview/evaluate/sum vs the
    reference's newviewGTRGAMMAPROT_AVX_GAPPED_SAVE family on gappy AA
    data (undetermined code 22)."""
    from tests.helpers import _model_arrays
    rng = np.random.default_rng(13)
    aa = np.load(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examl_amd", "data", "aa_models.npz"))
    m = ea.ProtGtrModel(aa["frequencies"][4], aa["rates190"][4], 0.9)
    EIGN, EV, EI, tipVector, g = _model_arrays(m)
    n = 160
    t1 = rng.integers(1, 23, n).astype(np.uint8)
    t1[rng.random(n) < 0.3] = 22
    t2 = rng.integers(1, 23, n).astype(np.uint8)
    t2[rng.random(n) < 0.3] = 22
    wgt = np.ones(n, dtype=np.int32)
    gvl = n // 32 + 1

    def gap_of(tips):
        gv = np.zeros(gvl, dtype=np.uint32)
        idx = np.nonzero(tips == 22)[0]
        np.bitwise_or.at(gv, idx // 32,
                         (np.uint32(1) << (idx % 32).astype(np.uint32)))
        return gv

    g1, g2 = gap_of(t1), gap_of(t2)
    left, right = O.make_p(-0.2, -0.5, g, EI, EIGN, 4, 20)

    def dp(a):
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_double))

    def u8(a):
        return (a.ctypes.data_as(ctypes.POINTER(ctypes.c_ubyte))
                if a is not None else
                ctypes.cast(None, ctypes.POINTER(ctypes.c_ubyte)))

    def ip(a):
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int))

    def up(a):
        return (a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint))
                if a is not None else
                ctypes.cast(None, ctypes.POINTER(ctypes.c_uint)))

    ndp = ctypes.cast(None, ctypes.POINTER(ctypes.c_double))
    ref = O._ref
    tvg = O.aligned(20)
    tvg[:] = tipVector[22 * 20:23 * 20]

    def run(tc, x1r, x1o, x2r, x2o, ga, gb, gca_r, gca_o, gcb_r, gcb_o, ta,
            tb):
        g3 = ga & gb
        nz = int(n - sum(bin(int(w)).count("1") for w in g3))
        x3r = O.aligned(nz * 80 + 80)
        x3o = O.aligned(nz * 80 + 80)
        gr = O.aligned(80)
        go = O.aligned(80)
        ir = ctypes.c_int(0)
        io = ctypes.c_int(0)
        ref.newviewGTRGAMMAPROT_AVX_GAPPED_SAVE(
            tc, dp(x1r) if x1r is not None else ndp,
            dp(x2r) if x2r is not None else ndp, dp(x3r), dp(EV),
            dp(tipVector), None, u8(ta), u8(tb), ctypes.c_int(n), dp(left),
            dp(right), ip(wgt), ctypes.byref(ir), ctypes.c_int(1), up(ga),
            up(gb), up(g3), dp(gca_r), dp(gcb_r), dp(gr))
        O._orc.oracle_newview_prot_gamma_save(
            tc, dp(x1o) if x1o is not None else ndp,
            dp(x2o) if x2o is not None else ndp, dp(x3o), dp(EV),
            dp(tipVector), u8(ta), u8(tb), ctypes.c_int(n), dp(left),
            dp(right), ip(wgt), ctypes.byref(io), up(ga), up(gb), up(g3),
            dp(gca_o), dp(gcb_o), dp(go))
        assert np.array_equal(x3r[:nz * 80], x3o[:nz * 80])
        assert np.array_equal(gr, go) and ir.value == io.value
        return g3, x3r, x3o, gr, go

    g3, ar, ao, gr, go = run(0, None, None, None, None, g1, g2, tvg, tvg,
                             tvg, tvg, t1, t2)
    g3b, br, bo, gr2, go2 = run(1, None, None, ar, ao, g1, g3, tvg, tvg, gr,
                                go, t1, None)
    run(2, ar, ao, br, bo, g3, g3b, gr, go, gr2, go2, None, None)

    diag = O.calc_diagptable(0.7, 20, 4, g, EIGN)
    ref.evaluateGTRGAMMAPROT_GAPPED_SAVE.restype = ctypes.c_double
    O._orc.oracle_evaluate_prot_gamma_save.restype = ctypes.c_double
    lr = ref.evaluateGTRGAMMAPROT_GAPPED_SAVE(
        ip(wgt), dp(ar), dp(br), dp(tipVector), None, ctypes.c_int(n),
        dp(diag), dp(gr), dp(gr2), up(g3), up(g3b))
    lo = O._orc.oracle_evaluate_prot_gamma_save(
        ip(wgt), dp(ao), dp(bo), dp(tipVector), None, ctypes.c_int(n),
        dp(diag), dp(go), dp(go2), up(g3), up(g3b))
    assert lr == lo

    sr = O.aligned(n * 80)
    so = O.aligned(n * 80)
    ref.sumGAMMAPROT_GAPPED_SAVE(2, dp(sr), dp(ar), dp(br), dp(tipVector),
                                 None, None, ctypes.c_int(n), dp(gr),
                                 dp(gr2), up(g3), up(g3b))
    O._orc.oracle_sum_prot_gamma_save(2, dp(so), dp(ao), dp(bo),
                                      dp(tipVector), u8(None), u8(None),
                                      ctypes.c_int(n), dp(go), dp(go2),
                                      up(g3), up(g3b))
    assert np.array_equal(sr, so)


@pytest.mark.skipif(not O.have_ref(), reason="reference libref.so not built")
def test_cat_save_kernels_bit_exact_vs_reference():
    """DNA PSR (-S + CAT) GAPPED_SAVE kernels vs the reference's
    newviewGTRCAT_AVX_GAPPED_SAVE (avxLikelihood.c:2306),
    evaluateGTRCAT_SAVE (evaluateGenericSpecial.c:1537) and sumCAT_SAVE
    (makenewzGenericSpecial.c:1648), including the saveMem extra
    rate-1.0 P slot at maxCats (newviewGenericSpecial.c:78 makeP)."""
    from tests.helpers import _model_arrays
    rng = np.random.default_rng(9)
    m = ea.DnaGtrModel([0.3, 0.2, 0.26, 0.24],
                       [1.2, 2.4, 0.7, 0.9, 3.1, 1.0], 0.8)
    EIGN, EV, EI, tipVector, g = _model_arrays(m)
    n, nc, maxc = 200, 5, 7
    rates = O.aligned(nc)
    rates[:] = [0.2, 0.6, 1.0, 1.7, 3.0]
    cptr = rng.integers(0, nc, n).astype(np.int32)
    t1 = rng.integers(1, 16, n).astype(np.uint8)
    t1[rng.random(n) < 0.3] = 15
    t2 = rng.integers(1, 16, n).astype(np.uint8)
    t2[rng.random(n) < 0.3] = 15
    wgt = np.ones(n, dtype=np.int32)
    gvl = n // 32 + 1

    def gap_of(tips):
        gv = np.zeros(gvl, dtype=np.uint32)
        idx = np.nonzero(tips == 15)[0]
        np.bitwise_or.at(gv, idx // 32,
                         (np.uint32(1) << (idx % 32).astype(np.uint32)))
        return gv

    g1, g2 = gap_of(t1), gap_of(t2)

    def dp(a):
        return (a.ctypes.data_as(ctypes.POINTER(ctypes.c_double))
                if a is not None else
                ctypes.cast(None, ctypes.POINTER(ctypes.c_double)))

    def u8(a):
        return (a.ctypes.data_as(ctypes.POINTER(ctypes.c_ubyte))
                if a is not None else
                ctypes.cast(None, ctypes.POINTER(ctypes.c_ubyte)))

    def ip(a):
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int))

    def up(a):
        return (a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint))
                if a is not None else
                ctypes.cast(None, ctypes.POINTER(ctypes.c_uint)))

    ref = O._ref

    # makeP saveMem branch: extra P pair at rate 1.0 in slot maxCats
    left_o = O.aligned((maxc + 1) * 16)
    right_o = O.aligned((maxc + 1) * 16)
    O._orc.oracle_make_p_save(
        ctypes.c_double(-0.2), ctypes.c_double(-0.5), dp(rates), dp(EI),
        dp(EIGN), ctypes.c_int(nc), dp(left_o), dp(right_o),
        ctypes.c_int(maxc), ctypes.c_int(4))
    left_r = O.aligned((maxc + 1) * 16)
    right_r = O.aligned((maxc + 1) * 16)
    ref.makeP(ctypes.c_double(-0.2), ctypes.c_double(-0.5), dp(rates),
              dp(EI), dp(EIGN), ctypes.c_int(nc), dp(left_r), dp(right_r),
              ctypes.c_int(1), ctypes.c_int(maxc), ctypes.c_int(4))
    assert np.array_equal(left_o, left_r)
    assert np.array_equal(right_o, right_r)

    tv_gap = O.aligned(4)
    tv_gap[:] = tipVector[15 * 4:16 * 4]

    def run_pair(tc, x1r, x1o, x2r, x2o, ga, gb, gca_r, gca_o, gcb_r, gcb_o,
                 ta, tb):
        g3 = ga & gb
        nz = int(n - sum(bin(int(w)).count("1") for w in g3))
        x3r = O.aligned(nz * 4 + 4)
        x3o = O.aligned(nz * 4 + 4)
        gr = O.aligned(4)
        go = O.aligned(4)
        ir = ctypes.c_int(0)
        io = ctypes.c_int(0)
        ref.newviewGTRCAT_AVX_GAPPED_SAVE(
            tc, dp(EV), ip(cptr), dp(x1r), dp(x2r), dp(x3r), dp(tipVector),
            None, u8(ta), u8(tb), ctypes.c_int(n), dp(left_r), dp(right_r),
            ip(wgt), ctypes.byref(ir), ctypes.c_int(1), up(ga), up(gb),
            up(g3), dp(gca_r), dp(gcb_r), dp(gr), ctypes.c_int(maxc))
        O._orc.oracle_newview_dna_cat_save(
            tc, dp(EV), ip(cptr), dp(x1o), dp(x2o), dp(x3o), dp(tipVector),
            u8(ta), u8(tb), ctypes.c_int(n), dp(left_o), dp(right_o),
            ip(wgt), ctypes.byref(io), up(ga), up(gb), up(g3), dp(gca_o),
            dp(gcb_o), dp(go), ctypes.c_int(maxc))
        assert np.array_equal(x3r[:nz * 4], x3o[:nz * 4])
        assert np.array_equal(gr, go)
        assert ir.value == io.value
        return g3, x3r, x3o, gr, go

    g3, x3r, x3o, gcr, gco = run_pair(0, None, None, None, None, g1, g2,
                                      tv_gap, tv_gap, tv_gap, tv_gap, t1, t2)
    g3b, x3r2, x3o2, gcr2, gco2 = run_pair(1, None, None, x3r, x3o, g1, g3,
                                           tv_gap, tv_gap, gcr, gco, t1,
                                           None)
    run_pair(2, x3r, x3o, x3r2, x3o2, g3, g3b, gcr, gco, gcr2, gco2, None,
             None)

    diag = O.calc_diagptable(0.7, 4, nc, rates, EIGN)
    ref.evaluateGTRCAT_SAVE.restype = ctypes.c_double
    O._orc.oracle_evaluate_dna_cat_save.restype = ctypes.c_double
    lr = ref.evaluateGTRCAT_SAVE(
        ip(cptr), ip(wgt), dp(x3r), dp(x3r2), dp(tipVector), None,
        ctypes.c_int(n), dp(diag), dp(gcr), dp(gcr2), up(g3), up(g3b))
    lo = O._orc.oracle_evaluate_dna_cat_save(
        ip(cptr), ip(wgt), dp(x3o), dp(x3o2), dp(tipVector), u8(None),
        ctypes.c_int(n), dp(diag), dp(gco), dp(gco2), up(g3), up(g3b))
    assert lr == lo

    sr = O.aligned(n * 4)
    so = O.aligned(n * 4)
    ref.sumCAT_SAVE(2, dp(sr), dp(x3r), dp(x3r2), dp(tipVector), None, None,
                    ctypes.c_int(n), dp(gcr), dp(gcr2), up(g3), up(g3b))
    O._orc.oracle_sum_dna_cat_save(2, dp(so), dp(x3o), dp(x3o2),
                                   dp(tipVector), u8(None), u8(None),
                                   ctypes.c_int(n), dp(gco), dp(gco2),
                                   up(g3), up(g3b))
    assert np.array_equal(sr, so)


@pytest.mark.skipif(not O.have_ref(), reason="reference libref.so not built")
def test_prot_cat_save_kernels_bit_exact_vs_reference():
    """Protein PSR (-S + CAT, span 20) GAPPED_SAVE kernels vs the
    reference's newviewGTRCATPROT_AVX_GAPPED_SAVE (avxLikelihood.c:2607),
    evaluateGTRCATPROT_SAVE (evaluateGenericSpecial.c:1537) and
    sumGTRCATPROT_SAVE (makenewzGenericSpecial.c:2218), sharing the
    saveMem makeP rate-1.0 slot at maxCats."""
    from tests.helpers import _model_arrays
    rng = np.random.default_rng(17)
    aa = np.load(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examl_amd", "data", "aa_models.npz"))
    m = ea.ProtGtrModel(aa["frequencies"][4], aa["rates190"][4], 1.0)
    EIGN, EV, EI, tipVector, _ = _model_arrays(m)
    n, nc, maxc = 160, 5, 7
    rates = O.aligned(nc)
    rates[:] = [0.2, 0.6, 1.0, 1.7, 3.0]
    cptr = rng.integers(0, nc, n).astype(np.int32)
    t1 = rng.integers(1, 23, n).astype(np.uint8)
    t1[rng.random(n) < 0.3] = 22
    t2 = rng.integers(1, 23, n).astype(np.uint8)
    t2[rng.random(n) < 0.3] = 22
    wgt = np.ones(n, dtype=np.int32)
    gvl = n // 32 + 1

    def gap_of(tips):
        gv = np.zeros(gvl, dtype=np.uint32)
        idx = np.nonzero(tips == 22)[0]
        np.bitwise_or.at(gv, idx // 32,
                         (np.uint32(1) << (idx % 32).astype(np.uint32)))
        return gv

    g1, g2 = gap_of(t1), gap_of(t2)

    def dp(a):
        return (a.ctypes.data_as(ctypes.POINTER(ctypes.c_double))
                if a is not None else
                ctypes.cast(None, ctypes.POINTER(ctypes.c_double)))

    def u8(a):
        return (a.ctypes.data_as(ctypes.POINTER(ctypes.c_ubyte))
                if a is not None else
                ctypes.cast(None, ctypes.POINTER(ctypes.c_ubyte)))

    def ip(a):
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int))

    def up(a):
        return (a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint))
                if a is not None else
                ctypes.cast(None, ctypes.POINTER(ctypes.c_uint)))

    ref = O._ref
    left_o = O.aligned((maxc + 1) * 400)
    right_o = O.aligned((maxc + 1) * 400)
    O._orc.oracle_make_p_save(
        ctypes.c_double(-0.2), ctypes.c_double(-0.45), dp(rates), dp(EI),
        dp(EIGN), ctypes.c_int(nc), dp(left_o), dp(right_o),
        ctypes.c_int(maxc), ctypes.c_int(20))
    left_r = O.aligned((maxc + 1) * 400)
    right_r = O.aligned((maxc + 1) * 400)
    ref.makeP(ctypes.c_double(-0.2), ctypes.c_double(-0.45), dp(rates),
              dp(EI), dp(EIGN), ctypes.c_int(nc), dp(left_r), dp(right_r),
              ctypes.c_int(1), ctypes.c_int(maxc), ctypes.c_int(20))
    assert np.array_equal(left_o, left_r)
    assert np.array_equal(right_o, right_r)

    tvg = O.aligned(20)
    tvg[:] = tipVector[22 * 20:23 * 20]

    def run_pair(tc, x1r, x1o, x2r, x2o, ga, gb, gca_r, gca_o, gcb_r, gcb_o,
                 ta, tb):
        g3 = ga & gb
        nz = int(n - sum(bin(int(w)).count("1") for w in g3))
        x3r = O.aligned(nz * 20 + 20)
        x3o = O.aligned(nz * 20 + 20)
        gr = O.aligned(20)
        go = O.aligned(20)
        ir = ctypes.c_int(0)
        io = ctypes.c_int(0)
        ref.newviewGTRCATPROT_AVX_GAPPED_SAVE(
            tc, dp(EV), ip(cptr), dp(x1r), dp(x2r), dp(x3r), dp(tipVector),
            None, u8(ta), u8(tb), ctypes.c_int(n), dp(left_r), dp(right_r),
            ip(wgt), ctypes.byref(ir), ctypes.c_int(1), up(ga), up(gb),
            up(g3), dp(gca_r), dp(gcb_r), dp(gr), ctypes.c_int(maxc))
        O._orc.oracle_newview_prot_cat_save(
            tc, dp(EV), ip(cptr), dp(x1o), dp(x2o), dp(x3o), dp(tipVector),
            u8(ta), u8(tb), ctypes.c_int(n), dp(left_o), dp(right_o),
            ip(wgt), ctypes.byref(io), up(ga), up(gb), up(g3), dp(gca_o),
            dp(gcb_o), dp(go), ctypes.c_int(maxc))
        assert np.array_equal(x3r[:nz * 20], x3o[:nz * 20])
        assert np.array_equal(gr, go)
        assert ir.value == io.value
        return g3, x3r, x3o, gr, go

    g3, ar, ao, gr, go = run_pair(0, None, None, None, None, g1, g2, tvg,
                                  tvg, tvg, tvg, t1, t2)
    g3b, br, bo, gr2, go2 = run_pair(1, None, None, ar, ao, g1, g3, tvg,
                                     tvg, gr, go, t1, None)
    run_pair(2, ar, ao, br, bo, g3, g3b, gr, go, gr2, go2, None, None)

    diag = O.calc_diagptable(0.6, 20, nc, rates, EIGN)
    ref.evaluateGTRCATPROT_SAVE.restype = ctypes.c_double
    O._orc.oracle_evaluate_prot_cat_save.restype = ctypes.c_double
    lr = ref.evaluateGTRCATPROT_SAVE(
        ip(cptr), ip(wgt), dp(ar), dp(br), dp(tipVector), None,
        ctypes.c_int(n), dp(diag), dp(gr), dp(gr2), up(g3), up(g3b))
    lo = O._orc.oracle_evaluate_prot_cat_save(
        ip(cptr), ip(wgt), dp(ao), dp(bo), dp(tipVector), u8(None),
        ctypes.c_int(n), dp(diag), dp(go), dp(go2), up(g3), up(g3b))
    assert lr == lo
    lr2 = ref.evaluateGTRCATPROT_SAVE(
        ip(cptr), ip(wgt), dp(None), dp(ar), dp(tipVector), u8(t1),
        ctypes.c_int(n), dp(diag), dp(None), dp(gr), up(g1), up(g3))
    lo2 = O._orc.oracle_evaluate_prot_cat_save(
        ip(cptr), ip(wgt), dp(None), dp(ao), dp(tipVector), u8(t1),
        ctypes.c_int(n), dp(diag), dp(None), dp(go), up(g1), up(g3))
    assert lr2 == lo2

    sr = O.aligned(n * 20)
    so = O.aligned(n * 20)
    ref.sumGTRCATPROT_SAVE(2, dp(sr), dp(ar), dp(br), dp(tipVector),
                           u8(None), u8(None), ctypes.c_int(n), dp(gr),
                           dp(gr2), up(g3), up(g3b))
    O._orc.oracle_sum_prot_cat_save(2, dp(so), dp(ao), dp(bo), dp(tipVector),
                                    u8(None), u8(None), ctypes.c_int(n),
                                    dp(go), dp(go2), up(g3), up(g3b))
    assert np.array_equal(sr, so)
    sr2 = O.aligned(n * 20)
    so2 = O.aligned(n * 20)
    ref.sumGTRCATPROT_SAVE(1, dp(sr2), dp(None), dp(ar), dp(tipVector),
                           u8(t1), u8(None), ctypes.c_int(n), dp(None),
                           dp(gr), up(g1), up(g3))
    O._orc.oracle_sum_prot_cat_save(1, dp(so2), dp(None), dp(ao),
                                    dp(tipVector), u8(t1), u8(None),
                                    ctypes.c_int(n), dp(None), dp(go),
                                    up(g1), up(g3))
    assert np.array_equal(sr2, so2)


def test_save_engine_transparent_with_per_gene_bl():
    """-S x -M interplay (uncovered combination): SaveDnaEngine under
    per-partition branch lengths equals the dense oracle engine on the
    same gappy 2-partition data — newview entries carry per-partition
    qz/rz (axml.h:434 qz[i]) and evaluate/makenewz use per-partition
    roots."""
    from examl_amd.search import TreeSearch
    from tests.helpers import OracleEngine, OracleSaveEngine
    tips1, wgt1 = _gappy(10, 240, 0.4, 11)
    tips2, wgt2 = _gappy(10, 180, 0.3, 12)
    mk = lambda: [ea.DnaGtrModel([0.3, 0.2, 0.26, 0.24],
                                 [1.2, 2.4, 0.7, 0.9, 3.1, 1.0], 0.8),
                  ea.DnaGtrModel([0.22, 0.28, 0.24, 0.26],
                                 [0.8, 1.4, 1.7, 0.9, 2.1, 1.0], 1.3)]
    t1 = ea.PhyloTree.random(10, seed=21, rng_z=True)
    t2 = ea.PhyloTree.random(10, seed=21, rng_z=True)
    m1, m2 = mk(), mk()
    ts_dense = TreeSearch(
        t1, [OracleEngine(tips1, wgt1, m1[0]),
             OracleEngine(tips2, wgt2, m1[1])], per_gene_bl=True)
    ts_save = TreeSearch(
        t2, [OracleSaveEngine(tips1, wgt1, m2[0]),
             OracleSaveEngine(tips2, wgt2, m2[1])], per_gene_bl=True)
    l1 = ts_dense.evaluate_generic(full=True)
    l2 = ts_save.evaluate_generic(full=True)
    assert l1 == l2, (l1, l2)
    a1 = ts_dense.tree_evaluate(1.0)
    a2 = ts_save.tree_evaluate(1.0)
    assert a1 == a2, (a1, a2)
