"""Protein CAT (-m PSR on AA partitions): span-20 per-site-rate kernels
(newviewGTRCATPROT_AVX, evaluateGTRCATPROT, sumGTRCATPROT, coreGTRCATPROT)
plus the protein evaluatePartial host probe.

Goldens: the reference's own kernels (bit-exact), and the full -f E -m PSR
pipeline on testData/140 re-partitioned as WAG+JTT (140psr.binary, built
with the reference parser) -> final lnL -121192.024397."""

import ctypes
import os

import numpy as np
import pytest

import examl_amd as ea
import oracle as O
from examl_amd.examl_io import read_byte_file, read_newick_topology
from examl_amd.search import TreeSearch

GOLDEN_FINAL_LNL = -121192.024397
TOL = abs(GOLDEN_FINAL_LNL) * 1e-6


@pytest.mark.skipif(not O.have_ref(), reason="reference libref.so not built")
def test_prot_cat_kernels_bit_exact_vs_reference():
    from tests.helpers import _model_arrays
    rng = np.random.default_rng(21)
    aa = np.load(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examl_amd", "data", "aa_models.npz"))
    m = ea.ProtGtrModel(aa["frequencies"][4], aa["rates190"][4], 1.0)
    EIGN, EV, EI, tipVector, _ = _model_arrays(m)
    n, nc = 150, 5
    rates = O.aligned(nc)
    rates[:] = [0.2, 0.6, 1.0, 1.7, 3.0]
    cptr = rng.integers(0, nc, n).astype(np.int32)
    t1 = rng.integers(1, 23, n).astype(np.uint8)
    t2 = rng.integers(1, 23, n).astype(np.uint8)
    wgt = np.ones(n, dtype=np.int32)
    left, right = O.make_p(-0.2, -0.45, rates, EI, EIGN, nc, 20)
    ref = O._ref

    def dp(a):
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_double))

    def u8(a):
        return (a.ctypes.data_as(ctypes.POINTER(ctypes.c_ubyte))
                if a is not None else
                ctypes.cast(None, ctypes.POINTER(ctypes.c_ubyte)))

    def ip(a):
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int))

    null = ctypes.cast(None, ctypes.POINTER(ctypes.c_double))

    def nv(tc, x1, x2, ta, tb):
        x3r = O.aligned(n * 20)
        ir = ctypes.c_int(0)
        ref.newviewGTRCATPROT_AVX(
            tc, dp(EV), ip(cptr), dp(x1) if x1 is not None else null,
            dp(x2) if x2 is not None else null, dp(x3r), dp(tipVector),
            u8(ta), u8(tb), ctypes.c_int(n), dp(left), dp(right), ip(wgt),
            ctypes.byref(ir))
        x3o, io = O.newview_prot_cat(tc, EV, cptr, x1, x2, tipVector, ta,
                                     tb, n, left, right, wgt)
        assert np.array_equal(x3r, x3o) and ir.value == io
        return x3o

    x3a = nv(0, None, None, t1, t2)
    x3b = nv(1, None, x3a, t1, None)
    x3c = nv(2, x3a, x3b, None, None)

    diag = O.calc_diagptable(0.6, 20, nc, rates, EIGN)
    ref.evaluateGTRCATPROT.restype = ctypes.c_double
    lr = ref.evaluateGTRCATPROT(ip(cptr), ip(wgt), dp(x3a), dp(x3b),
                                dp(tipVector), u8(None), ctypes.c_int(n),
                                dp(diag))
    lo = O.evaluate_prot_cat(cptr, wgt, x3a, x3b, tipVector, None, n, diag)
    assert lr == lo

    sr = O.aligned(n * 20)
    ref.sumGTRCATPROT(2, dp(sr), dp(x3a), dp(x3b), dp(tipVector), u8(None),
                      u8(None), ctypes.c_int(n))
    so = O.sum_prot_cat(2, x3a, x3b, tipVector, None, None, n)
    assert np.array_equal(sr, so)

    r1 = ctypes.c_double()
    r2 = ctypes.c_double()
    ref.coreGTRCATPROT(dp(EIGN), ctypes.c_double(-0.3), ctypes.c_int(nc),
                       dp(rates), ip(cptr), ctypes.c_int(n), ip(wgt),
                       ctypes.byref(r1), ctypes.byref(r2), dp(sr))
    o1, o2 = O.core_prot_cat(n, nc, so, wgt, rates, EIGN, cptr, -0.3)
    assert r1.value == o1 and r2.value == o2


@pytest.mark.skipif(not O.have_ref(), reason="reference libref.so not built")
def test_prot_evaluate_partial_bit_exact_vs_reference():
    """examl_host_evaluate_partial_prot_cat vs the reference's
    evaluatePartialGTRCATPROT (the CAT optimizer's per-site probe)."""
    from tests.helpers import make_synthetic_aa, OracleEngine, _model_arrays
    NUMB = 256

    class RefTI(ctypes.Structure):
        _fields_ = [("tipCase", ctypes.c_int), ("pNumber", ctypes.c_int),
                    ("qNumber", ctypes.c_int), ("rNumber", ctypes.c_int),
                    ("qz", ctypes.c_double * NUMB),
                    ("rz", ctypes.c_double * NUMB)]

    ntips, width = 10, 120
    tips, wgt = make_synthetic_aa(ntips, width, seed=77)
    aa = np.load(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examl_amd", "data", "aa_models.npz"))
    model = ea.ProtGtrModel(aa["frequencies"][4], aa["rates190"][4], 1.0)
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
    n_ti = len(entries) + 1
    arr = (RefTI * n_ti)()
    arr[0].pNumber, arr[0].qNumber, arr[0].qz[0] = p, q, root_z
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
    O._ref.evaluatePartialGTRCATPROT.restype = ctypes.c_double
    L = ea.lib()

    def dp(a):
        return a.ctypes.data_as(ctypes.POINTER(ctypes.c_double))

    ops_arr = (ea.TravEntry * len(entries))(*entries)
    for site in (0, 17, 60, 119):
        for ki in (0.2, 1.0, 2.7):
            r = O._ref.evaluatePartialGTRCATPROT(
                ctypes.c_int(site), ctypes.c_double(ki), ctypes.c_int(n_ti),
                arr, ctypes.c_double(root_z), ctypes.c_int(int(wgt[site])),
                dp(EIGN), dp(EI), dp(EV), dp(tipVector), rows,
                ctypes.c_int(0), ctypes.c_int(ntips))
            m = L.examl_host_evaluate_partial_prot_cat(
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


def _engines(parts, cls):
    aa = np.load(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examl_amd", "data", "aa_models.npz"))
    out = []
    for p in parts:
        freqs = aa["frequencies"][p.protModels] if p.protFreqs == 0 \
            else p.frequencies
        m = ea.ProtGtrModel(freqs, aa["rates190"][p.protModels], 1.0)
        w = p.upper - p.lower
        out.append(cls(p.tips, p.wgt, m, np.zeros(w, dtype=np.int32),
                       np.array([1.0])))
    return out


def _load(golden_dir):
    taxa, parts = read_byte_file(os.path.join(golden_dir, "140psr.binary"))
    tree = read_newick_topology(os.path.join(golden_dir, "140.tree"), taxa)
    return parts, tree


def test_full_prot_psr_f_E_cpu_oracle(golden_dir):
    """~30 s on the CPU oracle (CAT is span-20, far cheaper than GAMMA)."""
    from tests.helpers import OracleProtCatEngine
    parts, tree = _load(golden_dir)
    ts = TreeSearch(tree, _engines(parts, OracleProtCatEngine),
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts],
                    rate_het="CAT")
    lnl = ts.tree_evaluation_mode()
    assert abs(lnl - GOLDEN_FINAL_LNL) < TOL, lnl


@pytest.mark.gpu
def test_full_prot_psr_f_E_gpu(golden_dir):
    """The whole 140 WAG+JTT -m PSR -f E flow on the MI355X ProtCatEngine."""
    import torch
    assert torch.cuda.is_available()
    parts, tree = _load(golden_dir)
    ts = TreeSearch(
        tree,
        _engines(parts, lambda t, w, m, c, r: ea.ProtCatEngine(
            t, w, m, c, r, device="cuda:0")),
        opt_freq_flags=[bool(p.optimizeBaseFrequencies) for p in parts],
        rate_het="CAT")
    lnl = ts.tree_evaluation_mode()
    assert abs(lnl - GOLDEN_FINAL_LNL) < TOL, lnl
