"""GPU parity for the protein (20-state) kernels: bit-exact vs the
reference golden vectors and full-pipeline agreement with the CPU oracle."""

import ctypes
import math
import os

import numpy as np
import pytest

import examl_amd as ea
from tests.helpers import make_synthetic_aa, oracle_full_lnl, oracle_makenewz

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu


def vp(t):
    return ctypes.c_void_p(t.data_ptr())


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


@pytest.fixture(scope="module")
def kern(golden_dir):
    return np.load(os.path.join(golden_dir, "kernels_prot_gamma.npz"))


def _to_dev(a, dev):
    return torch.from_numpy(np.ascontiguousarray(a)).to(dev)


@pytest.mark.parametrize("tag", ["norm", "tiny"])
@pytest.mark.parametrize("tc", [ea.TIP_TIP, ea.TIP_INNER, ea.INNER_INNER])
def test_newview_prot_kernel_bit_exact(kern, dev, tag, tc):
    n = len(kern[f"{tag}_wgt"])
    d_x1 = _to_dev(kern[f"{tag}_x1"], dev)
    d_x2 = _to_dev(kern[f"{tag}_x2"], dev)
    d_x3 = torch.zeros(n * 80, dtype=torch.float64, device=dev)
    d_P = _to_dev(np.concatenate([kern["left"], kern["right"]]), dev)
    d_EV = _to_dev(kern["EV"], dev)
    d_tv = _to_dev(kern["tipVector"], dev)
    d_t1 = _to_dev(kern[f"{tag}_tipX1"], dev)
    d_t2 = _to_dev(kern[f"{tag}_tipX2"], dev)
    d_wgt = _to_dev(kern[f"{tag}_wgt"], dev)
    d_inc = torch.zeros(1, dtype=torch.int32, device=dev)
    null = ctypes.c_void_p(0)
    ea.check(ea.lib().examl_hip_newview_prot_gamma(
        tc,
        vp(d_x1) if tc == ea.INNER_INNER else null,
        vp(d_x2) if tc != ea.TIP_TIP else null,
        vp(d_x3), vp(d_EV), vp(d_tv),
        vp(d_t1) if tc != ea.INNER_INNER else null,
        vp(d_t2) if tc == ea.TIP_TIP else null,
        ctypes.c_long(n), vp(d_P),
        ctypes.c_void_p(d_P.data_ptr() + 1600 * 8), vp(d_wgt), vp(d_inc),
        ctypes.c_void_p(0)), "newview_prot")
    torch.cuda.synchronize()
    assert int(d_inc.item()) == int(kern[f"{tag}_newview_tc{tc}_inc"])
    assert np.array_equal(d_x3.cpu().numpy(),
                          kern[f"{tag}_newview_tc{tc}_x3"])


def test_evaluate_prot_kernel_vs_golden(kern, dev):
    n = len(kern["norm_wgt"])
    d_x1 = _to_dev(kern["norm_x1"], dev)
    d_x2 = _to_dev(kern["norm_x2"], dev)
    d_tv = _to_dev(kern["tipVector"], dev)
    d_t1 = _to_dev(kern["norm_tipX1"], dev)
    d_wgt = _to_dev(kern["norm_wgt"], dev)
    d_diag = _to_dev(kern["diag"], dev)
    d_lnl = torch.zeros(1, dtype=torch.float64, device=dev)
    d_part = torch.zeros(8192, dtype=torch.float64, device=dev)
    null = ctypes.c_void_p(0)
    ea.check(ea.lib().examl_hip_evaluate_prot_gamma(
        vp(d_wgt), vp(d_x1), vp(d_x2), vp(d_tv), null, ctypes.c_long(n),
        vp(d_diag), null, null, ctypes.c_double(0.0), vp(d_part), vp(d_lnl),
        ctypes.c_void_p(0)), "evaluate_prot")
    torch.cuda.synchronize()
    assert np.isclose(d_lnl.item(), float(kern["eval_II"]), rtol=1e-12)
    d_lnl.zero_()
    ea.check(ea.lib().examl_hip_evaluate_prot_gamma(
        vp(d_wgt), null, vp(d_x2), vp(d_tv), vp(d_t1), ctypes.c_long(n),
        vp(d_diag), null, null, ctypes.c_double(0.0), vp(d_part), vp(d_lnl),
        ctypes.c_void_p(0)), "evaluate_prot")
    torch.cuda.synchronize()
    assert np.isclose(d_lnl.item(), float(kern["eval_TIP"]), rtol=1e-12)


@pytest.mark.parametrize("tc", [ea.TIP_TIP, ea.TIP_INNER, ea.INNER_INNER])
def test_sum_core_prot_kernels(kern, dev, tc):
    n = len(kern["norm_wgt"])
    d_x1 = _to_dev(kern["norm_x1"], dev)
    d_x2 = _to_dev(kern["norm_x2"], dev)
    d_tv = _to_dev(kern["tipVector"], dev)
    d_t1 = _to_dev(kern["norm_tipX1"], dev)
    d_t2 = _to_dev(kern["norm_tipX2"], dev)
    d_sum = torch.zeros(n * 80, dtype=torch.float64, device=dev)
    null = ctypes.c_void_p(0)
    ea.check(ea.lib().examl_hip_sum_prot_gamma(
        tc, vp(d_sum),
        vp(d_x1) if tc == ea.INNER_INNER else null,
        vp(d_x2) if tc != ea.TIP_TIP else null,
        vp(d_tv),
        vp(d_t1) if tc != ea.INNER_INNER else null,
        vp(d_t2) if tc == ea.TIP_TIP else null,
        ctypes.c_long(n), ctypes.c_void_p(0)), "sum_prot")
    torch.cuda.synchronize()
    assert np.array_equal(d_sum.cpu().numpy(), kern[f"sum_tc{tc}"])

    d_wgt = _to_dev(kern["norm_wgt"], dev)
    d_dtab = torch.zeros(240, dtype=torch.float64, device=dev)
    d_out2 = torch.zeros(2, dtype=torch.float64, device=dev)
    d_part = torch.zeros(2 * 8192, dtype=torch.float64, device=dev)
    EIGN = np.ascontiguousarray(kern["EIGN"])
    g = np.ascontiguousarray(kern["gammaRates"])
    ea.check(ea.lib().examl_hip_core_root_prot_gamma(
        ctypes.c_long(n), vp(d_sum),
        EIGN.ctypes.data_as(ctypes.c_void_p),
        g.ctypes.data_as(ctypes.c_void_p),
        ctypes.c_double(float(kern["lz_core"])), vp(d_wgt), vp(d_dtab),
        vp(d_part), vp(d_out2), ctypes.c_void_p(0)), "core_prot")
    torch.cuda.synchronize()
    out = d_out2.cpu().numpy()
    assert np.isclose(out[0], float(kern[f"core_tc{tc}_d1"]), rtol=1e-11)
    assert np.isclose(out[1], float(kern[f"core_tc{tc}_d2"]), rtol=1e-11)


def test_prot_full_pipeline_vs_oracle(dev):
    ntips, width = 16, 2048
    tips, wgt = make_synthetic_aa(ntips, width, seed=321)
    model = ea.ProtGtrModel.lg(alpha=0.62)
    tree = ea.PhyloTree.random(ntips, seed=5, rng_z=True)
    eng = ea.DnaGammaEngine(tips, wgt, model, device=dev)
    entries, root = tree.full_traversal()
    lnl = eng.full_lnl(tree).item()
    ref, clv_ref, scalers_ref = oracle_full_lnl(entries, root, tree, model,
                                                tips, wgt, return_state=True)
    assert math.isfinite(lnl) and lnl < 0
    assert abs(lnl - ref) / abs(ref) < 1e-11
    clv = eng.d_clv.cpu().numpy()
    for slot, x in clv_ref.items():
        assert np.array_equal(clv[slot], x), f"prot CLV slot {slot} differs"
    sc = eng.d_scalers.cpu().numpy()
    for node in range(ntips + 1, 2 * ntips - 1):
        assert sc[node] == scalers_ref[node]


def test_prot_makenewz_vs_oracle(dev):
    ntips, width = 12, 1024
    tips, wgt = make_synthetic_aa(ntips, width, seed=77)
    model = ea.ProtGtrModel.lg(alpha=0.9)
    tree = ea.PhyloTree.random(ntips, seed=31, rng_z=True)
    eng = ea.DnaGammaEngine(tips, wgt, model, device=dev)
    entries, root = tree.full_traversal()
    p, q, z0 = root
    eng.newview_traversal(entries)
    z_gpu = eng.makenewz(tree, p, q, z0)
    z_ref = oracle_makenewz(entries, root, tree, model, tips, wgt, z0)
    assert abs(z_gpu - z_ref) < 1e-9


def test_prot_fast_math_variant(dev):
    """The opt-in FMA protein kernel (the reference's _FMA build class)
    agrees with the bit-exact path to fp64-roundoff accuracy."""
    ntips, width = 14, 4096
    tips, wgt = make_synthetic_aa(ntips, width, seed=9)
    model = ea.ProtGtrModel.lg(alpha=0.7)
    tree = ea.PhyloTree.random(ntips, seed=3, rng_z=True)
    eng = ea.DnaGammaEngine(tips, wgt, model, device=dev)
    exact = eng.full_lnl(tree).item()
    ea.lib().examl_hip_fast_math(1)
    try:
        fast = eng.full_lnl(tree).item()
    finally:
        ea.lib().examl_hip_fast_math(0)
    assert abs(fast - exact) / abs(exact) < 1e-12
