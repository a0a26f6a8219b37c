"""CPU-only parity: the oracle restatement vs the committed golden vectors
generated from the reference's own compiled kernels (oracle/gen_golden.py).

Everything here must be BIT-EXACT: the oracle deliberately mirrors the
reference's AVX/SSE3 summation order.
"""

import os

import numpy as np
import pytest

import oracle as O


@pytest.fixture(scope="module")
def model(golden_dir):
    return np.load(os.path.join(golden_dir, "model_dna.npz"))


@pytest.fixture(scope="module")
def kern(golden_dir):
    return np.load(os.path.join(golden_dir, "kernels_dna_gamma.npz"))


def _aligned_copy(a):
    out = O.aligned(a.shape, a.dtype)
    out[:] = a
    return out


def test_init_gtr_matches_reference_golden(model):
    for name in ("m0", "m1"):
        EIGN, EV, EI, tipVector = O.init_gtr_dna(model[f"{name}_freqs"],
                                                 model[f"{name}_rates6"])
        assert np.array_equal(EIGN, model[f"{name}_EIGN"])
        assert np.array_equal(EV, model[f"{name}_EV"])
        assert np.array_equal(EI, model[f"{name}_EI"])
        assert np.array_equal(tipVector, model[f"{name}_tipVector"])


def test_gamma_cats_match_reference_golden(model):
    for name in ("m0", "m1"):
        g = O.make_gamma_cats(float(model[f"{name}_alpha"]))
        assert np.array_equal(g, model[f"{name}_gammaRates"])


def test_make_p_and_diag_match_golden(model, kern):
    name = "m1"
    EI = _aligned_copy(model[f"{name}_EI"])
    EIGN = _aligned_copy(model[f"{name}_EIGN"])
    g = _aligned_copy(model[f"{name}_gammaRates"])
    left, right = O.make_p(np.log(float(kern["z_q"])),
                           np.log(float(kern["z_r"])), g, EI, EIGN, 4, 4)
    assert np.array_equal(left, kern["left"])
    assert np.array_equal(right, kern["right"])
    diag = O.calc_diagptable(float(kern["z_root"]), 4, 4, g, EIGN)
    assert np.array_equal(diag, kern["diag"])


@pytest.mark.parametrize("tag", ["norm", "tiny", "tiny2"])
@pytest.mark.parametrize("tc", [O.TIP_TIP, O.TIP_INNER, O.INNER_INNER])
def test_newview_matches_golden(model, kern, tag, tc):
    name = "m1"
    EV = _aligned_copy(model[f"{name}_EV"])
    tipVector = _aligned_copy(model[f"{name}_tipVector"])
    left = _aligned_copy(kern["left"])
    right = _aligned_copy(kern["right"])
    x1 = _aligned_copy(kern[f"{tag}_x1"])
    x2 = _aligned_copy(kern[f"{tag}_x2"])
    wgt = np.ascontiguousarray(kern[f"{tag}_wgt"])
    tipX1 = np.ascontiguousarray(kern[f"{tag}_tipX1"])
    tipX2 = np.ascontiguousarray(kern[f"{tag}_tipX2"])
    n = len(wgt)
    args = {
        O.TIP_TIP: (None, None, tipX1, tipX2),
        O.TIP_INNER: (None, x2, tipX1, None),
        O.INNER_INNER: (x1, x2, None, None),
    }[tc]
    x3, inc = O.newview_dna_gamma(tc, args[0], args[1], EV, tipVector,
                                  args[2], args[3], n, left, right, wgt)
    assert inc == int(kern[f"{tag}_newview_tc{tc}_inc"])
    assert np.array_equal(x3, kern[f"{tag}_newview_tc{tc}_x3"])


def test_evaluate_matches_golden(model, kern):
    name = "m1"
    tipVector = _aligned_copy(model[f"{name}_tipVector"])
    x1 = _aligned_copy(kern["norm_x1"])
    x2 = _aligned_copy(kern["norm_x2"])
    wgt = np.ascontiguousarray(kern["norm_wgt"])
    tipX1 = np.ascontiguousarray(kern["norm_tipX1"])
    diag = _aligned_copy(kern["diag"])
    n = len(wgt)
    lnl = O.evaluate_dna_gamma(wgt, x1, x2, tipVector, None, n, diag)
    assert lnl == float(kern["eval_II"])
    lnl = O.evaluate_dna_gamma(wgt, None, x2, tipVector, tipX1, n, diag)
    assert lnl == float(kern["eval_TIP"])


@pytest.mark.parametrize("tc", [O.TIP_TIP, O.TIP_INNER, O.INNER_INNER])
def test_sum_and_core_match_golden(model, kern, tc):
    name = "m1"
    tipVector = _aligned_copy(model[f"{name}_tipVector"])
    EIGN = _aligned_copy(model[f"{name}_EIGN"])
    g = _aligned_copy(model[f"{name}_gammaRates"])
    x1 = _aligned_copy(kern["norm_x1"])
    x2 = _aligned_copy(kern["norm_x2"])
    wgt = np.ascontiguousarray(kern["norm_wgt"])
    tipX1 = np.ascontiguousarray(kern["norm_tipX1"])
    tipX2 = np.ascontiguousarray(kern["norm_tipX2"])
    n = len(wgt)
    args = {
        O.TIP_TIP: (None, None, tipX1, tipX2),
        O.TIP_INNER: (None, x2, tipX1, None),
        O.INNER_INNER: (x1, x2, None, None),
    }[tc]
    st = O.sum_dna_gamma(tc, args[0], args[1], tipVector, args[2], args[3], n)
    assert np.array_equal(st, kern[f"sum_tc{tc}"])
    d1, d2 = O.core_dna_gamma(n, st, EIGN, g, float(kern["lz_core"]), wgt)
    assert d1 == float(kern[f"core_tc{tc}_d1"])
    assert d2 == float(kern[f"core_tc{tc}_d2"])


@pytest.mark.skipif(not O.have_ref(), reason="oracle/_ref not built here")
def test_live_cross_check_vs_reference():
    """Random-input bit-exactness against the reference kernels themselves
    (only runs in the dev container where /root/reference is present)."""
    rng = np.random.default_rng(7)
    freqs = rng.dirichlet([10, 10, 10, 10])
    rates6 = rng.uniform(0.3, 4.0, 6)
    rates6[5] = 1.0
    a = O.init_gtr_dna(freqs, rates6)
    b = O.ref_init_gtr_dna(freqs, rates6)
    for x, y in zip(a, b):
        assert np.array_equal(x, y)
    EIGN, EV, EI, tipVector = a
    g = O.make_gamma_cats(float(rng.uniform(0.1, 3.0)))
    l1, r1 = O.make_p(np.log(0.4), np.log(0.95), g, EI, EIGN, 4, 4)
    l2, r2 = O.ref_make_p(np.log(0.4), np.log(0.95), g, EI, EIGN, 4, 4)
    assert np.array_equal(l1, l2) and np.array_equal(r1, r2)
    n = 257  # odd size on purpose
    x1 = O.aligned(n * 16); x1[:] = rng.uniform(1e-50, 1.0, n * 16)
    x2 = O.aligned(n * 16); x2[:] = rng.uniform(1e-50, 1.0, n * 16)
    wgt = np.ascontiguousarray(rng.integers(1, 100, n), dtype=np.int32)
    o3, oi = O.newview_dna_gamma(O.INNER_INNER, x1, x2, EV, tipVector, None,
                                 None, n, l1, r1, wgt)
    r3, ri = O.newview_dna_gamma(O.INNER_INNER, x1, x2, EV, tipVector, None,
                                 None, n, l1, r1, wgt, lib=O._ref)
    assert oi == ri and np.array_equal(o3, r3)
