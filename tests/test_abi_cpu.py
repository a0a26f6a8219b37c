"""CPU checks of the product library: the C-ABI .so loads, exports every
symbol include/examl_hip.h declares, and its HOST math (model prep, makeP,
diag, dtables) is bit-identical to the reference golden vectors."""

import ctypes
import os
import re

import numpy as np
import pytest

import examl_amd as ea

HDR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "include", "examl_hip.h")


def test_library_loads_and_reports_version():
    assert b"gfx950" in ea.lib().examl_hip_version()


def test_every_declared_symbol_is_exported():
    with open(HDR) as f:
        text = f.read()
    names = re.findall(r"\b(examl_(?:hip|host)_\w+)\s*\(", text)
    assert len(names) >= 14
    for name in set(names):
        assert hasattr(ea.lib(), name), f"missing export: {name}"


def test_product_model_prep_matches_golden(golden_dir):
    d = np.load(os.path.join(golden_dir, "model_dna.npz"))
    for name in ("m0", "m1"):
        m = ea.DnaGtrModel(d[f"{name}_freqs"], d[f"{name}_rates6"],
                           float(d[f"{name}_alpha"]))
        assert np.array_equal(m.EIGN, d[f"{name}_EIGN"])
        assert np.array_equal(m.EV, d[f"{name}_EV"])
        assert np.array_equal(m.EI, d[f"{name}_EI"])
        assert np.array_equal(m.tipVector, d[f"{name}_tipVector"])
        assert np.array_equal(m.gammaRates, d[f"{name}_gammaRates"])


def test_product_make_p_and_diag_match_golden(golden_dir):
    d = np.load(os.path.join(golden_dir, "model_dna.npz"))
    k = np.load(os.path.join(golden_dir, "kernels_dna_gamma.npz"))
    m = ea.DnaGtrModel(d["m1_freqs"], d["m1_rates6"], float(d["m1_alpha"]))
    left = np.zeros(64)
    right = np.zeros(64)
    L = ea.lib()

    def vp(a):
        return a.ctypes.data_as(ctypes.c_void_p)

    L.examl_host_make_p(
        ctypes.c_double(np.log(float(k["z_q"]))),
        ctypes.c_double(np.log(float(k["z_r"]))), vp(m.gammaRates), vp(m.EI),
        vp(m.EIGN), 4, vp(left), vp(right), 4)
    assert np.array_equal(left, k["left"])
    assert np.array_equal(right, k["right"])

    diag = np.zeros(16)
    L.examl_host_calc_diagptable(ctypes.c_double(float(k["z_root"])), 4, 4,
                                 vp(m.gammaRates), vp(m.EIGN), vp(diag))
    assert np.array_equal(diag, k["diag"])


def test_core_dtables_match_reference_formula(golden_dir):
    d = np.load(os.path.join(golden_dir, "model_dna.npz"))
    m = ea.DnaGtrModel(d["m1_freqs"], d["m1_rates6"], float(d["m1_alpha"]))
    out = np.zeros(48)
    lz = -0.7
    ea.lib().examl_host_core_dtables_dna(
        m.EIGN.ctypes.data_as(ctypes.c_void_p),
        m.gammaRates.ctypes.data_as(ctypes.c_void_p), ctypes.c_double(lz),
        out.ctypes.data_as(ctypes.c_void_p))
    for c in range(4):
        ki = m.gammaRates[c]
        assert out[c * 4] == 1.0 and out[16 + c * 4] == 0.0
        for l in range(1, 4):
            assert out[c * 4 + l] == np.exp(m.EIGN[l] * ki * lz)
            assert out[16 + c * 4 + l] == m.EIGN[l] * ki
            assert out[32 + c * 4 + l] == m.EIGN[l] ** 2 * ki ** 2


def test_engine_requires_gpu_no_silent_fallback():
    """The product path must fail loudly without a GPU, never fall back."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present")
    from tests.helpers import make_synthetic
    tips, wgt = make_synthetic(6, 64)
    m = ea.DnaGtrModel.jukes_cantor()
    with pytest.raises(RuntimeError):
        ea.DnaGammaEngine(tips, wgt, m, device="cuda")


def test_make_p_save_matches_oracle():
    """examl_host_make_p_save (saveMem rate-1.0 pair at slot maxCats,
    newviewGenericSpecial.c:140-165) is bit-identical to the oracle
    restatement for DNA and protein."""
    import ctypes as C
    import math

    import oracle as O
    rng = np.random.default_rng(3)
    for states, sq in ((4, 16), (20, 400)):
        if states == 4:
            m = ea.DnaGtrModel([0.3, 0.2, 0.26, 0.24],
                               [1.2, 2.4, 0.7, 0.9, 3.1, 1.0], 0.8)
        else:
            aa = np.load(os.path.join(os.path.dirname(os.path.dirname(
                os.path.abspath(__file__))), "examl_amd", "data",
                "aa_models.npz"))
            m = ea.ProtGtrModel(aa["frequencies"][4], aa["rates190"][4],
                                0.9)
        from tests.helpers import _model_arrays
        EIGN, EV, EI, tipVector, _ = _model_arrays(m)
        MAXC, nc = 25, 6
        rates = np.sort(rng.uniform(0.05, 4.0, nc))
        qz, rz = math.log(0.37), math.log(0.72)
        lr = np.zeros((MAXC + 1) * sq)
        rr = np.zeros((MAXC + 1) * sq)
        O._orc.oracle_make_p_save(
            C.c_double(qz), C.c_double(rz),
            rates.ctypes.data_as(C.POINTER(C.c_double)),
            EI.ctypes.data_as(C.POINTER(C.c_double)),
            EIGN.ctypes.data_as(C.POINTER(C.c_double)), C.c_int(nc),
            lr.ctypes.data_as(C.POINTER(C.c_double)),
            rr.ctypes.data_as(C.POINTER(C.c_double)), C.c_int(MAXC),
            C.c_int(states))
        lh = np.zeros((MAXC + 1) * sq)
        rh = np.zeros((MAXC + 1) * sq)
        ea.lib().examl_host_make_p_save(
            C.c_double(qz), C.c_double(rz),
            rates.ctypes.data_as(C.c_void_p),
            EI.ctypes.data_as(C.c_void_p),
            EIGN.ctypes.data_as(C.c_void_p), nc,
            lh.ctypes.data_as(C.c_void_p), rh.ctypes.data_as(C.c_void_p),
            MAXC, states)
        assert np.array_equal(lh, lr) and np.array_equal(rh, rr), states
