"""CPU-only parity for the protein (20-state) oracle vs the reference
golden vectors — bit-exact throughout (see test_oracle_cpu.py for DNA)."""

import os

import numpy as np
import pytest

import oracle as O


@pytest.fixture(scope="module")
def kern(golden_dir):
    return np.load(os.path.join(golden_dir, "kernels_prot_gamma.npz"))


def _al(a):
    out = O.aligned(a.shape, a.dtype)
    out[:] = a
    return out


@pytest.fixture(scope="module")
def lg():
    return np.load(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examl_amd", "data", "lg_model.npz"))


def test_init_gtr_aa_matches_golden(lg, kern):
    EIGN, EV, EI, tipVector = O.init_gtr_aa(lg["frequencies"],
                                            lg["rates190"])
    assert np.array_equal(EIGN, kern["EIGN"])
    assert np.array_equal(EV, kern["EV"])
    assert np.array_equal(EI, kern["EI"])
    assert np.array_equal(tipVector, kern["tipVector"])


def test_make_p_prot_matches_golden(kern):
    EIGN = _al(kern["EIGN"])
    EI = _al(kern["EI"])
    g = _al(kern["gammaRates"])
    left, right = O.make_p(np.log(float(kern["z_q"])),
                           np.log(float(kern["z_r"])), g, EI, EIGN, 4, 20)
    assert np.array_equal(left, kern["left"])
    assert np.array_equal(right, kern["right"])
    diag = O.calc_diagptable(float(kern["z_root"]), 20, 4, g, EIGN)
    assert np.array_equal(diag, kern["diag"])


@pytest.mark.parametrize("tag", ["norm", "tiny"])
@pytest.mark.parametrize("tc", [O.TIP_TIP, O.TIP_INNER, O.INNER_INNER])
def test_newview_prot_matches_golden(kern, tag, tc):
    EV = _al(kern["EV"])
    tipVector = _al(kern["tipVector"])
    left = _al(kern["left"])
    right = _al(kern["right"])
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
    x3, inc = O.newview_prot_gamma(tc, args[0], args[1], EV, tipVector,
                                   args[2], args[3], n, left, right, wgt)
    assert inc == int(kern[f"{tag}_newview_tc{tc}_inc"])
    assert np.array_equal(x3, kern[f"{tag}_newview_tc{tc}_x3"])


def test_evaluate_prot_matches_golden(kern):
    tipVector = _al(kern["tipVector"])
    x1 = _al(kern["norm_x1"])
    x2 = _al(kern["norm_x2"])
    wgt = np.ascontiguousarray(kern["norm_wgt"])
    t1 = np.ascontiguousarray(kern["norm_tipX1"])
    diag = _al(kern["diag"])
    n = len(wgt)
    assert O.evaluate_prot_gamma(wgt, x1, x2, tipVector, None, n,
                                 diag) == float(kern["eval_II"])
    assert O.evaluate_prot_gamma(wgt, None, x2, tipVector, t1, n,
                                 diag) == float(kern["eval_TIP"])


@pytest.mark.parametrize("tc", [O.TIP_TIP, O.TIP_INNER, O.INNER_INNER])
def test_sum_core_prot_match_golden(kern, tc):
    tipVector = _al(kern["tipVector"])
    EIGN = _al(kern["EIGN"])
    g = _al(kern["gammaRates"])
    x1 = _al(kern["norm_x1"])
    x2 = _al(kern["norm_x2"])
    wgt = np.ascontiguousarray(kern["norm_wgt"])
    t1 = np.ascontiguousarray(kern["norm_tipX1"])
    t2 = np.ascontiguousarray(kern["norm_tipX2"])
    n = len(wgt)
    args = {
        O.TIP_TIP: (None, None, t1, t2),
        O.TIP_INNER: (None, x2, t1, None),
        O.INNER_INNER: (x1, x2, None, None),
    }[tc]
    st = O.sum_prot_gamma(tc, args[0], args[1], tipVector, args[2], args[3],
                          n)
    assert np.array_equal(st, kern[f"sum_tc{tc}"])
    d1, d2 = O.core_prot_gamma(n, st, EIGN, g, float(kern["lz_core"]), wgt)
    assert d1 == float(kern[f"core_tc{tc}_d1"])
    assert d2 == float(kern[f"core_tc{tc}_d2"])


def test_product_aa_model_prep_matches_golden(lg, kern):
    import examl_amd as ea
    m = ea.ProtGtrModel(lg["frequencies"], lg["rates190"],
                        float(kern["alpha"]))
    assert np.array_equal(m.EIGN, kern["EIGN"])
    assert np.array_equal(m.EV, kern["EV"])
    assert np.array_equal(m.EI, kern["EI"])
    assert np.array_equal(m.tipVector, kern["tipVector"])
    assert np.array_equal(m.gammaRates, kern["gammaRates"])


def test_prot_host_logic_full_lnl():
    """Full protein pipeline through the oracle: rerooting invariance."""
    import examl_amd as ea
    from tests.helpers import make_synthetic_aa, oracle_full_lnl
    ntips, width = 10, 192
    tips, wgt = make_synthetic_aa(ntips, width, seed=17)
    model = ea.ProtGtrModel.lg(alpha=0.6)
    tree = ea.PhyloTree.random(ntips, seed=23, rng_z=True)
    edges = tree.edges()
    vals = []
    for edge in (edges[0], edges[-1]):
        entries, root = tree.full_traversal(edge)
        vals.append(oracle_full_lnl(entries, root, tree, model, tips, wgt))
    assert vals[0] < 0
    assert np.allclose(vals, vals[0], rtol=1e-9)
