"""Generate tests/golden/*.npz from the reference's own compiled kernels
(oracle/_ref/libref.so, built in place from /root/reference by
oracle/Makefile).  TEST INFRASTRUCTURE ONLY.

Run in the dev container (where /root/reference exists):
    python3 oracle/gen_golden.py
The fixtures are committed; the GPU box never needs /root/reference.
"""

import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import oracle as O  # noqa: E402

GOLDEN = os.path.join(os.path.dirname(__file__), "..", "tests", "golden")


def main():
    assert O.have_ref(), "oracle/_ref/libref.so missing — run make in oracle/"
    os.makedirs(GOLDEN, exist_ok=True)
    rng = np.random.default_rng(20260915)
    ref = O._ref

    # --- model inputs (two parameter sets: near-equal and skewed) ----------
    models = {
        "m0": dict(freqs=[0.25, 0.25, 0.25, 0.25], rates=[1, 1, 1, 1, 1, 1],
                   alpha=1.0),
        "m1": dict(freqs=[0.32, 0.18, 0.21, 0.29],
                   rates=[1.4, 3.2, 0.6, 0.9, 3.9, 1.0], alpha=0.47),
    }
    out = {}
    for name, m in models.items():
        freqs = np.array(m["freqs"], dtype=np.float64)
        rates6 = np.array(m["rates"], dtype=np.float64)
        EIGN, EV, EI, tipVector = O.ref_init_gtr_dna(freqs, rates6)
        g = O.ref_make_gamma_cats(m["alpha"])
        out[f"{name}_freqs"] = freqs
        out[f"{name}_rates6"] = rates6
        out[f"{name}_alpha"] = np.float64(m["alpha"])
        out[f"{name}_EIGN"] = EIGN
        out[f"{name}_EV"] = EV
        out[f"{name}_EI"] = EI
        out[f"{name}_tipVector"] = tipVector
        out[f"{name}_gammaRates"] = g
    np.savez_compressed(os.path.join(GOLDEN, "model_dna.npz"), **out)

    # --- kernel cases ------------------------------------------------------
    name = "m1"
    d = np.load(os.path.join(GOLDEN, "model_dna.npz"))
    EIGN = d[f"{name}_EIGN"].copy()
    EV = O.aligned(16); EV[:] = d[f"{name}_EV"]
    EI = O.aligned(16); EI[:] = d[f"{name}_EI"]
    tipVector = O.aligned(64); tipVector[:] = d[f"{name}_tipVector"]
    g = O.aligned(4); g[:] = d[f"{name}_gammaRates"]
    EIGNa = O.aligned(4); EIGNa[:] = EIGN

    z_q, z_r = 0.81, 0.13
    lzq, lzr = np.log(z_q), np.log(z_r)
    left, right = O.ref_make_p(lzq, lzr, g, EI, EIGNa, 4, 4)

    cases = {}
    n = 640
    for tag, scalemag in [("norm", 1.0), ("tiny", 1e-40), ("tiny2", 1e-80)]:
        x1 = O.aligned(n * 16)
        x1[:] = rng.uniform(0.01, 1.0, n * 16) * scalemag
        x2 = O.aligned(n * 16)
        x2[:] = rng.uniform(0.01, 1.0, n * 16) * scalemag
        wgt = np.ascontiguousarray(rng.integers(1, 5, n), dtype=np.int32)
        tipX1 = np.ascontiguousarray(rng.integers(1, 16, n), dtype=np.uint8)
        tipX2 = np.ascontiguousarray(rng.integers(1, 16, n), dtype=np.uint8)
        cases[f"{tag}_x1"] = x1
        cases[f"{tag}_x2"] = x2
        cases[f"{tag}_wgt"] = wgt
        cases[f"{tag}_tipX1"] = tipX1
        cases[f"{tag}_tipX2"] = tipX2
        for tc, a1, a2, t1, t2 in [
            (O.TIP_TIP, None, None, tipX1, tipX2),
            (O.TIP_INNER, None, x2, tipX1, None),
            (O.INNER_INNER, x1, x2, None, None),
        ]:
            x3, inc = O.newview_dna_gamma(tc, a1, a2, EV, tipVector, t1, t2,
                                          n, left, right, wgt, lib=ref)
            cases[f"{tag}_newview_tc{tc}_x3"] = x3
            cases[f"{tag}_newview_tc{tc}_inc"] = np.int64(inc)

    # evaluate / sum / core on the "norm" inputs
    x1 = cases["norm_x1"]; x2 = cases["norm_x2"]
    wgt = cases["norm_wgt"]; tipX1 = cases["norm_tipX1"]
    tipX2 = cases["norm_tipX2"]
    z_root = 0.77
    diag = O.ref_calc_diagptable(z_root, 4, 4, g, EIGNa)
    cases["z_root"] = np.float64(z_root)
    cases["eval_II"] = np.float64(
        O.evaluate_dna_gamma(wgt, x1, x2, tipVector, None, n, diag, lib=ref))
    cases["eval_TIP"] = np.float64(
        O.evaluate_dna_gamma(wgt, None, x2, tipVector, tipX1, n, diag,
                             lib=ref))
    lz = np.log(0.61)
    cases["lz_core"] = np.float64(lz)
    for tc, a1, a2, t1, t2 in [
        (O.TIP_TIP, None, None, tipX1, tipX2),
        (O.TIP_INNER, None, x2, tipX1, None),
        (O.INNER_INNER, x1, x2, None, None),
    ]:
        st = O.sum_dna_gamma(tc, a1, a2, tipVector, t1, t2, n, lib=ref)
        cases[f"sum_tc{tc}"] = st
        d1, d2 = O.core_dna_gamma(n, st, EIGNa, g, lz, wgt, lib=ref)
        cases[f"core_tc{tc}_d1"] = np.float64(d1)
        cases[f"core_tc{tc}_d2"] = np.float64(d2)

    cases["z_q"] = np.float64(z_q)
    cases["z_r"] = np.float64(z_r)
    cases["left"] = left
    cases["right"] = right
    cases["diag"] = diag
    cases["model"] = np.bytes_(name.encode())
    np.savez_compressed(os.path.join(GOLDEN, "kernels_dna_gamma.npz"),
                        **cases)

    # ---- protein (LG+GAMMA) ----------------------------------------------
    lg = np.load(os.path.join(os.path.dirname(__file__), "..", "examl_amd",
                              "data", "lg_model.npz"))
    alpha_aa = 0.8
    EIGN, EV, EI, tipVector = O.ref_init_gtr_aa(lg["frequencies"],
                                                lg["rates190"])
    g = O.ref_make_gamma_cats(alpha_aa)
    pc = {
        "alpha": np.float64(alpha_aa),
        "EIGN": EIGN, "EV": EV, "EI": EI, "tipVector": tipVector,
        "gammaRates": np.asarray(g),
    }
    z_q, z_r = 0.77, 0.21
    lzq, lzr = np.log(z_q), np.log(z_r)
    left, right = O.ref_make_p(lzq, lzr, g, EI, EIGN, 4, 20)
    pc["z_q"], pc["z_r"] = np.float64(z_q), np.float64(z_r)
    pc["left"], pc["right"] = left, right
    n = 320
    for tag, mag in [("norm", 1.0), ("tiny", 1e-80)]:
        x1 = O.aligned(n * 80); x1[:] = rng.uniform(0.01, 1.0, n * 80) * mag
        x2 = O.aligned(n * 80); x2[:] = rng.uniform(0.01, 1.0, n * 80) * mag
        wgt = np.ascontiguousarray(rng.integers(1, 5, n), dtype=np.int32)
        t1 = np.ascontiguousarray(rng.integers(1, 23, n), dtype=np.uint8)
        t2 = np.ascontiguousarray(rng.integers(1, 23, n), dtype=np.uint8)
        pc[f"{tag}_x1"], pc[f"{tag}_x2"] = x1, x2
        pc[f"{tag}_wgt"], pc[f"{tag}_tipX1"], pc[f"{tag}_tipX2"] = wgt, t1, t2
        for tc, a1, a2, u1, u2 in [
            (O.TIP_TIP, None, None, t1, t2),
            (O.TIP_INNER, None, x2, t1, None),
            (O.INNER_INNER, x1, x2, None, None),
        ]:
            x3, inc = O.newview_prot_gamma(tc, a1, a2, EV, tipVector, u1, u2,
                                           n, left, right, wgt, lib=ref)
            pc[f"{tag}_newview_tc{tc}_x3"] = x3
            pc[f"{tag}_newview_tc{tc}_inc"] = np.int64(inc)
    x1, x2 = pc["norm_x1"], pc["norm_x2"]
    wgt, t1, t2 = pc["norm_wgt"], pc["norm_tipX1"], pc["norm_tipX2"]
    z_root = 0.69
    diag = O.ref_calc_diagptable(z_root, 20, 4, g, EIGN)
    pc["z_root"], pc["diag"] = np.float64(z_root), diag
    pc["eval_II"] = np.float64(O.evaluate_prot_gamma(
        wgt, x1, x2, tipVector, None, n, diag, lib=ref))
    pc["eval_TIP"] = np.float64(O.evaluate_prot_gamma(
        wgt, None, x2, tipVector, t1, n, diag, lib=ref))
    lz = np.log(0.55)
    pc["lz_core"] = np.float64(lz)
    for tc, a1, a2, u1, u2 in [
        (O.TIP_TIP, None, None, t1, t2),
        (O.TIP_INNER, None, x2, t1, None),
        (O.INNER_INNER, x1, x2, None, None),
    ]:
        st = O.sum_prot_gamma(tc, a1, a2, tipVector, u1, u2, n, lib=ref)
        pc[f"sum_tc{tc}"] = st
        d1, d2 = O.core_prot_gamma(n, st, EIGN, g, lz, wgt, lib=ref)
        pc[f"core_tc{tc}_d1"] = np.float64(d1)
        pc[f"core_tc{tc}_d2"] = np.float64(d2)
    np.savez_compressed(os.path.join(GOLDEN, "kernels_prot_gamma.npz"), **pc)
    print("golden fixtures written to", GOLDEN)



def gen_cat():
    """CAT (PSR) golden fixtures — reference kernels via _ref."""
    assert O.have_ref()
    rng = np.random.default_rng(424242)
    ref = O._ref
    d = np.load(os.path.join(GOLDEN, "model_dna.npz"))
    EIGN = O.aligned(4); EIGN[:] = d["m1_EIGN"]
    EV = O.aligned(16); EV[:] = d["m1_EV"]
    EI = O.aligned(16); EI[:] = d["m1_EI"]
    tipVector = O.aligned(64); tipVector[:] = d["m1_tipVector"]
    num_cats = 9
    rptr = O.aligned(num_cats)
    rptr[:] = rng.uniform(0.03, 5.0, num_cats)
    z_q, z_r = 0.84, 0.18
    left, right = O.ref_make_p(np.log(z_q), np.log(z_r), rptr, EI, EIGN,
                               num_cats, 4)
    n = 768
    cc = {"num_cats": np.int64(num_cats), "rptr": np.asarray(rptr),
          "left": np.asarray(left), "right": np.asarray(right),
          "z_q": np.float64(z_q), "z_r": np.float64(z_r)}
    cptr = np.ascontiguousarray(rng.integers(0, num_cats, n), np.int32)
    cc["cptr"] = cptr
    for tag, mag in [("norm", 1.0), ("tiny", 1e-80)]:
        x1 = O.aligned(n * 4); x1[:] = rng.uniform(0.01, 1.0, n * 4) * mag
        x2 = O.aligned(n * 4); x2[:] = rng.uniform(0.01, 1.0, n * 4) * mag
        wgt = np.ascontiguousarray(rng.integers(1, 5, n), np.int32)
        t1 = np.ascontiguousarray(rng.integers(1, 16, n), np.uint8)
        t2 = np.ascontiguousarray(rng.integers(1, 16, n), np.uint8)
        cc[f"{tag}_x1"], cc[f"{tag}_x2"] = x1, x2
        cc[f"{tag}_wgt"], cc[f"{tag}_tipX1"], cc[f"{tag}_tipX2"] = wgt, t1, t2
        for tc, a1, a2, u1, u2 in [
            (O.TIP_TIP, None, None, t1, t2),
            (O.TIP_INNER, None, x2, t1, None),
            (O.INNER_INNER, x1, x2, None, None),
        ]:
            x3, inc = O.newview_dna_cat(tc, EV, cptr, a1, a2, tipVector, u1,
                                        u2, n, left, right, wgt, lib=ref)
            cc[f"{tag}_newview_tc{tc}_x3"] = x3
            cc[f"{tag}_newview_tc{tc}_inc"] = np.int64(inc)
    x1, x2 = cc["norm_x1"], cc["norm_x2"]
    wgt, t1, t2 = cc["norm_wgt"], cc["norm_tipX1"], cc["norm_tipX2"]
    z_root = 0.71
    diag = O.ref_calc_diagptable(z_root, 4, num_cats, rptr, EIGN)
    cc["z_root"], cc["diag"] = np.float64(z_root), np.asarray(diag)
    cc["eval_II"] = np.float64(O.evaluate_dna_cat(
        cptr, wgt, x1, x2, tipVector, None, n, diag, lib=ref))
    cc["eval_TIP"] = np.float64(O.evaluate_dna_cat(
        cptr, wgt, None, x2, tipVector, t1, n, diag, lib=ref))
    lz = np.log(0.52)
    cc["lz_core"] = np.float64(lz)
    for tc, a1, a2, u1, u2 in [
        (O.TIP_TIP, None, None, t1, t2),
        (O.TIP_INNER, None, x2, t1, None),
        (O.INNER_INNER, x1, x2, None, None),
    ]:
        st = O.sum_dna_cat(tc, a1, a2, tipVector, u1, u2, n, lib=ref)
        cc[f"sum_tc{tc}"] = np.asarray(st)
        d1, d2 = O.core_dna_cat(n, num_cats, st, wgt, rptr, EIGN, cptr, lz,
                                lib=ref)
        cc[f"core_tc{tc}_d1"] = np.float64(d1)
        cc[f"core_tc{tc}_d2"] = np.float64(d2)
    np.savez_compressed(os.path.join(GOLDEN, "kernels_dna_cat.npz"), **cc)
    print("CAT golden fixtures written")


if __name__ == "__main__":
    # regenerate the FULL golden kernel-fixture set (model_dna,
    # kernels_dna_gamma, kernels_prot_gamma, kernels_dna_cat); the
    # end-to-end fixtures (12*/49/140 binaries, trees, checkpoints) come
    # from tests/golden/gen_12.py + the reference binaries in oracle/_ref
    main()
    gen_cat()
