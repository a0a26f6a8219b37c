"""-a (median gamma discretization, makeGammaCats useMedian=TRUE,
models.c:3795): host math bit-exact vs the reference, and the 49 -f E -a
pipeline against the reference's own per-pass trace (full golden
-16180.242640)."""

import ctypes
import os

import numpy as np
import pytest

import examl_amd as ea
import oracle as O
from examl_amd.examl_io import read_byte_file, read_newick_topology
from examl_amd.search import TreeSearch

GOLDEN_FINAL = -16180.242640
GOLDEN_START = -18031.240945   # after treeEvaluate(1)
GOLDEN_PASS_1 = -16332.308564  # end of modOpt pass 1


@pytest.mark.skipif(not O.have_ref(), reason="reference libref.so not built")
def test_median_gamma_cats_bit_exact():
    L = ea.lib()
    for alpha in (0.05, 0.3, 1.0, 2.7, 11.0, 77.7):
        ours = np.zeros(4)
        L.examl_host_make_gamma_cats_median(
            ctypes.c_double(alpha), ours.ctypes.data_as(ctypes.c_void_p),
            ctypes.c_int(4))
        theirs = O.aligned(4)
        O._ref.makeGammaCats(
            ctypes.c_double(alpha),
            theirs.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
            ctypes.c_int(4), ctypes.c_int(1))
        assert np.array_equal(ours, theirs), alpha


def _search(golden_dir):
    from tests.helpers import OracleEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "49.binary"))
    tree = read_newick_topology(os.path.join(golden_dir, "49.tree"), taxa)
    engines = [OracleEngine(p.tips, p.wgt,
                            ea.DnaGtrModel(p.frequencies, [1.0] * 6, 1.0,
                                           use_median=True))
               for p in parts]
    return TreeSearch(
        tree, engines,
        opt_freq_flags=[bool(p.optimizeBaseFrequencies) for p in parts])


def test_median_f_E_one_pass_cpu_oracle(golden_dir):
    """Bounded: treeEvaluate(1) + modOpt pass 1 land on the reference's
    own -a trace; full pipeline is the opt-in test below."""
    ts = _search(golden_dir)
    ts.evaluate_generic(full=True)
    start = ts.tree_evaluate(1.0)
    assert abs(start - GOLDEN_START) < 5e-6
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
    assert abs(ts.likelihood - GOLDEN_PASS_1) < 5e-6, ts.likelihood


@pytest.mark.skipif(not os.environ.get("EXAML_E2E_MEDIAN"),
                    reason="full -a -f E on CPU oracle (~3 min): set "
                           "EXAML_E2E_MEDIAN=1")
def test_full_median_f_E_cpu_oracle(golden_dir):
    ts = _search(golden_dir)
    lnl = ts.tree_evaluation_mode()
    assert abs(lnl - GOLDEN_FINAL) < abs(GOLDEN_FINAL) * 1e-6, lnl
