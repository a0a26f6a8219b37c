"""End-to-end parity for the CAT (-m PSR) model on testData/49: the full
-f E pipeline — evaluate + treeEvaluate + modOpt with the catOpt<3
optimizeRateCategories schedule (optimizeModel.c:3096-3110) — must land on
the reference's final lnL (-14702.970620, measured from
oracle/_ref/examl-AVX -m PSR -f E on the same inputs; our CPU replay lands
at 3e-11 relative).

This exercises the whole §8f CAT row: per-site rate search over
evaluatePartialGeneric, rate clustering, mean-1 rescale, and the CAT
kernels (newview/evaluate/sum/core) under the optimizer.
"""

import os

import numpy as np
import pytest

import examl_amd as ea
from examl_amd.examl_io import read_byte_file, read_newick_topology
from examl_amd.search import TreeSearch

GOLDEN_FINAL_LNL = -14702.970620  # reference examl-AVX -m PSR -f E
TOL_ABS = abs(GOLDEN_FINAL_LNL) * 1e-6


def _engines(parts, engine_cls):
    engines = []
    for p in parts:
        model = ea.DnaGtrModel(p.frequencies, [1.0] * 6, alpha=1.0)
        w = p.upper - p.lower
        # CAT initial state: one category, rate 1 (models.c:4194-4201)
        engines.append(engine_cls(p.tips, p.wgt, model,
                                  np.zeros(w, dtype=np.int32),
                                  np.array([1.0])))
    return engines


def _load(golden_dir):
    taxa, parts = read_byte_file(os.path.join(golden_dir, "49.binary"))
    tree = read_newick_topology(os.path.join(golden_dir, "49.tree"), taxa)
    return parts, tree


def test_full_psr_f_E_pipeline_cpu_oracle(golden_dir):
    """-f E -m PSR on the CPU oracle engines (about 2.5 min)."""
    from tests.helpers import OracleCatEngine
    parts, tree = _load(golden_dir)
    ts = TreeSearch(tree, _engines(parts, OracleCatEngine),
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts],
                    rate_het="CAT")
    lnl = ts.tree_evaluation_mode()
    assert abs(lnl - GOLDEN_FINAL_LNL) < TOL_ABS, lnl
    assert all(e.num_cats == 25 for e in ts.engines)
    # GLOBAL weighted mean rate == 1 for numBranches==1
    # (checkPerSiteRates, optimizeModel.c:2022-2036; the per-partition
    # means only hold under -M)
    rsum = wsum = 0.0
    for e in ts.engines:
        rsum += float((e.host_wgt * e.per_site_rates[e.cptr]).sum())
        wsum += float(e.host_wgt.sum())
    assert abs(rsum / wsum - 1.0) < 1e-5


@pytest.mark.gpu
def test_full_psr_f_E_pipeline_gpu(golden_dir):
    """The same flow on the MI355X CAT engines (HIP span-4 kernels +
    the host evaluatePartial probe)."""
    import torch
    assert torch.cuda.is_available()
    parts, tree = _load(golden_dir)
    ts = TreeSearch(
        tree,
        _engines(parts, lambda tips, wgt, model, cptr, rates:
                 ea.DnaCatEngine(tips, wgt, model, cptr, rates,
                                 device="cuda:0")),
        opt_freq_flags=[bool(p.optimizeBaseFrequencies) for p in parts],
        rate_het="CAT")
    lnl = ts.tree_evaluation_mode()
    assert abs(lnl - GOLDEN_FINAL_LNL) < TOL_ABS, lnl


def test_full_psr_M_f_E_pipeline_cpu_oracle(golden_dir):
    """-m PSR with -M (per-partition branch lengths): exercises the
    vectorized NR under CAT and updatePerSiteRates' per-partition
    rescale branch (optimizeModel.c:2072-2082).  Reference golden
    -14532.602263; our replay lands at 1.5e-11 relative (~70 s)."""
    from tests.helpers import OracleCatEngine
    parts, tree = _load(golden_dir)
    ts = TreeSearch(tree, _engines(parts, OracleCatEngine),
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts],
                    rate_het="CAT", per_gene_bl=True)
    lnl = ts.tree_evaluation_mode()
    golden = -14532.602263
    assert abs(lnl - golden) < abs(golden) * 1e-6, lnl
    # per-partition weighted mean rate == 1 under -M (checkPerSiteRates)
    for e in ts.engines:
        mean = float((e.host_wgt * e.per_site_rates[e.cptr]).sum()
                     / e.host_wgt.sum())
        assert abs(mean - 1.0) < 1e-5


@pytest.mark.gpu
def test_full_psr_M_f_E_pipeline_gpu(golden_dir):
    import torch
    assert torch.cuda.is_available()
    parts, tree = _load(golden_dir)
    ts = TreeSearch(
        tree,
        _engines(parts, lambda tips, wgt, model, cptr, rates:
                 ea.DnaCatEngine(tips, wgt, model, cptr, rates,
                                 device="cuda:0")),
        opt_freq_flags=[bool(p.optimizeBaseFrequencies) for p in parts],
        rate_het="CAT", per_gene_bl=True)
    lnl = ts.tree_evaluation_mode()
    golden = -14532.602263
    assert abs(lnl - golden) < abs(golden) * 1e-6, lnl
