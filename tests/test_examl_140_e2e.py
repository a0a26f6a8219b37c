"""End-to-end parity on testData/140 (BASELINE.json configs[4]): the
partitioned-protein -f E pipeline — WAG + two AUTO partitions, so this
exercises the AUTO protein-model selection (optimizeModel.c:2669) on top of
treeEvaluate/modOpt — against the reference's final lnL (-121288.814123,
measured from oracle/_ref/examl-AVX on the same inputs, 858 s on the dev
box's host CPU).  GPU-only: the CPU-oracle replay of 40 model trials would
take tens of minutes."""

import os

import numpy as np
import pytest

import examl_amd as ea
from examl_amd.examl_io import read_byte_file, read_newick_topology
from examl_amd.search import TreeSearch

GOLDEN_FINAL_LNL = -121288.814123
TOL_ABS = abs(GOLDEN_FINAL_LNL) * 1e-6  # the 1e-6-relative north-star bar


def _setup(golden_dir):
    taxa, parts = read_byte_file(os.path.join(golden_dir, "140.binary"))
    tree = read_newick_topology(os.path.join(golden_dir, "140.tree"), taxa)
    aa = np.load(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examl_amd", "data", "aa_models.npz"))
    return taxa, parts, tree, aa


@pytest.mark.gpu
def test_140_initial_evaluation_vs_oracle(golden_dir):
    """Bounded 140 coverage: the partitioned-protein initial full-tree
    evaluation on the HIP engines matches the CPU oracle replay, and one
    treeEvaluate pass improves it (runs in ~1 min; the full -f E pipeline
    with AUTO selection is the opt-in test below)."""
    import torch
    from tests.helpers import oracle_full_lnl
    assert torch.cuda.is_available()
    taxa, parts, tree, aa = _setup(golden_dir)
    engines = []
    models = []
    for p in parts:
        freqs = p.frequencies if (p.protModels == 19 and p.protFreqs == 0) \
            else aa["frequencies"][4 if p.protModels == 19 else p.protModels]
        m = ea.ProtGtrModel(freqs, aa["rates190"][4 if p.protModels == 19
                                                  else p.protModels], 1.0)
        models.append(m)
        engines.append(ea.DnaGammaEngine(p.tips, p.wgt, m, device="cuda:0"))
    ts = TreeSearch(tree, engines)
    fused = ts.fused
    ts.fused = None  # per-partition path first: 1e-11 vs the oracle
    lnl = ts.evaluate_generic(full=True)
    ref = 0.0
    entries, root = tree.full_traversal((1, next(iter(tree.adj[1]))))
    for p, m in zip(parts, models):
        ref += oracle_full_lnl(entries, root, tree, m, p.tips, p.wgt)
    assert abs(lnl - ref) / abs(ref) < 1e-11
    if fused is not None:
        # fused mseg path: device-exp P matrices -> <=1e-9 rel here
        ts.fused = fused
        lnl_f = ts.evaluate_generic(full=True)
        assert abs(lnl_f - ref) / abs(ref) < 1e-9, (lnl_f, ref)
    after = ts.tree_evaluate(1.0)
    assert after > lnl


@pytest.mark.gpu
@pytest.mark.skipif(not os.environ.get("EXAML_E2E_140"),
                    reason="Python-host-bound (>30 min even on the fused "
                           "executors: ~1e5 tiny optimizer probes through "
                           "Python/ctypes); the UNGATED hardware validation "
                           "of this config is tests/test_hybrid.py::"
                           "test_hybrid_140_partitioned_protein — the "
                           "reference's own C search + AUTO selection on "
                           "the same kernels in ~3 min. Set EXAML_E2E_140=1 "
                           "to run this Python replay too")
@pytest.mark.timeout(3000)
def test_full_f_E_pipeline_140_gpu(golden_dir):
    import torch
    assert torch.cuda.is_available()
    taxa, parts = read_byte_file(os.path.join(golden_dir, "140.binary"))
    tree = read_newick_topology(os.path.join(golden_dir, "140.tree"), taxa)
    aa = np.load(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examl_amd", "data", "aa_models.npz"))
    wag = 4  # protModels ids, globalVariables.h protModels[]
    AUTO = 19
    engines = []
    auto_flags = []
    empirical = []
    for p in parts:
        assert p.states == 20
        if p.protModels == AUTO:
            # AUTO starts as WAG (models.c:4222); protFreqs==0 -> empirical
            # frequencies initially (models.c:3528-3534)
            freqs = p.frequencies if p.protFreqs == 0 \
                else aa["frequencies"][wag]
            model = ea.ProtGtrModel(freqs, aa["rates190"][wag], alpha=1.0)
            auto_flags.append(True)
        else:
            # fixed matrix; protFreqs==0 -> the matrix's own frequencies
            # (models.c:3536-3551)
            freqs = aa["frequencies"][p.protModels] if p.protFreqs == 0 \
                else p.frequencies
            model = ea.ProtGtrModel(freqs, aa["rates190"][p.protModels],
                                    alpha=1.0)
            auto_flags.append(False)
        empirical.append(p.frequencies)
        engines.append(ea.DnaGammaEngine(p.tips, p.wgt, model,
                                         device="cuda:0"))
    ts = TreeSearch(tree, engines,
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts],
                    auto_flags=auto_flags, empirical_freqs=empirical)
    # initial prot_freqs per byte file
    ts.prot_freqs = [p.protFreqs for p in parts]
    lnl = ts.tree_evaluation_mode()
    assert abs(lnl - GOLDEN_FINAL_LNL) < TOL_ABS, lnl
