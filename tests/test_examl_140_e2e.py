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


@pytest.mark.gpu
@pytest.mark.timeout(1200)
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
