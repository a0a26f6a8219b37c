"""Multi-tree -f E / -f e (fast tree evaluation): optimizeTrees'
per-tree loop — tree 0 full modOpt, later trees resetBranches +
(fast: treeEvaluate(2) only | slow: full modOpt), model state carried
across trees.  Goldens from the reference on a 2-topology input
(49x2.trees: the 49 tree and a Seq1<->Seq40 label swap)."""

import os

import pytest

import examl_amd as ea
from examl_amd.examl_io import read_byte_file, read_newick_trees
from examl_amd.search import evaluate_trees

# reference examl-AVX -s 49 -t 49x2.trees
GOLDEN_SLOW = (-16205.671990, -16825.095741)  # -f E
GOLDEN_FAST = (-16205.671990, -16838.427884)  # -f e


def _setup(golden_dir, engine_cls):
    taxa, parts = read_byte_file(os.path.join(golden_dir, "49.binary"))
    trees = read_newick_trees(os.path.join(golden_dir, "49x2.trees"), taxa)
    assert len(trees) == 2
    engines = [engine_cls(p.tips, p.wgt,
                          ea.DnaGtrModel(p.frequencies, [1.0] * 6, 1.0))
               for p in parts]
    flags = [bool(p.optimizeBaseFrequencies) for p in parts]
    return trees, engines, flags


def test_fast_tree_evaluation_cpu(golden_dir):
    """-f e: the cheap second-tree path (tree 0 still runs full modOpt,
    ~3.5 min with the shared oracle engines is avoided by reusing the
    session fixture state? No — this runs its own tree-0 modOpt; the
    fast path only matters for tree 1)."""
    from tests.helpers import OracleEngine
    trees, engines, flags = _setup(golden_dir, OracleEngine)
    lnls = evaluate_trees(trees, engines, fast=True,
                          opt_freq_flags=flags)
    for got, want in zip(lnls, GOLDEN_FAST):
        assert abs(got - want) < abs(want) * 1e-6, (got, want)


@pytest.mark.gpu
def test_multi_tree_slow_gpu(golden_dir):
    """-f E over both topologies on the HIP engines."""
    import torch
    assert torch.cuda.is_available()
    trees, engines, flags = _setup(
        golden_dir, lambda t, w, m: ea.DnaGammaEngine(t, w, m,
                                                      device="cuda:0"))
    lnls = evaluate_trees(trees, engines, opt_freq_flags=flags)
    for got, want in zip(lnls, GOLDEN_SLOW):
        assert abs(got - want) < abs(want) * 1e-6, (got, want)
