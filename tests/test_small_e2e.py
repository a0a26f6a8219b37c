"""Full -f E pipelines on the 12-taxon synthetic goldens (seconds each,
so the WHOLE pipeline — not a bounded first pass — runs in the default
CPU suite for every model family).  Goldens from the reference
examl-AVX on the same inputs; see tests/golden/gen_12.py for how the
fixtures were built with the reference's own parser."""

import os

import numpy as np
import pytest

import examl_amd as ea
from examl_amd.examl_io import read_byte_file, read_newick_topology
from examl_amd.search import TreeSearch

LG4M, LG4X = 20, 21


def _aa():
    return np.load(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examl_amd", "data", "aa_models.npz"))


def _dna_engines(parts, cat):
    from tests.helpers import OracleCatEngine, OracleEngine
    out = []
    for p in parts:
        m = ea.DnaGtrModel(p.frequencies, [1.0] * 6, alpha=1.0)
        if cat:
            w = p.upper - p.lower
            out.append(OracleCatEngine(p.tips, p.wgt, m,
                                       np.zeros(w, dtype=np.int32),
                                       np.array([1.0])))
        else:
            out.append(OracleEngine(p.tips, p.wgt, m))
    return out


def _prot_engines(parts):
    from tests.helpers import OracleEngine, OracleLg4Engine
    aa = _aa()
    out = []
    for p in parts:
        if p.protModels == LG4M:
            out.append(OracleLg4Engine(p.tips, p.wgt, ea.Lg4Model.lg4m()))
        elif p.protModels == LG4X:
            out.append(OracleLg4Engine(p.tips, p.wgt, ea.Lg4Model.lg4x()))
        else:
            freqs = aa["frequencies"][p.protModels] if p.protFreqs == 0 \
                else p.frequencies
            out.append(OracleEngine(
                p.tips, p.wgt,
                ea.ProtGtrModel(freqs, aa["rates190"][p.protModels], 1.0)))
    return out


def _run(golden_dir, binary, golden, cat=False, M=False, prot=False):
    taxa, parts = read_byte_file(os.path.join(golden_dir, binary))
    tree = read_newick_topology(os.path.join(golden_dir, "12.tree"), taxa)
    engines = _prot_engines(parts) if prot else _dna_engines(parts, cat)
    kw = dict(opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                              for p in parts])
    if cat:
        kw["rate_het"] = "CAT"
    if M:
        kw["per_gene_bl"] = True
    ts = TreeSearch(tree, engines, **kw)
    lnl = ts.tree_evaluation_mode()
    assert abs(lnl - golden) < abs(golden) * 1e-6, (lnl, golden)


def test_12_gamma_f_E(golden_dir):
    _run(golden_dir, "12.binary", -3650.993621)


def test_12_psr_f_E(golden_dir):
    _run(golden_dir, "12.binary", -3233.904617, cat=True)


def test_12_gamma_M_f_E(golden_dir):
    _run(golden_dir, "12m.binary", -3634.341296, M=True)


def test_12_psr_M_f_E(golden_dir):
    _run(golden_dir, "12m.binary", -3219.085042, cat=True, M=True)


def test_12_prot_wag_jtt_f_E(golden_dir):
    _run(golden_dir, "12aa.binary", -7246.699416, prot=True)


def test_12_lg4x_lg4m_f_E(golden_dir):
    _run(golden_dir, "12lg4.binary", -7387.472983, prot=True)


def test_12_psr_c10_f_E(golden_dir):
    """-c 10 (tr->maxCategories, axml.c:1117): the PSR pipeline under a
    non-default category cap lands on the reference's -3268.753243
    (vs -3233.904617 at the default 25 — the cap genuinely binds)."""
    from tests.helpers import OracleCatEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "12.binary"))
    tree = read_newick_topology(os.path.join(golden_dir, "12.tree"), taxa)
    engines = []
    for p in parts:
        m = ea.DnaGtrModel(p.frequencies, [1.0] * 6, 1.0)
        w = p.upper - p.lower
        engines.append(OracleCatEngine(p.tips, p.wgt, m,
                                       np.zeros(w, dtype=np.int32),
                                       np.array([1.0])))
    ts = TreeSearch(tree, engines,
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts],
                    rate_het="CAT", max_categories=10)
    lnl = ts.tree_evaluation_mode()
    assert abs(lnl - (-3268.753243)) < abs(3268.753243) * 1e-6, lnl
