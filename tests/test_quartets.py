"""Quartet evaluation (-f q) parity on testData/49: the RANDOM_QUARTETS
flavor with -r 30 -p 12345 against the reference's own quartet file
(tests/golden/49.quartets.txt, generated from oracle/_ref/examl-AVX on the
same inputs).  Both sides first optimize the model with the identical
treeEvaluate(1) + modOpt(0.1) preamble (quartets.c:408-417), so the PRNG
selection must match EXACTLY and each quartet-tree lnL to 1e-6 relative."""

import os
import re

import numpy as np
import pytest

from examl_amd.quartets import compute_quartets, randum

GOLDEN_RE = re.compile(r"^(\d+) (\d+) \| (\d+) (\d+): (-?\d+\.\d+)")


def _parse_golden(path):
    out = []
    for line in open(path):
        m = GOLDEN_RE.match(line)
        if m:
            out.append((int(m.group(1)), int(m.group(2)), int(m.group(3)),
                        int(m.group(4)), float(m.group(5))))
    return out


def test_randum_matches_reference_semantics():
    """axml.c:353 PRNG restatement: 12-bit limb arithmetic."""
    s = 12345
    vals = []
    for _ in range(4):
        v, s = randum(s)
        vals.append(v)
    # deterministic sequence (regression lock)
    assert abs(vals[0] - 0.7843347128946334) < 1e-15
    assert all(0.0 <= v < 1.0 for v in vals)


def test_random_quartets_cpu_vs_reference(golden_dir, optimized_49_cpu):
    golden = _parse_golden(os.path.join(golden_dir, "49.quartets.txt"))
    assert len(golden) == 90  # 30 quartets x 3 topologies
    ts, _ = optimized_49_cpu
    out = compute_quartets(ts.engines, ts.tree.ntips,
                           random_quartets=30, seed=12345)
    assert len(out) == len(golden)
    for (a, b, c, d, lnl), (ga, gb, gc, gd, glnl) in zip(out, golden):
        assert (a, b, c, d) == (ga, gb, gc, gd)  # PRNG selection identical
        assert abs(lnl - glnl) < max(abs(glnl) * 1e-6, 1e-4), \
            ((a, b, c, d), lnl, glnl)


@pytest.mark.gpu
def test_random_quartets_gpu(golden_dir):
    """The same quartet sweep on the HIP engines, against the same golden:
    exercises newview/evaluate/sum/core on tiny 4-taxon trees (the
    launch-latency stress case of SURVEY §8f row 4)."""
    import torch
    import examl_amd as ea
    from examl_amd.examl_io import read_byte_file, read_newick_topology
    from examl_amd.search import TreeSearch
    assert torch.cuda.is_available()
    golden = _parse_golden(os.path.join(golden_dir, "49.quartets.txt"))
    taxa, parts = read_byte_file(os.path.join(golden_dir, "49.binary"))
    tree = read_newick_topology(os.path.join(golden_dir, "49.tree"), taxa)
    engines = [ea.DnaGammaEngine(p.tips, p.wgt,
                                 ea.DnaGtrModel(p.frequencies, [1.0] * 6,
                                                1.0), device="cuda:0")
               for p in parts]
    ts = TreeSearch(
        tree, engines,
        opt_freq_flags=[bool(p.optimizeBaseFrequencies) for p in parts])
    ts.tree_evaluation_mode()
    out = compute_quartets(engines, tree.ntips, random_quartets=30,
                           seed=12345)
    assert len(out) == len(golden)
    for (a, b, c, d, lnl), (ga, gb, gc, gd, glnl) in zip(out, golden):
        assert (a, b, c, d) == (ga, gb, gc, gd)
        assert abs(lnl - glnl) < max(abs(glnl) * 1e-6, 1e-4), \
            ((a, b, c, d), lnl, glnl)


def test_grouped_quartets_match_reference_12(golden_dir):
    """GROUPED_QUARTETS (-Y, quartets.c:585-600): all 81 cross-group
    quartets x 3 topologies on the 12-taxon golden match the
    reference's ExaML_quartets output line for line (groups file:
    tests/golden/12.groups)."""
    import examl_amd as ea
    from examl_amd.examl_io import read_byte_file, read_newick_topology
    from examl_amd.search import TreeSearch
    from tests.helpers import OracleEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "12.binary"))
    tree = read_newick_topology(os.path.join(golden_dir, "12.tree"), taxa)
    engines = [OracleEngine(p.tips, p.wgt,
                            ea.DnaGtrModel(p.frequencies, [1.0] * 6, 1.0))
               for p in parts]
    ts = TreeSearch(tree, engines,
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts])
    ts.tree_evaluation_mode()
    out = compute_quartets(ts.engines, 12,
                           groups=[[1, 2, 3], [4, 5, 6], [7, 8, 9],
                                   [10, 11, 12]])
    golden = _parse_golden(os.path.join(golden_dir,
                                        "12.quartets.grouped.txt"))
    assert len(golden) == 243 and len(out) == 243
    for o, g in zip(out, golden):
        assert (o[0], o[1], o[2], o[3]) == (g[0], g[1], g[2], g[3])
        assert abs(o[4] - g[4]) < 1e-5, (o, g)


def test_quartet_resume_from_reference_checkpoint(golden_dir):
    """-f q -I / -R: resume RANDOM_QUARTETS from a reference-written
    QUARTETS-state checkpoint (counter 20 of 30, seed 123).  The PRNG
    stream replays from the stored initial seed with the first
    quartetCounter evaluations skipped (quartets.c:560), and the
    resumed output matches the reference's own resumed file tail line
    for line."""
    import examl_amd as ea
    from examl_amd.checkpoint import read_checkpoint
    from examl_amd.examl_io import read_byte_file
    from tests.helpers import OracleEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "12.binary"))
    ck = read_checkpoint(os.path.join(golden_dir, "12.quartets.ckpt.bin"),
                         12, [4])
    assert ck.state == 5 and ck.quartet_counter == 20 and ck.seed == 123
    engines = [OracleEngine(p.tips, p.wgt,
                            ea.DnaGtrModel(m["frequencies"],
                                           m["substRates"], m["alpha"]))
               for p, m in zip(parts, ck.models)]
    out = compute_quartets(engines, 12, random_quartets=30, seed=ck.seed,
                           start_counter=ck.quartet_counter)
    golden = _parse_golden(os.path.join(golden_dir,
                                        "12.quartets.random.txt"))
    assert len(golden) == 90 and len(out) == 30
    for o, g in zip(out, golden[60:]):
        assert (o[0], o[1], o[2], o[3]) == (g[0], g[1], g[2], g[3])
        assert abs(o[4] - g[4]) < 1e-5, (o, g)
