"""Binary checkpoint interchange with the reference (-R restart):

- READ: parse a reference-written checkpoint (tests/golden/49.ckpt2.bin =
  the third MOD_OPT checkpoint of examl-AVX -f E on testData/49), rebuild
  engines from the stored model arrays and the tree from the node image,
  land BIT-EXACTLY on the lnL the reference prints at its own restart
  (-16309.42131282399714), and resume modOpt to the -f E golden.
- WRITE: emit a checkpoint from our own state that our reader round-trips
  and that the reference binary itself accepts and resumes to the same
  final lnL (exercised when oracle/_ref/examl-AVX is present)."""

import os
import shutil
import subprocess

import numpy as np
import pytest

import examl_amd as ea
from examl_amd.checkpoint import (read_checkpoint, write_checkpoint,
                                  build_model_entry)
from examl_amd.examl_io import read_byte_file, read_newick_topology
from examl_amd.search import TreeSearch

RESTART_LNL = -16309.42131282399714  # reference: "ExaML Restart with ..."
GOLDEN_FINAL_LNL = -16205.671990
TOL_ABS = abs(GOLDEN_FINAL_LNL) * 1e-6

_REF_BIN = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "oracle", "_ref", "examl-AVX")


def test_read_reference_checkpoint_and_resume(golden_dir):
    from tests.helpers import OracleEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "49.binary"))
    ck = read_checkpoint(os.path.join(golden_dir, "49.ckpt2.bin"), 49,
                         [4] * 4)
    assert ck.state == 4 and not ck.per_gene_bl
    assert len(ck.tree.edges()) == 2 * 49 - 3
    engines = []
    for p, m in zip(parts, ck.models):
        model = ea.DnaGtrModel(m["frequencies"], m["substRates"], m["alpha"])
        # our host model math must reproduce the reference's stored
        # eigendecomposition bit-for-bit
        assert np.array_equal(model.EIGN[:4], m["EIGN"])
        assert np.array_equal(model.EV, m["EV"])
        assert np.array_equal(model.tipVector, m["tipVector"])
        assert np.array_equal(model.gammaRates, m["gammaRates"])
        engines.append(OracleEngine(p.tips, p.wgt, model))
    ts = TreeSearch(ck.tree, engines,
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts])
    lnl = ts.evaluate_generic(full=True)
    assert lnl == RESTART_LNL  # bit-exact restore
    fin = ts.mod_opt(0.1)
    assert abs(fin - GOLDEN_FINAL_LNL) < TOL_ABS, fin


def test_checkpoint_roundtrip(tmp_path):
    t = ea.PhyloTree.random(10, seed=5, rng_z=True)
    m = ea.DnaGtrModel([0.3, 0.2, 0.26, 0.24],
                       [1.2, 2.4, 0.7, 0.9, 3.1, 1.0], 0.7)
    path = str(tmp_path / "rt.ckpt")
    write_checkpoint(path, t, [build_model_entry(m)], 10)
    ck = read_checkpoint(path, 10, [4])
    assert sorted(ck.tree.edges()) == sorted(t.edges())
    for a, b in t.edges():
        assert ck.tree.get_z(a, b) == t.get_z(a, b)
    assert ck.models[0]["alpha"] == 0.7
    assert np.array_equal(ck.models[0]["substRates"], m.rates6)
    assert np.array_equal(ck.models[0]["EV"], m.EV)
    assert np.array_equal(ck.models[0]["EIGN"], m.EIGN[:4])


@pytest.mark.skipif(not os.path.exists(_REF_BIN),
                    reason="reference examl-AVX not built")
def test_reference_resumes_from_our_checkpoint(golden_dir, tmp_path):
    """The strongest drop-in claim: the unmodified reference binary
    restarts from a checkpoint WE wrote (state = after treeEvaluate(1))
    and reaches its own -f E golden."""
    from tests.helpers import OracleEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "49.binary"))
    tree = read_newick_topology(os.path.join(golden_dir, "49.tree"), taxa)
    engines = [OracleEngine(p.tips, p.wgt,
                            ea.DnaGtrModel(p.frequencies, [1.0] * 6, 1.0))
               for p in parts]
    ts = TreeSearch(tree, engines,
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts])
    ts.evaluate_generic(full=True)
    mid = ts.tree_evaluate(1.0)
    path = str(tmp_path / "ours.ckpt")
    write_checkpoint(path, tree, [build_model_entry(e.model)
                                  for e in engines], 49,
                     likelihoods=[mid])
    shutil.copy(os.path.join(golden_dir, "49.binary"),
                str(tmp_path / "t49.binary"))
    shutil.copy(os.path.join(golden_dir, "49.tree"),
                str(tmp_path / "49.tree"))
    r = subprocess.run(
        [_REF_BIN, "-s", "t49.binary", "-t", "49.tree", "-m", "GAMMA",
         "-f", "E", "-R", "ours.ckpt", "-n", "ours"],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=600)
    txt = r.stdout + r.stderr
    restart = [l for l in txt.splitlines() if "Restart with likelihood" in l]
    final = [l for l in txt.splitlines() if "Likelihood tree 0" in l]
    assert restart and final, txt[-2000:]
    restored = float(restart[0].split(":")[1])
    assert abs(restored - mid) < 1e-6 * abs(mid)
    fin = float(final[0].split(":")[1])
    assert abs(fin - GOLDEN_FINAL_LNL) < TOL_ABS, fin
