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


RESTART_FAST_LNL = -2744.20054491362134285736829042434692382812500
RESTART_SLOW_LNL = -2741.47310239244689000770449638366699218750000
SPR_FINAL = -2741.473102


def _resume_spr(golden_dir, ckpt_name, restart_lnl):
    from examl_amd.checkpoint import spr_tree
    from examl_amd.spr import SprSearch
    from tests.helpers import OracleEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "12.binary"))
    ck = read_checkpoint(os.path.join(golden_dir, ckpt_name), 12, [4])
    st = spr_tree(ck, 12)
    engines = [OracleEngine(p.tips, p.wgt,
                            ea.DnaGtrModel(m["frequencies"],
                                           m["substRates"], m["alpha"]))
               for p, m in zip(parts, ck.models)]
    ts = TreeSearch(st, engines,
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts])
    assert ts.evaluate_generic(full=True) == restart_lnl  # bit-exact
    sp = SprSearch(ts)
    lnl = sp.compute_big_rapid(estimate_model=True, checkpoint=ck)
    assert abs(lnl - SPR_FINAL) < abs(SPR_FINAL) * 1e-6, lnl


def test_resume_spr_search_from_fast_checkpoint(golden_dir):
    """-R restart mid-FAST_SPRS: the exact ring structure and loop
    variables come from a reference-written checkpoint
    (ExaML_binaryCheckpoint.R1_3 of examl-AVX on the 12-taxon golden);
    the restored lnL is bit-exact vs the reference's own "ExaML Restart
    with likelihood" line and the resumed search lands on the same
    final tree score the reference's -R run prints."""
    _resume_spr(golden_dir, "12.spr_fast.ckpt.bin", RESTART_FAST_LNL)


def test_resume_spr_search_from_slow_checkpoint(golden_dir):
    """-R restart mid-SLOW_SPRS (thorough loop), same contract."""
    _resume_spr(golden_dir, "12.spr_slow.ckpt.bin", RESTART_SLOW_LNL)


def test_spr_checkpoint_fields(golden_dir):
    """The search-state fields of checkPointState (axml.h:679-720)
    parse at the documented offsets."""
    ck = read_checkpoint(os.path.join(golden_dir, "12.spr_slow.ckpt.bin"),
                         12, [4])
    assert ck.state == 3  # SLOW_SPRS
    assert ck.fast_iterations == 2 and ck.thorough_iterations == 1
    assert ck.best_trav == 5 and ck.thorough == 1 and ck.impr == 0
    assert (ck.rearrangements_min, ck.rearrangements_max) == (1, 5)
    assert ck.tr_it_count == 3


@pytest.mark.skipif(not os.path.exists(_REF_BIN),
                    reason="reference examl-AVX not built")
def test_reference_resumes_from_our_spr_checkpoint(golden_dir, tmp_path):
    """Write interchange for SPR-state checkpoints: OUR search writes
    FAST_SPRS/SLOW_SPRS checkpoints at the reference's write points and
    the unmodified reference binary -R restarts from them — its printed
    restart lnL is bit-identical to the one it prints when restarting
    from its OWN checkpoint of the same state, and it completes to the
    same final score."""
    from examl_amd.checkpoint import build_model_entry, write_checkpoint
    from examl_amd.spr import SprSearch, SprTree
    from tests.helpers import OracleEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "12.binary"))
    t = read_newick_topology(os.path.join(golden_dir, "12.tree"), taxa)
    st = SprTree.from_phylo(t)
    engines = [OracleEngine(p.tips, p.wgt,
                            ea.DnaGtrModel(p.frequencies, [1.0] * 6, 1.0))
               for p in parts]
    ts = TreeSearch(st, engines,
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts])
    sp = SprSearch(ts)
    written = []

    def writer(state, fields):
        path = str(tmp_path / f"our_{len(written)}")
        write_checkpoint(path, st,
                         [build_model_entry(e.model) for e in ts.engines],
                         12, state=state, spr=fields,
                         start_number=st.start)
        written.append((path, state, fields.get("fast_iterations", 0),
                        fields.get("thorough_iterations", 0)))

    sp.checkpoint_writer = writer
    sp.compute_big_rapid(estimate_model=True)
    rearr = next(p for p, s, fi, ti in written if s == 1)
    fast = next(p for p, s, fi, ti in written if s == 2 and fi == 1)
    slow = next(p for p, s, fi, ti in written if s == 3 and ti == 1)
    shutil.copy(os.path.join(golden_dir, "12.binary"),
                tmp_path / "12.binary")
    # the image serializes our LIVE rings, so the reference's resumed
    # traversals replay ITS OWN restart trajectories exactly: the
    # finals below are what examl-AVX prints when restarting from its
    # own checkpoints of the same states
    for path, expect_restart, expect_final in (
            (rearr, None, -2741.473101),
            (fast, "-2744.20054491362134285736829042434692382812500",
             -2741.473102),
            (slow, "-2741.47310239244689000770449638366699218750000",
             -2741.473102)):
        name = "RES" + os.path.basename(path)
        subprocess.run(
            [_REF_BIN, "-s", "12.binary", "-R", path, "-m", "GAMMA",
             "-n", name], cwd=tmp_path, check=True,
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
            timeout=300)
        info = open(tmp_path / f"ExaML_info.{name}").read()
        if expect_restart is not None:
            restart = [ln for ln in info.splitlines()
                       if "Restart with likelihood" in ln][0]
            assert expect_restart in restart, restart
        final = float([ln for ln in info.splitlines()
                       if ln.startswith("Likelihood of best tree")]
                      [0].split(":")[1])
        assert abs(final - expect_final) < 5e-6, (path, final)


def test_resume_spr_search_from_rearr_checkpoint(golden_dir):
    """-R restart mid-REARR_SETTING (radius search): the resumed run
    replays the reference's own -R run from the same checkpoint — same
    bit-exact restored lnL (-3651.20215442176...) and the same final
    -2741.473101 the reference's restart prints (one trajectory digit
    off the uninterrupted run's -2741.473102, matching the reference's
    restart exactly)."""
    from examl_amd.checkpoint import spr_tree
    from examl_amd.spr import SprSearch
    from tests.helpers import OracleEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "12.binary"))
    ck = read_checkpoint(os.path.join(golden_dir,
                                      "12.spr_rearr.ckpt.bin"), 12, [4])
    assert ck.state == 1 and ck.maxtrav == 10
    st = spr_tree(ck, 12)
    engines = [OracleEngine(p.tips, p.wgt,
                            ea.DnaGtrModel(m["frequencies"],
                                           m["substRates"], m["alpha"]))
               for p, m in zip(parts, ck.models)]
    ts = TreeSearch(st, engines,
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts])
    assert ts.evaluate_generic(full=True) == \
        -3651.20215442176322540035471320152282714843750
    sp = SprSearch(ts)
    lnl = sp.compute_big_rapid(estimate_model=True, checkpoint=ck)
    assert abs(lnl - (-2741.473101)) < abs(2741.473101) * 1e-6, lnl


def test_read_lg4_checkpoint_and_resume(golden_dir):
    """LG4 checkpoints (per-category eigensystem block between
    weightExponents and alpha, writeCheckpointInner
    searchAlgo.c:1248-1262): restore an LG4X+LG4M -f E checkpoint from
    the reference, land bit-exactly on its "ExaML Restart with
    likelihood" value, and resume modOpt to the -f E golden."""
    import numpy as np

    from tests.helpers import OracleLg4Engine
    taxa, parts = read_byte_file(os.path.join(golden_dir,
                                              "12lg4.binary"))
    pm = [p.protModels for p in parts]
    ck = read_checkpoint(os.path.join(golden_dir, "12lg4.ckpt.bin"), 12,
                         [20, 20], prot_models=pm)
    engines = []
    for p, m in zip(parts, ck.models):
        mdl = ea.Lg4Model.lg4x() if p.protModels == 21 \
            else ea.Lg4Model.lg4m()
        mdl.EIGN4 = np.concatenate(m["EIGN_LG4"])
        mdl.EIGN4_raw = np.concatenate(m["rawEIGN_LG4"])
        mdl.EV4 = np.concatenate(m["EV_LG4"])
        mdl.EI4 = np.concatenate(m["EI_LG4"])
        mdl.tipVector4 = np.concatenate(m["tipVector_LG4"])
        mdl.gammaRates = np.asarray(m["gammaRates"])
        mdl.weights = np.asarray(m["weights"])
        mdl.weightExponents = np.asarray(m["weightExponents"])
        mdl.alpha = m["alpha"]
        engines.append(OracleLg4Engine(p.tips, p.wgt, mdl))
    ts = TreeSearch(ck.tree, engines,
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts])
    lnl = ts.evaluate_generic(full=True)
    assert lnl == -7387.55350209685184381669387221336364746093750
    fin = ts.mod_opt(0.1)
    assert abs(fin - (-7387.472983)) < abs(7387.472983) * 1e-6, fin


def test_resume_spr_search_M_from_checkpoint(golden_dir):
    """-R restart under -M: per-partition branch vectors come out of
    the node image (z[0..nb-1] per edge); restored lnL is bit-exact vs
    the reference's restart line and the resumed search lands on the
    -M -f d golden."""
    from examl_amd.checkpoint import spr_tree
    from examl_amd.spr import SprSearch
    from tests.helpers import OracleEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "12m.binary"))
    ck = read_checkpoint(os.path.join(golden_dir,
                                      "12m.spr_fast.ckpt.bin"), 12,
                         [4, 4])
    assert ck.per_gene_bl and ck.state == 2
    st = spr_tree(ck, 12)
    engines = [OracleEngine(p.tips, p.wgt,
                            ea.DnaGtrModel(m["frequencies"],
                                           m["substRates"], m["alpha"]))
               for p, m in zip(parts, ck.models)]
    ts = TreeSearch(st, engines,
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts],
                    per_gene_bl=True)
    assert ts.evaluate_generic(full=True) == \
        -2730.70260654699541191803291440010070800781250
    sp = SprSearch(ts)
    lnl = sp.compute_big_rapid(estimate_model=True, checkpoint=ck)
    assert abs(lnl - (-2728.477352)) < abs(2728.477352) * 1e-6, lnl


def test_resume_spr_search_psr_from_checkpoint(golden_dir):
    """-R restart under -m PSR: rate categories (rateCategory), per-site
    rates (patrat) and the optimizeRateCategoryInvocations counter come
    out of the checkpoint; restored lnL is bit-exact vs the reference's
    restart line and the resumed search lands on the PSR -f d golden."""
    import numpy as np

    from examl_amd.checkpoint import spr_tree
    from examl_amd.spr import SprSearch
    from tests.helpers import OracleCatEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "12.binary"))
    cl = sum(p.upper - p.lower for p in parts)
    ck = read_checkpoint(
        os.path.join(golden_dir, "12psr.spr_fast.ckpt.bin"), 12, [4],
        rate_het="CAT", crunched_length=cl)
    assert ck.state == 2 and ck.optimize_rate_category_invocations == 6
    st = spr_tree(ck, 12)
    engines = []
    for p, m in zip(parts, ck.models):
        model = ea.DnaGtrModel(m["frequencies"], m["substRates"],
                               m["alpha"])
        cptr = np.asarray(ck.rate_category[p.lower:p.upper],
                          dtype=np.int32).copy()
        rates = np.asarray(m["per_site_rates"][:m["num_cats"]]).copy()
        engines.append(OracleCatEngine(p.tips, p.wgt, model, cptr, rates))
    ts = TreeSearch(st, engines,
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts],
                    rate_het="CAT")
    ts.rate_cat_invocations = ck.optimize_rate_category_invocations
    ts.cat_patrat = [np.asarray(ck.patrat[p.lower:p.upper]).copy()
                     for p in parts]
    assert ts.evaluate_generic(full=True) == \
        -2520.03789080780052245245315134525299072265625
    sp = SprSearch(ts)
    lnl = sp.compute_big_rapid(estimate_model=True, checkpoint=ck)
    assert abs(lnl - (-2507.657682)) < abs(2507.657682) * 1e-6, lnl


def test_resume_spr_search_psr_M_from_checkpoint(golden_dir):
    """-R restart under -m PSR -M together (per-partition branch
    vectors AND rate-category state from one checkpoint); bit-exact
    restore, reference-golden final."""
    import numpy as np

    from examl_amd.checkpoint import spr_tree
    from examl_amd.spr import SprSearch
    from tests.helpers import OracleCatEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "12m.binary"))
    cl = sum(p.upper - p.lower for p in parts)
    ck = read_checkpoint(
        os.path.join(golden_dir, "12psrm.spr_slow.ckpt.bin"), 12, [4, 4],
        rate_het="CAT", crunched_length=cl)
    assert ck.state == 3 and ck.per_gene_bl
    st = spr_tree(ck, 12)
    engines = []
    for p, m in zip(parts, ck.models):
        model = ea.DnaGtrModel(m["frequencies"], m["substRates"],
                               m["alpha"])
        cptr = np.asarray(ck.rate_category[p.lower:p.upper],
                          dtype=np.int32).copy()
        rates = np.asarray(m["per_site_rates"][:m["num_cats"]]).copy()
        engines.append(OracleCatEngine(p.tips, p.wgt, model, cptr, rates))
    ts = TreeSearch(st, engines,
                    opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                    for p in parts],
                    rate_het="CAT", per_gene_bl=True)
    ts.rate_cat_invocations = ck.optimize_rate_category_invocations
    ts.cat_patrat = [np.asarray(ck.patrat[p.lower:p.upper]).copy()
                     for p in parts]
    assert ts.evaluate_generic(full=True) == \
        -2496.48818503920119837857782840728759765625
    sp = SprSearch(ts)
    lnl = sp.compute_big_rapid(estimate_model=True, checkpoint=ck)
    assert abs(lnl - (-2496.442393)) < abs(2496.442393) * 1e-6, lnl
