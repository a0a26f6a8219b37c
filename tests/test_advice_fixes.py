"""Dedicated CPU coverage for the round-1 ADVICE fixes:
- the checkpoint commandLine block carries the run's actual options
  (checkCommandLineArguments, searchAlgo.c:1383 hard-fails on mismatch)
- LG4 per-category eigensystem blocks round-trip through
  write_checkpoint/read_checkpoint
- -D resume re-populates the RF-convergence table from the stored
  tree0/tree1 topology strings (readCheckpoint, searchAlgo.c:1545-1580)
  and the strings themselves are written."""

import struct

import numpy as np
import pytest

import examl_amd as ea
from examl_amd.checkpoint import (MAX_CATEGORIES, read_checkpoint,
                                  write_checkpoint)
from examl_amd.examl_io import to_newick_topology
from examl_amd.spr import RfConvergence, SprTree


def _tree(ntips=8, seed=3):
    t = ea.PhyloTree.random(ntips, seed=seed, rng_z=True)
    return t


def _models(n=1, states=4, lg4=False):
    out = []
    rng = np.random.default_rng(5)
    for _ in range(n):
        m = {
            "num_cats": 1,
            "per_site_rates": np.array([1.0]),
            "EIGN": rng.uniform(-2, 0, 20 if states == 20 else 4),
            "EV": rng.uniform(-1, 1, states * states),
            "EI": rng.uniform(-1, 1, states * states),
            "freqExponents": np.zeros(states),
            "frequencies": np.full(states, 1.0 / states),
            "tipVector": rng.uniform(0, 1, 460 if states == 20 else 64),
            "substRates": rng.uniform(0.1, 3.0,
                                      190 if states == 20 else 6),
            "alpha": 0.73,
            "gammaRates": np.array([0.1, 0.4, 1.0, 2.5]),
            "protModels": 20 if lg4 else (2 if states == 20 else 0),
            "autoProtModels": 2,
        }
        if lg4:
            for key, ln in (("rawEIGN_LG4", 20), ("EIGN_LG4", 20),
                            ("EV_LG4", 400), ("EI_LG4", 400),
                            ("frequencies_LG4", 20),
                            ("tipVector_LG4", 460),
                            ("substRates_LG4", 190)):
                m[key] = [rng.uniform(-1, 1, ln) for _ in range(4)]
        out.append(m)
    return out


def test_cmd_block_carries_run_options(tmp_path):
    """write_checkpoint stores -a/-B/-S/-D/-c/-e/-i in the commandLine
    block at the reference's offsets (axml.h:660-679)."""
    p = str(tmp_path / "ck.bin")
    t = _tree()
    write_checkpoint(p, t, _models(), 8, likelihoods=[-1234.5],
                     use_median=True, save_best_trees=7, save_memory=True,
                     search_convergence=True, categories=13,
                     initial_set=True, initial=15,
                     likelihood_epsilon=0.03)
    d = open(p, "rb").read()
    c = 1248
    assert struct.unpack_from("<i", d, c + 0)[0] == 1    # useMedian
    assert struct.unpack_from("<i", d, c + 4)[0] == 7    # saveBestTrees
    assert struct.unpack_from("<i", d, c + 8)[0] == 1    # saveMemory
    assert struct.unpack_from("<i", d, c + 12)[0] == 1   # searchConv.
    assert struct.unpack_from("<d", d, c + 24)[0] == 0.03
    assert struct.unpack_from("<i", d, c + 32)[0] == 13  # categories
    assert struct.unpack_from("<i", d, c + 44)[0] == 1   # initialSet
    assert struct.unpack_from("<i", d, c + 48)[0] == 15  # initial
    # and the defaults stay default-shaped when not passed
    p2 = str(tmp_path / "ck2.bin")
    write_checkpoint(p2, t, _models(), 8, likelihoods=[-1.0])
    d2 = open(p2, "rb").read()
    assert struct.unpack_from("<i", d2, c + 0)[0] == 0
    assert struct.unpack_from("<i", d2, c + 32)[0] == MAX_CATEGORIES


def test_lg4_checkpoint_write_roundtrip(tmp_path):
    """The four per-category eigensystem blocks between weightExponents
    and alpha (writeCheckpointInner, searchAlgo.c:1244-1260) round-trip
    through our writer and reader."""
    p = str(tmp_path / "lg4.bin")
    t = _tree()
    models = _models(states=20, lg4=True)
    write_checkpoint(p, t, models, 8, likelihoods=[-99.0])
    ck = read_checkpoint(p, 8, [20], prot_models=[20])
    m = ck.models[0]
    for key in ("rawEIGN_LG4", "EIGN_LG4", "EV_LG4", "EI_LG4",
                "frequencies_LG4", "tipVector_LG4", "substRates_LG4"):
        for k in range(4):
            assert np.array_equal(m[key][k], models[0][key][k]), (key, k)
    assert m["alpha"] == models[0]["alpha"]
    assert m["protModels"] == 20


def test_rfconv_seed_from_newick_matches_store():
    """Seeding a fresh RF table from the topology STRING of a tree gives
    the same bipartition slots as storing the live tree — the -D resume
    path (readCheckpoint: treeReadTopologyString +
    bitVectorInitravSpecial)."""
    ntips = 9
    taxa = [f"T{i}" for i in range(1, ntips + 1)]
    st = SprTree.random(ntips, seed=11) if hasattr(SprTree, "random") \
        else None
    if st is None:
        from examl_amd.checkpoint import spr_tree  # noqa: F401
        pt = ea.PhyloTree.random(ntips, seed=11, rng_z=True)
        st = SprTree.from_phylo(pt)
    live = RfConvergence(st)
    live.store(0)
    s0 = to_newick_topology(st, taxa)

    seeded = RfConvergence(st)
    seeded.seed_from_newick(s0, 0, taxa)
    assert seeded.table == live.table

    # slot-1 semantics: a second topology lands in bit 2
    seeded2 = RfConvergence(st)
    seeded2.seed_from_newick(s0.encode() + b"\0garbage", 1, taxa)
    assert set(seeded2.table) == set(live.table)
    assert all(v == 2 for v in seeded2.table.values())
    # rrf of identical trees in both slots is 0
    both = RfConvergence(st)
    both.seed_from_newick(s0, 0, taxa)
    both.seed_from_newick(s0, 1, taxa)
    assert both.rrf() == 0.0
