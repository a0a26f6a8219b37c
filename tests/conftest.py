import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    try:
        import torch
        have_gpu = torch.cuda.is_available()
    except Exception:
        have_gpu = False
    if have_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def golden_dir():
    return os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")


@pytest.fixture(scope="session")
def optimized_49_cpu(golden_dir):
    """The 49-taxa -f E pipeline run ONCE on the CPU oracle engines
    (shared by the e2e parity test and the quartet test, which the
    reference also runs on the modOpt-optimized state)."""
    import examl_amd as ea
    from examl_amd.examl_io import read_byte_file, read_newick_topology
    from examl_amd.search import TreeSearch
    from tests.helpers import OracleEngine
    taxa, parts = read_byte_file(os.path.join(golden_dir, "49.binary"))
    tree = read_newick_topology(os.path.join(golden_dir, "49.tree"), taxa)
    engines = [OracleEngine(p.tips, p.wgt,
                            ea.DnaGtrModel(p.frequencies, [1.0] * 6, 1.0))
               for p in parts]
    ts = TreeSearch(
        tree, engines,
        opt_freq_flags=[bool(p.optimizeBaseFrequencies) for p in parts])
    lnl = ts.tree_evaluation_mode()
    return ts, lnl
