"""Command-line drop-in for the reference `examl` binary.

Mirrors the reference's option surface (axml.c get_args:941) over the
MI355X engines: the same flags mean the same things, the same output
files appear (ExaML_info/ExaML_result/ExaML_quartets/goodTrees), and a
reference user can point their invocation here unchanged:

    python -m examl_amd -s 49.binary -t 49.tree -m GAMMA -f E -n RUN

Supported: -s -t -g -p -m -f(e|E|d|o|q) -n -w -a -M -S -D -B -c -e -i
-r -Y -v -R -I, plus AUTO protein partitions (per-partition model
selection); the binary checkpoint read/write interchange itself lives
in examl_amd.checkpoint.  This is the PRODUCT path: it requires the
HIP extension and an AMD GPU and fails loudly without one (no CPU
fallback).
"""

import os
import sys

import numpy as np


def _usage():
    sys.stderr.write(
        "examl_amd (MI355X-native ExaML)\n"
        "  -s byteFile -n runName and one of -t startingTree | "
        "-g constraintTree -p seed\n"
        "  [-m GAMMA|PSR] [-f e|E|d|o|q] [-w workdir] [-a] [-M] [-S]\n"
        "  [-D] [-B numBestTrees] [-c numRateCategories] "
        "[-e likelihoodEpsilon]\n"
        "  [-i initialRearrangementSetting] [-r randomQuartets -p seed] "
        "[-Y quartetGroupingFile]\n")


def _parse_args(argv):
    opts = {"m": "GAMMA", "f": "d", "w": os.getcwd(), "a": False,
            "M": False, "S": False, "D": False, "B": 0, "c": 25,
            "e": 0.1, "i": None, "r": 0, "p": None, "s": None, "t": None,
            "g": None, "n": None, "Y": None, "R": None}
    flags = set("aMSD")
    valued = set("stgpmfnwBceirYR")
    i = 0
    while i < len(argv):
        a = argv[i]
        if not a.startswith("-") or len(a) != 2:
            sys.exit(f"unknown argument {a!r}")
        c = a[1]
        if c == "h":
            _usage()
            sys.exit(0)
        if c == "v":
            print("examl_amd: MI355X-native ExaML-compatible engine")
            sys.exit(0)
        if c in flags:
            opts[c] = True
            i += 1
        elif c in valued:
            if i + 1 >= len(argv):
                sys.exit(f"option -{c} needs a value")
            v = argv[i + 1]
            if c in "Bci":
                v = int(v)
            elif c == "e":
                v = float(v)
            elif c == "p":
                v = int(v)
            elif c == "r":
                v = int(v)
            opts[c] = v
            i += 2
        else:
            sys.exit(f"unknown option -{c}")
    return opts


LG4M, LG4X, AUTO = 20, 21, 19
WAG = 4


def _build_engines(parts, opts, device):
    """One engine per partition, by data type and rate model — the
    initializePartitions role (axml.c:1936).  Returns (engines,
    auto_flags, empirical_freqs) for the AUTO selection machinery."""
    import examl_amd as ea
    aa = np.load(os.path.join(os.path.dirname(os.path.abspath(__file__)),
                              "data", "aa_models.npz"))
    psr = opts["m"] == "PSR"
    engines = []
    auto_flags = []
    empirical = []
    for p in parts:
        auto_flags.append(p.states == 20 and p.protModels == AUTO)
        empirical.append(p.frequencies)
        if p.states == 4:
            model = ea.DnaGtrModel(p.frequencies, [1.0] * 6, 1.0,
                                   use_median=opts["a"])
            if psr:
                w = p.upper - p.lower
                cat_cls = ea.SaveCatEngine if opts["S"] else ea.DnaCatEngine
                engines.append(cat_cls(
                    p.tips, p.wgt, model, np.zeros(w, dtype=np.int32),
                    np.array([1.0]), device=device))
            elif opts["S"]:
                engines.append(ea.SaveDnaEngine(p.tips, p.wgt, model,
                                                device=device))
            else:
                engines.append(ea.DnaGammaEngine(p.tips, p.wgt, model,
                                                 device=device))
        else:
            if p.protModels == AUTO:
                # AUTO starts as WAG (models.c:4222); protFreqs==0 ->
                # empirical frequencies initially (models.c:3528)
                freqs = (p.frequencies if p.protFreqs == 0
                         else aa["frequencies"][WAG])
                model = ea.ProtGtrModel(freqs, aa["rates190"][WAG], 1.0,
                                        use_median=opts["a"])
                engines.append(ea.DnaGammaEngine(p.tips, p.wgt, model,
                                                 device=device))
                continue
            if p.protModels == LG4M:
                engines.append(ea.Lg4Engine(p.tips, p.wgt,
                                            ea.Lg4Model.lg4m(),
                                            device=device))
                continue
            if p.protModels == LG4X:
                engines.append(ea.Lg4Engine(p.tips, p.wgt,
                                            ea.Lg4Model.lg4x(),
                                            device=device))
                continue
            freqs = (aa["frequencies"][p.protModels] if p.protFreqs == 0
                     else p.frequencies)
            model = ea.ProtGtrModel(freqs, aa["rates190"][p.protModels],
                                    1.0, use_median=opts["a"])
            if psr:
                w = p.upper - p.lower
                cat_cls = ea.SaveCatEngine if opts["S"] else ea.ProtCatEngine
                engines.append(cat_cls(
                    p.tips, p.wgt, model, np.zeros(w, dtype=np.int32),
                    np.array([1.0]), device=device))
            elif opts["S"]:
                engines.append(ea.SaveProtEngine(p.tips, p.wgt, model,
                                                 device=device))
            else:
                engines.append(ea.DnaGammaEngine(p.tips, p.wgt, model,
                                                 device=device))
    return engines, auto_flags, empirical


def main(argv=None, device=None):
    opts = _parse_args(sys.argv[1:] if argv is None else argv)
    if not opts["s"] or not opts["n"]:
        _usage()
        sys.exit("-s and -n are required")
    if not opts["t"] and not opts["g"] and not opts["R"]:
        _usage()
        sys.exit("specify a starting tree: -t treeFile, "
                 "-g constraintTree -p seed, or -R checkpointFile")

    if opts["g"] and opts["p"] is None:
        sys.exit("you must specify a random number seed via -p when "
                 "using a constraint tree")
    if opts["f"] == "q" and opts["r"] == 0 and not opts["Y"]:
        sys.exit("you must specify either -r randomQuartetNumber or "
                 "-Y quartetGroupingFileName with -f q")
    if device is None:
        # PRODUCT path: MI355X only, no CPU fallback
        import torch
        if not torch.cuda.is_available():
            sys.exit("examl_amd requires an AMD GPU (the HIP engines); "
                     "torch.cuda is not available on this host")
        device = "cuda:0"

    import examl_amd as ea
    from examl_amd.examl_io import (read_byte_file, read_newick_trees,
                                    to_newick)
    from examl_amd.search import TreeSearch, evaluate_trees
    from examl_amd.spr import (SprSearch, SprTree, read_constraint_tree)

    name = opts["n"]
    wdir = opts["w"]
    info_path = os.path.join(wdir, f"ExaML_info.{name}")
    info_f = open(info_path, "w")

    def log(msg):
        info_f.write(msg + "\n")
        info_f.flush()
        print(msg)

    taxa, parts = read_byte_file(opts["s"])
    log(f"partitions: {len(parts)}, taxa: {len(taxa)}, model {opts['m']}")

    if opts["R"]:
        # restart (searchAlgo.c:1726): tree from the checkpoint's node
        # image, engines rebuilt from the stored model arrays (our
        # eigendecomposition reproduces the stored one bit-for-bit)
        import examl_amd as ea
        from examl_amd.checkpoint import (FAST_SPRS, MOD_OPT,
                                          REARR_SETTING, SLOW_SPRS,
                                          read_checkpoint, spr_tree)
        from examl_amd.search import TreeSearch
        from examl_amd.spr import SprSearch
        from examl_amd.examl_io import to_newick
        psr = opts["m"] == "PSR"
        cl = sum(p.upper - p.lower for p in parts)
        ckpt = read_checkpoint(opts["R"], len(taxa),
                               [p.states for p in parts],
                               rate_het="CAT" if psr else "GAMMA",
                               crunched_length=cl if psr else None)
        # engines via the (monkeypatchable) factory, then install the
        # checkpoint's model/rate state in place (readCheckpoint's
        # restore, searchAlgo.c:1502+)
        engines, _af, _ef = _build_engines(parts, opts, device)
        for p, m, eng in zip(parts, ckpt.models, engines):
            assert p.states == 4, "-R restart wired for DNA"
            model = ea.DnaGtrModel(m["frequencies"], m["substRates"],
                                   m["alpha"], use_median=opts["a"])
            eng.model = model
            eng.upload_model()
            if psr:
                cptr = np.asarray(ckpt.rate_category[p.lower:p.upper],
                                  dtype=np.int32).copy()
                rates = np.asarray(
                    m["per_site_rates"][:m["num_cats"]]).copy()
                eng.set_site_rates(cptr, rates)
        kw = dict(opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                                  for p in parts],
                  max_categories=opts["c"])
        if opts["M"]:
            kw["per_gene_bl"] = True
        if psr:
            kw["rate_het"] = "CAT"
        res_path = os.path.join(wdir, f"ExaML_result.{name}")
        if opts["f"] in ("d", "o"):
            assert ckpt.state in (REARR_SETTING, FAST_SPRS,
                                  SLOW_SPRS), \
                "checkpoint state does not match -f d"
            st = spr_tree(ckpt, len(taxa))
            ts = TreeSearch(st, engines, **kw)
            if psr:
                ts.rate_cat_invocations = \
                    ckpt.optimize_rate_category_invocations
                ts.cat_patrat = [
                    np.asarray(ckpt.patrat[p.lower:p.upper]).copy()
                    for p in parts]
            sp = SprSearch(ts, do_cutoff=(opts["f"] == "d"),
                           convergence_criterion=opts["D"],
                           save_best_trees=opts["B"], log=log)
            if opts["D"]:
                from examl_amd.examl_io import to_newick_topology
                sp.seed_rfconv_from_checkpoint(ckpt, taxa)
                sp.topology_string_fn = (
                    lambda: to_newick_topology(st, taxa))
            final = sp.compute_big_rapid(estimate_model=True,
                                         checkpoint=ckpt)
            tree_out = st
        else:
            assert ckpt.state == MOD_OPT, "checkpoint state/mode mismatch"
            ts = TreeSearch(ckpt.tree, engines, **kw)
            ts.evaluate_generic(full=True)
            final = ts.mod_opt(opts["e"])
            tree_out = ckpt.tree
        log(f"Likelihood of best tree: {final:.6f}")
        with open(res_path, "w") as f:
            f.write(to_newick(tree_out, taxa) + "\n")
        log(f"Final tree written to: {res_path}")
        info_f.close()
        return 0

    engines, auto_flags, empirical = _build_engines(parts, opts, device)
    kw = dict(opt_freq_flags=[bool(p.optimizeBaseFrequencies)
                              for p in parts],
              max_categories=opts["c"])
    if any(auto_flags):
        kw["auto_flags"] = auto_flags
        kw["empirical_freqs"] = empirical
    if opts["m"] == "PSR":
        kw["rate_het"] = "CAT"
    if opts["M"]:
        kw["per_gene_bl"] = True

    prot_freqs0 = [p.protFreqs for p in parts]

    prot_freqs0 = [p.protFreqs for p in parts]

    def _ts(tree):
        ts = TreeSearch(tree, engines, **kw)
        ts.prot_freqs = list(prot_freqs0)
        return ts

    mode = opts["f"]
    if mode in ("e", "E"):
        trees = read_newick_trees(opts["t"], taxa)
        if any(auto_flags):
            # AUTO + multi-tree: evaluate per tree with fresh searches
            lnls = []
            for i, tree in enumerate(trees):
                ts = _ts(tree)
                if i > 0:
                    ts.reset_branches()
                lnls.append(ts.tree_evaluation_mode(epsilon=opts["e"]))
        else:
            lnls = evaluate_trees(trees, engines, fast=(mode == "e"),
                                  epsilon=opts["e"], **kw)
        best = int(np.argmax(lnls))
        for i, v in enumerate(lnls):
            log(f"Likelihood tree {i}: {v:.6f}")
        # trees are optimized in place; emit the best-scoring one
        result = to_newick(trees[best], taxa)
        final = lnls[best]
    elif mode in ("d", "o"):
        trees = read_newick_trees(opts["t"], taxa) if opts["t"] else None
        if opts["g"]:
            with open(opts["g"]) as f:
                st, cv = read_constraint_tree(f.read(), taxa, opts["p"])
        else:
            st = SprTree.from_phylo(trees[0])
            cv = None
        ts = _ts(st)
        sp = SprSearch(ts, do_cutoff=(mode == "d"),
                       convergence_criterion=opts["D"],
                       save_best_trees=opts["B"], log=log)
        sp.constraint = cv
        if (all(p.states == 4 for p in parts)
                and opts["m"] in ("GAMMA", "PSR")
                and cv is None):
            # resumable like the reference: one binary checkpoint per
            # SPR cycle (writeCheckpointInner, searchAlgo.c:1153); PSR
            # additionally carries rateCategory/patrat + the per-model
            # category state (searchAlgo.c:1188-1201)
            from examl_amd.checkpoint import (build_model_entry,
                                              write_checkpoint)
            counter = [0]

            def _writer(state, fields):
                path = os.path.join(
                    wdir, f"ExaML_binaryCheckpoint.{name}_{counter[0]}")
                kw2 = {}
                if opts["m"] == "PSR":
                    kw2 = dict(
                        rate_het="CAT",
                        invocations=ts.rate_cat_invocations,
                        rate_category=np.concatenate(
                            [e.cptr for e in ts.engines]),
                        patrat=np.concatenate(ts.cat_patrat),
                    )
                    entries = [build_model_entry(
                        e.model, num_cats=e.num_cats,
                        per_site_rates=e.per_site_rates)
                        for e in ts.engines]
                else:
                    entries = [build_model_entry(e.model)
                               for e in ts.engines]
                write_checkpoint(
                    path, st, entries,
                    len(taxa), state=state, spr=fields,
                    start_number=st.start, per_gene_bl=opts["M"],
                    likelihood_epsilon=opts["e"],
                    use_median=opts["a"], save_best_trees=opts["B"],
                    save_memory=opts["S"], search_convergence=opts["D"],
                    categories=opts["c"],
                    initial_set=opts["i"] is not None,
                    initial=10 if opts["i"] is None else opts["i"],
                    tree0=sp.slot_tree_strings[0],
                    tree1=sp.slot_tree_strings[1], **kw2)
                counter[0] += 1

            sp.checkpoint_writer = _writer
            if opts["D"]:
                from examl_amd.examl_io import to_newick_topology
                sp.topology_string_fn = (
                    lambda: to_newick_topology(st, taxa))
        final = sp.compute_big_rapid(estimate_model=True,
                                     initial_trav=opts["i"])
        log(f"Likelihood of best tree: {final:.6f}")
        # the result file holds the best tree (written BEFORE the -B
        # good-trees replay, which leaves the tree at the last entry)
        result = to_newick(st, taxa)
        if opts["B"]:
            gt_path = os.path.join(
                wdir, f"RAxML_{len(sp.good_trees)}_goodTrees.{name}")
            with open(gt_path, "w") as f:
                for i, lnl in enumerate(sp.good_trees):
                    sp.best_ml.recall(i + 1, ts)
                    f.write(to_newick(st, taxa) + "\n")
                    log(f"tree {i + 1} likelihood {lnl:.6f}")
    elif mode == "q":
        from examl_amd.quartets import compute_quartets
        trees = read_newick_trees(opts["t"], taxa)
        ts = _ts(trees[0])
        ts.tree_evaluation_mode(epsilon=opts["e"])
        groups = None
        if opts["Y"]:
            groups = _parse_groups(opts["Y"], taxa)
        quartets = compute_quartets(
            ts.engines, len(taxa), random_quartets=opts["r"],
            seed=opts["p"] or 0, groups=groups,
            per_gene_bl=opts["M"])
        q_path = os.path.join(wdir, f"ExaML_quartets.{name}")
        with open(q_path, "w") as f:
            f.write("Taxon names and indices:\n\n")
            for i, t in enumerate(taxa):
                f.write(f"{t} {i + 1}\n")
            f.write("\n")
            for a, b, c, d, lnl in quartets:
                f.write(f"{a} {b} | {c} {d}: {lnl:.6f}\n")
        log(f"quartets written to {q_path}")
        info_f.close()
        return 0
    else:
        sys.exit(f"unknown -f mode {mode!r}")

    res_path = os.path.join(wdir, f"ExaML_result.{name}")
    with open(res_path, "w") as f:
        f.write(result + "\n")
    log(f"Final tree written to: {res_path}")
    info_f.close()
    return 0


def _parse_groups(path, taxa):
    """groupingParser (quartets.c:69): four parenthesized taxon lists."""
    tip_no = {t: i + 1 for i, t in enumerate(taxa)}
    with open(path) as f:
        text = f.read()
    groups = []
    for chunk in text.split(")")[:-1]:
        chunk = chunk.split("(")[-1]
        names = [x.strip() for x in chunk.split(",") if x.strip()]
        groups.append([tip_no[n] for n in names])
    if len(groups) != 4:
        sys.exit("quartet grouping file must define exactly 4 groups")
    return groups


if __name__ == "__main__":
    sys.exit(main())
