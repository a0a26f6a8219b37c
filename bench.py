#!/usr/bin/env python3
"""Headline benchmark: ExaML's north-star workload on MI355X.

Workload (BASELINE.json configs[1]): synthetic DNA, 50 taxa x 1,000,000
sites per GPU, single partition, GTRGAMMA, fp64.  One "step" = one full-tree
evaluateGeneric: a full post-order traversal (48 newview CLV updates over
all sites) + the root lnL evaluation + the per-partition lnL all-reduce
(RCCL when world>1) — the per-partition body of SURVEY.md §3.1.

  python bench.py --gpus N --steps K --warmup W

N>1 is launched by the driver as one rank per GPU via torch.distributed.run;
sites shard one-million-per-rank (weak scaling — per-GPU work fixed, the
reference's site sharding with the C1 all-reduce in the loop).

Output: ONE JSON line from rank 0 (see the repo contract), including
  roofline     — the dominant kernel (newview INNER_INNER), algorithmic
                 bytes/launch over its HIP-event launch time vs 8 TB/s HBM3E
  cpu_baseline — the reference's own AVX kernel (oracle/_ref/libref.so)
                 timed on this box's host cores (kind "reference"), or the
                 oracle C restatement (kind "port") if _ref is absent.
"""

import argparse
import ctypes
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch  # noqa: E402

import examl_amd as ea  # noqa: E402
from examl_amd.synthetic import make_alignment, make_alignment_aa  # noqa: E402

NTAXA = 50
SITES_PER_GPU = 1_000_000
SITES_PER_GPU_PROT = 200_000
# algorithmic HBM traffic per site for newview INNER_INNER:
#   DNA: 2x16 fp64 read + 16 fp64 write + 4 B wgt = 388 B
#   protein: 2x80 fp64 read + 80 fp64 write + 4 B wgt = 1924 B
BYTES_PER_SITE_II = {4: 388.0, 20: 1924.0}
HBM_PEAK = 8.0e12  # B/s, MI355X HBM3E spec


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, file=sys.stderr, flush=True)


def cpu_baseline_leg(width=200_000, budget_s=10.0, protein=False):
    """Time the reference AVX newview (INNER_INNER) single-threaded on this
    host, bounded to ~budget_s; returns the cpu_baseline dict."""
    try:
        import oracle as O
        lib = O._ref if O.have_ref() else None
        kind = "reference" if lib is not None else "port"
        st = 20 if protein else 4
        span = 4 * st
        if protein:
            width = min(width, 20_000)
        rng = np.random.default_rng(1)
        x1 = O.aligned(width * span)
        x1[:] = rng.uniform(0.01, 1.0, width * span)
        x2 = O.aligned(width * span)
        x2[:] = rng.uniform(0.01, 1.0, width * span)
        wgt = np.ones(width, dtype=np.int32)
        if protein:
            lg = np.load(os.path.join(os.path.dirname(
                os.path.abspath(__file__)), "examl_amd", "data",
                "lg_model.npz"))
            EIGN, EV, EI, tipVector = O.init_gtr_aa(lg["frequencies"],
                                                    lg["rates190"])
            nv = O.newview_prot_gamma
            name = "newviewGTRGAMMAPROT_AVX"
        else:
            EIGN, EV, EI, tipVector = O.init_gtr_dna(
                [0.25] * 4, [1.0, 2.0, 0.8, 1.1, 3.0, 1.0])
            nv = O.newview_dna_gamma
            name = "newviewGTRGAMMA_AVX"
        g = O.make_gamma_cats(0.5)
        left, right = O.make_p(np.log(0.9), np.log(0.7), g, EI, EIGN, 4, st)
        # one calibration call, then fill the budget
        t0 = time.perf_counter()
        nv(ea.INNER_INNER, x1, x2, EV, tipVector, None, None, width, left,
           right, wgt, lib=lib)
        per = time.perf_counter() - t0
        reps = max(3, int(budget_s / max(per, 1e-3)))
        t0 = time.perf_counter()
        for _ in range(reps):
            nv(ea.INNER_INNER, x1, x2, EV, tipVector, None, None, width,
               left, right, wgt, lib=lib)
        el = time.perf_counter() - t0
        return {
            "value": reps * width / el,
            "unit": "site-updates/s",
            "cores": 1,
            "kind": kind,
            "sample": f"{name} INNER_INNER, {width} sites x {reps} reps, "
                      f"1 thread" + ("" if kind == "reference"
                                     else " (oracle restatement)"),
        }
    except Exception as e:  # pragma: no cover
        log(f"cpu_baseline failed: {e}")
        return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=60)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--sites", type=int, default=0)
    ap.add_argument("--taxa", type=int, default=NTAXA)
    ap.add_argument("--protein", action="store_true",
                    help="config 4: 50 taxa x 200k sites, LG+GAMMA")
    ap.add_argument("--partitions", type=int, default=1,
                    help="partitions per GPU, each on its own HIP stream "
                         "(config 3 shard shape: --partitions 16 "
                         "--sites 125000)")
    ap.add_argument("--fast-math", action="store_true",
                    help="FMA protein newview (the reference's _FMA build "
                         "class; ~1 ulp/op vs the bit-exact default)")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl")
    device = torch.device(f"cuda:{local_rank}")

    ntips = args.taxa
    width = args.sites or (SITES_PER_GPU_PROT if args.protein
                           else SITES_PER_GPU)
    log(f"generating synthetic alignment: {ntips} taxa x {width} sites "
        f"(rank {rank}/{world}, {'LG+GAMMA' if args.protein else 'GTRGAMMA'})")
    if args.protein:
        tips, wgt = make_alignment_aa(ntips, width, seed=42 + rank)
        model = ea.ProtGtrModel.lg(alpha=0.75)
    else:
        tips, wgt = make_alignment(ntips, width, seed=42 + rank)
        model = ea.DnaGtrModel([0.28, 0.22, 0.24, 0.26],
                               [1.2, 2.9, 0.7, 1.0, 3.2, 1.0], alpha=0.6)
    if args.fast_math:
        ea.lib().examl_hip_fast_math(1)
    tree = ea.PhyloTree.random(ntips, seed=7)
    P = args.partitions
    assert width % P == 0
    pw = width // P
    engines = [ea.DnaGammaEngine(np.ascontiguousarray(tips[:, i*pw:(i+1)*pw]),
                                 wgt[i*pw:(i+1)*pw], model, device=device)
               for i in range(P)]
    eng = engines[0]
    multi = ea.MultiDnaEngine(engines) if P > 1 else None
    entries, (p, q, z) = tree.full_traversal()
    n_ops = len(entries)
    tc_counts = [sum(1 for e in entries if e.tipCase == t) for t in range(3)]

    def step():
        if P == 1:
            eng.newview_traversal(entries)
            return eng.evaluate_root(tree, p, q, z, all_reduce=world > 1)
        # partitioned: the fused mseg executors — one launch per
        # (traversal level x tipCase) covering all P partitions, then ONE
        # all-reduce of the per-partition lnL vector (the C1 collective)
        multi.newview_traversal(entries)
        lnl_vec = multi.evaluate_root(tree, p, q, z, all_reduce=world > 1)
        return lnl_vec.sum()

    # warmup (also captures the traversal's hipGraph)
    for _ in range(args.warmup):
        lnl_t = step()
    eng.sync()
    lnl0 = float(lnl_t.sum().cpu())

    # timed region: graph-replayed traversals, no profiling overhead
    if dist:
        dist.barrier()
    eng.sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    eng.sync()
    if dist:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # roofline pass: HIP-event per-launch timing of the same kernels
    # (profiling bypasses the graph cache; kernel durations are identical,
    # only the launch gaps differ)
    L = ea.lib()
    L.examl_hip_profile_reset()
    L.examl_hip_profile_enable(1)
    for _ in range(max(3, args.steps // 10)):
        step()
    eng.sync()
    L.examl_hip_profile_enable(0)
    ms = np.zeros(3)
    cnt = np.zeros(3, dtype=np.int64)
    L.examl_hip_profile_get(ms.ctypes.data_as(ctypes.c_void_p),
                            cnt.ctypes.data_as(ctypes.c_void_p))

    # max over ranks
    if dist:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.cpu())

    if rank != 0:
        return

    # roofline.  P == 1: the dominant kernel (newview INNER_INNER), bytes
    # per launch over HIP-event launch time.  P > 1 (fused mseg path): one
    # launch covers many (op, partition) segments, so use TOTAL algorithmic
    # newview bytes over TOTAL newview kernel time for the profiled steps —
    # per-launch bytes would overcount.
    BPS_TC = {4: (130.0, 260.0, 388.0), 20: (164.0, 1124.0, 1924.0)}
    bps = BYTES_PER_SITE_II[model.states]
    ii_ms, ii_n = float(ms[2]), int(cnt[2])
    if P == 1:
        achieved = (bps * pw * ii_n) / (ii_ms * 1e-3) \
            if ii_ms > 0 else None
    else:
        steps_prof = max(3, args.steps // 10)
        bytes_step = sum(tc_counts[t] * BPS_TC[model.states][t] * pw * P
                         for t in range(3))
        tot_ms = float(ms.sum())
        achieved = (steps_prof * bytes_step) / (tot_ms * 1e-3) \
            if tot_ms > 0 else None
    # PMC-measured HBM traffic for this exact workload (collected in a
    # separate rocprofv3 --pmc pass, corrected per MI355X_MICROARCH.md §HBM;
    # see profiles/r01_pmc_traffic.json)
    traffic = None
    cal = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                       "profiles", "r01_pmc_traffic.json")
    if pw == SITES_PER_GPU and model.states == 4 and os.path.exists(cal):
        with open(cal) as f:
            traffic = json.load(f)["traffic_bytes_per_launch"]
    roofline = {
        "bound": "hbm",
        "achieved": achieved / 1e9 if achieved else None,
        "peak": HBM_PEAK / 1e9,
        "unit": "GB/s",
        "frac": achieved / HBM_PEAK if achieved else None,
        "traffic": traffic,
    }

    cpu = None
    if not args.no_cpu_baseline and world == 1 \
            and not os.environ.get("EXAML_BENCH_NO_CPU"):
        log("timing CPU baseline (reference AVX kernel, 1 core)...")
        cpu = cpu_baseline_leg(protein=args.protein)

    site_updates = float(args.steps) * n_ops * width * world
    value = site_updates / elapsed
    out = {
        "metric": "newview site-lnL updates/sec",
        "value": value,
        "unit": "site-updates/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": elapsed / args.steps * 1e3,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "f64",
        "data": "synthetic",
        "config": {
            "workload": ("synthetic_prot_50taxa_200ksites_lg_gamma "
                         "(BASELINE.json configs[3])" if args.protein else
                         (f"synthetic_dna_50taxa_{P}x{pw}sites_gtrgamma "
                          "(BASELINE.json configs[2] shard shape, fused "
                          "mseg path)" if P > 1 else
                          "synthetic_dna_50taxa_1Msites_gtrgamma "
                          "(BASELINE.json configs[1])"))
                        + "; full-tree evaluateGeneric per step",
            "taxa": ntips,
            "sites_per_gpu": width,
            "partitions": P,
            "newview_ops_per_step": n_ops,
            "tipcase_counts": {"TT": tc_counts[0], "TI": tc_counts[1],
                               "II": tc_counts[2]},
            "full_tree_eval_ms": elapsed / args.steps * 1e3,
            "lnl": lnl0,
            "parallelism": f"dp{world} site-sharded, 1 RCCL all-reduce/step",
            "fast_math": bool(args.fast_math),
        },
        "roofline": roofline,
        "cpu_baseline": cpu,
    }
    print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
