"""Randomized soak campaigns (CPU): fabrics, flows, front-end, oracles.

The CI suite pins a handful of seeds; this tool runs the same property
checks over arbitrary seed ranges and fabric scales — the campaigns that
caught the synth fan-in-saturation bug (seed 1014) and validated the
sink-parallel merge oracle across 124 random wide nets.

  python tools/soak.py flows --seeds 1000:1200 --het 0.4
  python tools/soak.py big   --seeds 3000:3120
  python tools/soak.py blif  --seeds 5000:5150
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import numpy as np


def seed_range(spec):
    a, b = spec.split(":")
    return range(int(a), int(b))


def soak_flows(seeds, het_prob, big=False):
    from tests.test_fuzz import random_arch
    from parallel_eda_amd.arch.archdef import ArchDef
    from parallel_eda_amd.io.synth import (synth_netlist, spec_for_arch,
                                           synth_placed_netlist)
    from parallel_eda_amd.place.placer import anneal_place
    from parallel_eda_amd.route.router import pathfinder_route
    from parallel_eda_amd.timing.sta import STA
    from parallel_eda_amd import rrgraph

    t0 = time.time()
    flows = 0
    for seed in seeds:
        rng = np.random.default_rng(seed)
        if big:
            L = int(rng.integers(1, 6))
            kw = {}
            if rng.random() < 0.5:
                kw = dict(ram_col_every=int(rng.integers(4, 9)),
                          dsp_col_every=int(rng.integers(9, 17)),
                          ram_in=int(rng.integers(8, 24)),
                          ram_out=int(rng.integers(4, 12)),
                          dsp_in=int(rng.integers(8, 24)),
                          dsp_out=int(rng.integers(4, 12)))
            arch = ArchDef(name="soak", nx=int(rng.integers(18, 33)),
                           ny=int(rng.integers(18, 33)),
                           W=2 * int(rng.integers(max(8, 2 * L), 33)), L=L,
                           fc_in=int(rng.integers(4, 12)),
                           fc_out=int(rng.integers(4, 12)),
                           clb_in=int(rng.integers(8, 24)),
                           clb_out=int(rng.integers(2, 8)),
                           io_cap=int(rng.integers(2, 6)), **kw)
        else:
            arch = random_arch(rng, het_prob=het_prob)
        g = rrgraph.build_rr_graph(arch)
        try:
            rrgraph.check_rr_graph(g, arch)
        except rrgraph.RRGraphError:
            continue
        if big:
            nl, pl = synth_placed_netlist(
                arch, fill=float(rng.uniform(0.35, 0.6)), seed=seed)
        else:
            spec = spec_for_arch(arch, fill=0.4, seed=seed)
            if spec.n_clb < 2:
                continue
            nl = synth_netlist(spec)
            if int((nl.block_type == 0).sum()) > arch.num_io_slots():
                continue
            pl = anneal_place(nl, arch, seed=seed, timing_tradeoff=0.5,
                              sta=STA(nl, arch))
        res = pathfinder_route(nl, pl, g, arch, sta=STA(nl, arch),
                               max_iters=70, incremental=(seed % 2 == 1))
        if res.success:
            ok, err = res.router.check_routed()
            assert ok, (seed, err)
        else:
            assert res.overused > 0 or res.router.unrouted_sinks() > 0, seed
        flows += 1
        if flows % 25 == 0:
            print(f"...{flows} flows at {time.time()-t0:.0f}s", flush=True)
    print(f"SOAK OK: {flows} flows in {time.time()-t0:.0f}s")


def soak_blif(seeds):
    from tests.test_fuzz import _gen_blif
    from parallel_eda_amd.io.blif import parse_blif
    from parallel_eda_amd.io.pack import pack_blif
    from parallel_eda_amd.io.net_file import check_netlist
    from parallel_eda_amd.arch.archdef import get_arch
    from parallel_eda_amd.place.placer import anneal_place
    from parallel_eda_amd.route.router import pathfinder_route
    from parallel_eda_amd.timing.sta import STA
    from parallel_eda_amd import rrgraph

    t0 = time.time()
    arch = get_arch("tiny_het")
    g = rrgraph.build_rr_graph(arch)
    routed = 0
    for seed in seeds:
        rng = np.random.default_rng(seed)
        nl, _, _ = pack_blif(parse_blif(_gen_blif(rng)), arch, n_ble=4)
        errs, _ = check_netlist(nl)
        assert not errs, (seed, errs)
        STA(nl, arch)
        bt = np.asarray(nl.block_type)
        counts = {t: int((bt == t).sum()) for t in range(4)}
        if counts[0] > arch.num_io_slots() or any(
                counts[t] > arch.num_tiles_of_type(t) for t in (1, 2, 3)):
            continue
        pl = anneal_place(nl, arch, seed=seed, timing_tradeoff=0.0)
        res = pathfinder_route(nl, pl, g, arch, sta=STA(nl, arch),
                               max_iters=60)
        if res.success:
            ok, err = res.router.check_routed()
            assert ok, (seed, err)
            routed += 1
    print(f"BLIF SOAK OK: {len(list(seeds))} designs, {routed} routed, "
          f"{time.time()-t0:.0f}s")


def _dist_worker(rank, world, port, seed, tmpdir):
    import os, pickle
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    from tests.test_fuzz import random_arch
    from parallel_eda_amd.io.synth import synth_netlist, spec_for_arch
    from parallel_eda_amd.parallel.full_flow import run_flow_dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    rng = np.random.default_rng(seed)
    arch = random_arch(rng, het_prob=0.5)
    arch.W = max(arch.W, 8 * arch.L + 8)
    spec = spec_for_arch(arch, fill=0.35, seed=seed)
    res = {"skip": True}
    if spec.n_clb >= 4:
        nl = synth_netlist(spec)
        if int((nl.block_type == 0).sum()) <= arch.num_io_slots():
            out = run_flow_dist(nl, arch, rank=rank, world_size=world,
                                seed=seed, timing_driven=True,
                                max_route_iters=70)
            res = {"skip": False, "success": out["success"],
                   "wl": out["wirelength"], "cpd": out["cpd"]}
    with open(os.path.join(tmpdir, f"r{rank}.pkl"), "wb") as f:
        pickle.dump(res, f)
    dist.barrier()
    dist.destroy_process_group()


def soak_dist(seeds, world=2):
    """Random-fabric DISTRIBUTED full flows: catches rank-divergent
    control flow (collectives must match exactly across ranks) — found
    the conditional-all-reduce bug at seed 90006."""
    import tempfile, os, pickle
    import torch.multiprocessing as mp
    t0 = time.time()
    ran = ok = 0
    for i, seed in enumerate(seeds):
        with tempfile.TemporaryDirectory() as d:
            mp.spawn(_dist_worker, args=(world, 29700 + (i % 200), seed, d),
                     nprocs=world, join=True)
            rs = [pickle.load(open(os.path.join(d, f"r{r}.pkl"), "rb"))
                  for r in range(world)]
        if rs[0]["skip"]:
            continue
        ran += 1
        for r in rs[1:]:
            assert r == rs[0], (seed, rs)
        ok += rs[0]["success"]
    print(f"DIST SOAK OK: {ran} random world-{world} flows, {ok} routed, "
          f"all rank-identical ({time.time()-t0:.0f}s)")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("mode", choices=["flows", "big", "blif", "dist"])
    ap.add_argument("--seeds", default="1000:1100")
    ap.add_argument("--het", type=float, default=0.4)
    ap.add_argument("--world", type=int, default=2)
    args = ap.parse_args()
    seeds = seed_range(args.seeds)
    if args.mode == "blif":
        soak_blif(seeds)
    elif args.mode == "dist":
        soak_dist(seeds, world=args.world)
    else:
        soak_flows(seeds, args.het, big=args.mode == "big")
    return 0


if __name__ == "__main__":
    sys.exit(main())
