"""Place+route WALL-CLOCK flow benchmark (BASELINE's headline metric).

bench.py's driver contract measures steady-state per-iteration routing
throughput; THIS tool measures the end-to-end metric BASELINE.json names:
place + route wall-clock to a feasible, timing-clean result at a named
config — time-to-feasible with selective reroute, not rip-all steps.
Setup (netlist synthesis, rr build, upload) is reported separately and
excluded from the headline number, matching the reference's stats
(route time vs total time in final_stats).

GPU run (round 2):
  gpurun --timeout 1800 -- 'python tools/bench_flow.py LU32PEEng > gpurun_out/flow_lu32.json'
CPU smoke (any box):
  python tools/bench_flow.py tseng --placer cpu --router cpu
"""
import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import numpy as np


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("config", nargs="?", default="LU32PEEng")
    ap.add_argument("--fill", type=float, default=0.55)
    ap.add_argument("--seed", type=int, default=7)
    ap.add_argument("--placer", choices=["gpu", "cpu", "synthetic"],
                    default=None,
                    help="synthetic = placement generated with the "
                         "netlist (big-config default; real benchmarks "
                         "start from a quality placement)")
    ap.add_argument("--router", choices=["gpu", "cpu"], default=None)
    ap.add_argument("--max_iters", type=int, default=60)
    ap.add_argument("--incremental", action="store_true",
                    help="partial rip-up on selective iterations")
    args = ap.parse_args()

    from parallel_eda_amd.arch.archdef import get_arch
    from parallel_eda_amd.io.synth import (synth_netlist, spec_for_arch,
                                           synth_placed_netlist)
    from parallel_eda_amd.place.placer import anneal_place
    from parallel_eda_amd.route.router import pathfinder_route
    from parallel_eda_amd.timing.sta import STA
    from parallel_eda_amd import rrgraph

    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    big = False
    arch = get_arch(args.config)
    big = arch.nx * arch.ny > 64 * 64
    placer = args.placer or ("synthetic" if big else
                             ("gpu" if has_gpu else "cpu"))
    router = args.router or ("gpu" if has_gpu else "cpu")

    t0 = time.perf_counter()
    if placer == "synthetic":
        nl, pl = synth_placed_netlist(arch, fill=args.fill, seed=args.seed)
        t_synth = time.perf_counter() - t0
        t_place = 0.0
    else:
        nl = synth_netlist(spec_for_arch(arch, fill=args.fill,
                                         seed=args.seed))
        t_synth = time.perf_counter() - t0
        sta_p = STA(nl, arch)
        t1 = time.perf_counter()
        pl = anneal_place(nl, arch, seed=args.seed, timing_tradeoff=0.5,
                          sta=sta_p, engine=placer)
        t_place = time.perf_counter() - t1
    t1 = time.perf_counter()
    g = rrgraph.build_rr_graph(arch)
    t_rr = time.perf_counter() - t1

    sta = STA(nl, arch)
    t1 = time.perf_counter()
    res = pathfinder_route(nl, pl, g, arch, sta=sta,
                           max_iters=args.max_iters, engine=router,
                           incremental=args.incremental)
    t_route = time.perf_counter() - t1

    wall = t_place + t_route   # the headline: place+route wall-clock
    print(json.dumps({
        "metric": "place+route wall-clock to feasible (s)",
        "value": round(wall, 3),
        "higher_is_better": False,
        "config": {"model": args.config, "grid": f"{arch.nx}x{arch.ny}",
                   "W": arch.W, "nets": int(nl.num_nets),
                   "sinks": int(nl.num_conns), "placer": placer,
                   "router": router, "incremental": args.incremental},
        "success": bool(res.success),
        "route_iterations": int(res.iterations),
        "wirelength": int(res.wirelength),
        "crit_path_ns": round(res.crit_path_delay * 1e9, 4),
        "phase_s": {"netlist_synth": round(t_synth, 3),
                    "place": round(t_place, 3),
                    "rr_build": round(t_rr, 3),
                    "route": round(t_route, 3)},
        "data": "synthetic (no network)",
    }))
    return 0 if res.success else 1


if __name__ == "__main__":
    sys.exit(main())
