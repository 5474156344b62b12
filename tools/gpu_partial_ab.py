"""GPU A/B: partial rip-up (P.partial) vs full-rip selective reroute.

Round-2 validation for the flag-gated kernel path (see
csrc/hip/router_kernel.hip partial branches; CPU oracle
SerialRouter::route_net_incremental, measured 2.6x at ~1% WL on CPU).
Runs on an MI355X via gpurun:

  gpurun --timeout 900 -- 'python tools/gpu_partial_ab.py tseng 0.6;
                           python tools/gpu_partial_ab.py LU32PEEng'
"""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import numpy as np


def main():
    name = sys.argv[1] if len(sys.argv) > 1 else "tseng"
    fill = float(sys.argv[2]) if len(sys.argv) > 2 else 0.55
    from parallel_eda_amd.arch.archdef import get_arch
    from parallel_eda_amd.io.synth import (synth_netlist, spec_for_arch,
                                           synth_placed_netlist)
    from parallel_eda_amd.place.placer import anneal_place
    from parallel_eda_amd.route.gpu_router import pathfinder_route_gpu
    from parallel_eda_amd.timing.sta import STA
    from parallel_eda_amd import rrgraph

    arch = get_arch(name)
    if arch.nx * arch.ny > 40 * 40:
        nl, pl = synth_placed_netlist(arch, fill=fill, seed=7)
    else:
        nl = synth_netlist(spec_for_arch(arch, fill=fill, seed=7))
        pl = anneal_place(nl, arch, seed=7, timing_tradeoff=0.0)
    g = rrgraph.build_rr_graph(arch)
    out = {}
    for label, inc in (("full", False), ("partial", True)):
        sta = STA(nl, arch)
        t0 = time.time()
        res = pathfinder_route_gpu(nl, pl, g, arch, sta=sta, max_iters=60,
                                   incremental=inc)
        t = time.time() - t0
        ok, err = (True, "")
        if res.success:
            ok = res.router.check_occ_recount()
            err = "occ recount mismatch" if not ok else ""
        out[label] = (res, t, ok, err)
        print(f"{name} {label}: ok={res.success} it={res.iterations} "
              f"wl={res.wirelength} cpd={res.crit_path_delay*1e9:.3f}ns "
              f"t={t:.2f}s occ_ok={ok} {err}", flush=True)
    rf, tf, okf, _ = out["full"]
    rp, tp, okp, errp = out["partial"]
    assert rp.success and okp, f"partial-rip failed: {errp}"
    assert rp.wirelength <= rf.wirelength * 1.10, "partial WL blowup"
    print(f"A/B: time {tf:.2f}s -> {tp:.2f}s "
          f"({tf/max(tp,1e-9):.2f}x), wl {rf.wirelength} -> {rp.wirelength} "
          f"({100*(rp.wirelength-rf.wirelength)/rf.wirelength:+.1f}%)")
    return 0


if __name__ == "__main__":
    sys.exit(main())
