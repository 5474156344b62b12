"""Command-line driver: the reference's single `Router` binary equivalent.

Mirrors vpr/SRC/main.c:310 main() + ReadOptions.c flag surface (the knobs
in t_router_opts / s_placer_opts): positional `circuit arch` plus stage and
engine flags; stages can be skipped by supplying their output files, like
VPR (--place_file to skip placement, etc.).

Usage examples:
  python -m parallel_eda_amd circuit.blif arch.xml --route_chan_width 64
  python -m parallel_eda_amd --synth tseng --engine gpu --stats_dir out/
"""
import argparse
import sys
import time

import numpy as np


def build_parser():
    p = argparse.ArgumentParser(prog="parallel_eda_amd")
    p.add_argument("circuit", nargs="?", help=".blif netlist")
    p.add_argument("arch", nargs="?", help="arch.xml architecture")
    p.add_argument("--synth", type=str, default=None,
                   help="use a named synthetic config instead of files")
    p.add_argument("--fill", type=float, default=0.6)
    p.add_argument("--seed", type=int, default=1)
    # engine selection (reference: --router_algorithm, OptionTokens.c:76-84)
    p.add_argument("--engine", choices=["cpu", "gpu"], default="cpu")
    p.add_argument("--router_algorithm",
                   choices=["timing_driven", "breadth_first"],
                   default="timing_driven")
    # placer opts (reference: s_placer_opts)
    p.add_argument("--pad_loc_file", type=str, default=None,
                   help="pin IO pads (or any named blocks) to fixed "
                        "locations: lines of 'name x y subblk'")
    p.add_argument("--place_file", type=str, default=None,
                   help="read placement instead of annealing")
    p.add_argument("--timing_tradeoff", type=float, default=0.5)
    p.add_argument("--criticality_exp", type=float, default=1.0,
                   help="criticality exponent (reference: -place_cost_exp"
                        " / criticality_exp router opt)")
    p.add_argument("--max_criticality", type=float, default=0.99)
    p.add_argument("--delay_matrix", choices=["analytic", "routed"],
                   default="analytic",
                   help="placer delay LUT: analytic Elmore or router-measured"
                        " (reference: compute_delay_lookup_tables)")
    p.add_argument("--inner_num", type=float, default=1.0)
    # router opts (reference: s_router_opts vpr_types.h:724-770)
    p.add_argument("--route_chan_width", type=int, default=None)
    p.add_argument("--max_router_iterations", type=int, default=60)
    p.add_argument("--initial_pres_fac", type=float, default=0.5)
    p.add_argument("--pres_fac_mult", type=float, default=1.3)
    p.add_argument("--acc_fac", type=float, default=1.0)
    p.add_argument("--astar_fac", type=float, default=1.2)
    p.add_argument("--bb_factor", type=int, default=3)
    p.add_argument("--min_channel_width", action="store_true",
                   help="binary search the minimum routable W")
    p.add_argument("--route_incremental", action="store_true",
                   help="CPU engine: partial rip-up + selective reroute "
                        "after a full warm-start pass (2.6x at ~1%% WL)")
    p.add_argument("--rip_up_always", action="store_true",
                   help="re-route every net every iteration instead of "
                        "congested-only selective reroute (GPU engine)")
    p.add_argument("--deterministic", action="store_true",
                   help="force the fixed wave schedule (GPU)")
    # outputs
    p.add_argument("--sdc", type=str, default=None,
                   help="SDC constraints (create_clock -period, subset)")
    p.add_argument("--draw_place", type=str, default=None,
                   help="render the placement to SVG (headless draw.c)")
    p.add_argument("--draw_route", type=str, default=None,
                   help="render the routing to SVG")
    p.add_argument("--power_report", type=str, default=None,
                   help="post-route power estimate (reference: power.c)")
    p.add_argument("--activity_file", type=str, default=None,
                   help="per-net switching activities (.act lines: "
                        "'netname activity')")
    p.add_argument("--route_file", type=str, default=None,
                   help="read this .route and ANALYZE it (occupancy + "
                        "connectivity validation, Elmore delays, STA) "
                        "instead of routing — VPR's ROUTE_NEVER flow")
    p.add_argument("--timing_report", type=str, default=None,
                   help="write a critical-path report here after routing")
    p.add_argument("--place_only", action="store_true",
                   help="stop after placement (reference: -place_only)")
    p.add_argument("--out_net", type=str, default=None,
                   help="write the packed .net netlist")
    p.add_argument("--out_place", type=str, default=None)
    p.add_argument("--out_route", type=str, default=None)
    p.add_argument("--out_verilog", type=str, default=None,
                   help="post-route structural verilog")
    p.add_argument("--out_sdf", type=str, default=None,
                   help="post-route SDF timing annotation")
    p.add_argument("--stats_dir", type=str, default=None)
    p.add_argument("--echo_routes", action="store_true",
                   help="with --stats_dir: dump the full .route traceback "
                        "after routing (reference: write_routes "
                        "routes_iter_N.txt)")
    p.add_argument("--settings", type=str, default=None,
                   help="TOML settings file: keys = CLI flag names; "
                        "explicit CLI flags win (reference: read_settings.c)")
    p.add_argument("--verbose", "-v", action="store_true")
    return p


def apply_settings(parser, args, argv):
    """Overlay a TOML settings file under explicit CLI flags (reference:
    base/read_settings.c — file supplies defaults, command line wins)."""
    import tomli
    with open(args.settings, "rb") as f:
        cfg = tomli.load(f)
    given = set()
    for tok in (argv if argv is not None else sys.argv[1:]):
        if tok.startswith("--"):
            given.add(tok[2:].split("=", 1)[0])
    for k, v in cfg.items():
        if k in given:
            continue  # explicit flag wins
        if not hasattr(args, k):
            raise SystemExit(f"settings: unknown option {k!r}")
        setattr(args, k, v)
    return args


def main(argv=None):
    parser = build_parser()
    args = parser.parse_args(argv)
    if args.settings:
        args = apply_settings(parser, args, argv)
    from .arch.archdef import get_arch
    from .arch.xml_parser import parse_arch_xml, size_grid_for_netlist
    from .io.synth import synth_netlist, spec_for_arch
    from .io.blif import read_blif
    from .io.pack import pack_blif
    from .io.place_file import write_place, read_place
    from .io.route_file import write_route
    from .place.placer import anneal_place
    from .route.router import pathfinder_route, net_rr_terminals
    from .timing.sta import STA
    from .utils.stats import StatsWriter, routing_stats
    from . import rrgraph
    from .flow import min_channel_width

    t_start = time.perf_counter()
    if args.synth:
        arch = get_arch(args.synth)
        netlist = synth_netlist(spec_for_arch(arch, fill=args.fill,
                                              seed=args.seed))
        print(f"synthetic netlist '{args.synth}': {netlist.num_blocks} blocks,"
              f" {netlist.num_nets} nets")
    elif args.circuit and args.circuit.endswith(".net"):
        # packed netlist input: skip the packer (reference: VPR reads
        # circuit.net directly when packing already happened)
        from .io.net_file import read_net
        if not args.arch:
            print("error: need arch.xml with a .net input", file=sys.stderr)
            return 2
        arch = parse_arch_xml(args.arch, W=args.route_chan_width or 64)
        netlist = read_net(args.circuit)
        size_grid_for_netlist(netlist, arch)
        print(f"read {args.circuit}: {netlist.num_blocks} blocks, "
              f"{netlist.num_nets} nets on {arch.nx}x{arch.ny} grid")
    else:
        if not args.circuit or not args.arch:
            print("error: need circuit.blif + arch.xml (or --synth NAME)",
                  file=sys.stderr)
            return 2
        model = read_blif(args.circuit)
        arch = parse_arch_xml(args.arch, W=args.route_chan_width or 64)
        netlist, _, _ = pack_blif(model, arch, n_ble=arch.clb_n_ble)
        size_grid_for_netlist(netlist, arch)
        print(f"read {args.circuit}: {len(model.prims)} primitives -> "
              f"{netlist.num_blocks} blocks on {arch.nx}x{arch.ny} grid")
    if args.route_chan_width:
        arch.W = args.route_chan_width + (args.route_chan_width % 2)

    timing = args.router_algorithm == "timing_driven"
    sta = STA(netlist, arch) if timing else None
    sdc_clocks = {}
    sdc_all = None
    if args.sdc:
        from .timing.report import parse_sdc_constraints
        with open(args.sdc) as f:
            sdc_all = parse_sdc_constraints(f.read())
        sdc_clocks = sdc_all["clocks"]
        for nm, p_ in sdc_clocks.items():
            print(f"SDC: clock '{nm}' period {p_*1e9:.3f} ns")
        for (f_, t_) in sdc_all["false_paths"]:
            print(f"SDC: false path {f_ or '*'} -> {t_ or '*'}")
        for (f_, t_, n_) in sdc_all["multicycle"]:
            print(f"SDC: multicycle {n_} {f_ or '*'} -> {t_ or '*'}")

    # ---- placement ----
    t0 = time.perf_counter()
    if args.place_file:
        placement = read_place(args.place_file, netlist)
        print(f"read placement from {args.place_file}")
    else:
        fixed = None
        if args.pad_loc_file:
            from .io.place_file import read_pad_loc
            fixed = read_pad_loc(args.pad_loc_file, netlist)
            print(f"pinned {len(fixed[0])} blocks from {args.pad_loc_file}")
        macros = getattr(netlist, "macros", None) or None
        if macros and args.engine != "cpu":
            print(f"note: {len(macros)} carry chains ignored on the "
                  f"{args.engine} engine (macros: CPU placer only)")
            macros = None
        elif macros:
            print(f"carry chains: {len(macros)} macros "
                  f"({sum(len(m) for m in macros)} blocks)")
        placement = anneal_place(
            netlist, arch, seed=args.seed,
            timing_tradeoff=args.timing_tradeoff if timing else 0.0,
            inner_num=args.inner_num, sta=sta, verbose=args.verbose,
            engine=args.engine, delay_matrix=args.delay_matrix, fixed=fixed,
            crit_exp=args.criticality_exp, macros=macros)
        print(f"placement: bb_cost={placement.bb_cost:.1f} "
              f"({time.perf_counter()-t0:.2f}s)")
    if args.out_net:
        from .io.net_file import write_net
        write_net(args.out_net, netlist)
        print(f"wrote {args.out_net}")
    if args.out_place:
        write_place(args.out_place, placement, netlist, arch)
        print(f"wrote {args.out_place}")
    if args.draw_place:
        from .utils.draw import write_placement_svg
        write_placement_svg(args.draw_place, placement, netlist, arch)
        print(f"wrote {args.draw_place}")
    if args.place_only:
        print(f"entire flow took {time.perf_counter()-t_start:.2f}s")
        return 0

    # ---- analysis-only: re-read an existing routing (reference: VPR's
    # ROUTE_NEVER / --route_file analysis flow) ----
    if args.route_file:
        from .io.route_file import read_route, tree_elmore_delays
        from .route.router import ConnMap
        g = rrgraph.build_rr_graph(arch)
        names_r, trees_r = read_route(args.route_file, g, arch)
        net_ids, src_rr, sink_ptr, sink_rr, conn_index = net_rr_terminals(
            netlist, placement, g, arch)
        if len(trees_r) != len(net_ids):
            print(f"route file has {len(trees_r)} nets, netlist expects "
                  f"{len(net_ids)}")
            return 1
        # occupancy + connectivity validation (check_route.c semantics)
        occ = np.zeros(g.num_nodes, dtype=np.int32)
        cap = np.asarray(g.capacity)
        sink_delays = np.zeros(len(sink_rr), dtype=np.float32)
        for k, (nodes_k, parents_k) in enumerate(trees_r):
            np.add.at(occ, nodes_k, 1)
            if len(nodes_k) == 0 or nodes_k[0] != src_rr[k]:
                print(f"net {k}: traceback does not start at its SOURCE")
                return 1
            d = tree_elmore_delays(g, nodes_k, parents_k)
            pos = {int(v): i for i, v in enumerate(nodes_k)}
            for s in range(sink_ptr[k], sink_ptr[k + 1]):
                v = int(sink_rr[s])
                if v not in pos:
                    print(f"net {k}: sink rr node {v} not in traceback")
                    return 1
                sink_delays[s] = d[pos[v]]
        over = int((occ > cap).sum())
        print(f"read {args.route_file}: {len(trees_r)} nets, "
              f"overused nodes: {over}")
        if over:
            return 1
        if sta is not None:
            cmap = ConnMap(conn_index, sink_ptr, netlist.num_conns,
                           len(sink_rr))
            cd = cmap.conn_delays(sink_delays,
                                  fill=float(arch.T_opin + arch.T_ipin))
            cpd, slack, crit = sta.analyze(cd)
            print(f"analysis: crit_path={cpd*1e9:.3f}ns")
            if args.timing_report:
                from .timing.report import write_timing_report
                write_timing_report(args.timing_report, netlist, sta, cd)
                print(f"wrote {args.timing_report}")
        print(f"entire flow took {time.perf_counter()-t_start:.2f}s")
        return 0

    # ---- routing ----
    if args.min_channel_width:
        w, res = min_channel_width(netlist, placement, arch,
                                   engine=args.engine, verbose=args.verbose)
        print(f"minimum channel width: {w}")
    else:
        t0 = time.perf_counter()
        g = rrgraph.build_rr_graph(arch)
        print(f"rr graph: {g.num_nodes} nodes, {g.num_edges} edges "
              f"({time.perf_counter()-t0:.2f}s)")
        t0 = time.perf_counter()
        sw = StatsWriter(args.stats_dir) if args.stats_dir else None
        res = pathfinder_route(
            netlist, placement, g, arch, sta=sta,
            max_iters=args.max_router_iterations,
            pres_fac_init=args.initial_pres_fac,
            pres_fac_mult=args.pres_fac_mult, acc_fac=args.acc_fac,
            astar_fac=args.astar_fac, verbose=args.verbose,
            engine=args.engine, rip_up_always=args.rip_up_always,
            deterministic=args.deterministic, bb_factor=args.bb_factor,
            crit_exp=args.criticality_exp,
            max_criticality=args.max_criticality,
            incremental=args.route_incremental)
        rt = time.perf_counter() - t0
        if not res.success:
            print(f"ROUTING FAILED: {res.overused} overused nodes after "
                  f"{res.iterations} iterations")
            return 1
        print(f"routed in {res.iterations} iterations ({rt:.2f}s): "
              f"wirelength={res.wirelength} "
              f"crit_path={res.crit_path_delay*1e9:.3f}ns")
        net_ids, *_ = net_rr_terminals(netlist, placement, g, arch)
        st = routing_stats(g, arch, net_ids, lambda k: res.router.tree(k))
        from .utils.stats import routing_serial_num, mem_usage_mb
        serial = routing_serial_num(net_ids, lambda k: res.router.tree(k))
        print(f"  segments={st['total_segments']} bends={st['total_bends']} "
              f"avg_wl/net={st['avg_wirelength_per_net']:.1f}")
        print(f"  serial_num {serial}  host_mem {mem_usage_mb():.0f} MiB")
        if sw:
            for h in res.stats["history"]:
                sw.iteration(h["iter"], h["overused"], cpd=h.get("cpd", 0.0))
            sw.final(res.success, res.wirelength, res.crit_path_delay, st)
        if args.out_route:
            write_route(args.out_route, g, arch, net_ids,
                        lambda k: res.router.tree(k), netlist=netlist)
            print(f"wrote {args.out_route}")
        if args.echo_routes and args.stats_dir:
            import os as _os
            ep = _os.path.join(args.stats_dir,
                               f"routes_iter_{res.iterations}.txt")
            write_route(ep, g, arch, net_ids,
                        lambda k: res.router.tree(k), netlist=netlist)
            print(f"wrote {ep}")
        if sdc_clocks and getattr(netlist, "block_clock", None) is not None \
                and len(sdc_clocks) >= 1 and sta is not None:
            # multi-clock analysis against the SDC constraints
            from .route.router import ConnMap
            _, src_rr3, sink_ptr3, sink_rr3, ci3 = net_rr_terminals(
                netlist, placement, g, arch)
            cmap3 = ConnMap(ci3, sink_ptr3, netlist.num_conns, len(sink_rr3))
            sd3 = np.asarray(res.router.sink_delays()) \
                if hasattr(res.router, "sink_delays") \
                else res.router.t_sink_delay.cpu().numpy()
            cd3 = cmap3.conn_delays(sd3)
            cnames = getattr(netlist, "clock_names", []) or []
            periods = [sdc_clocks.get(c, list(sdc_clocks.values())[0])
                       for c in cnames] or list(sdc_clocks.values())[:1]
            bc = getattr(netlist, "block_clock")
            ps = pm = None
            if sdc_all and (sdc_all["false_paths"] or sdc_all["multicycle"]):
                from .timing.report import pair_constraints
                ps, pm = pair_constraints(sdc_all, cnames)
            wp, sl, cr = sta.analyze_domains(cd3, bc, periods,
                                             pair_skip=ps, pair_mult=pm)
            ok_sdc = all(wp <= p_ * (1 + 1e-6) for p_ in periods[:1])
            print(f"SDC analysis: worst achieved period {wp*1e9:.3f} ns "
                  f"across {max(1, len(periods))} clock domain(s)")
        if args.out_verilog or args.out_sdf:
            from .io.verilog import write_verilog, write_sdf
            if args.out_verilog:
                write_verilog(args.out_verilog, netlist)
                print(f"wrote {args.out_verilog}")
            if args.out_sdf:
                from .route.router import ConnMap
                _, _s4, sink_ptr4, sink_rr4, ci4 = net_rr_terminals(
                    netlist, placement, g, arch)
                cmap4 = ConnMap(ci4, sink_ptr4, netlist.num_conns,
                                len(sink_rr4))
                sd4 = np.asarray(res.router.sink_delays()) \
                    if hasattr(res.router, "sink_delays") \
                    else res.router.t_sink_delay.cpu().numpy()
                write_sdf(args.out_sdf, netlist, arch, cmap4.conn_delays(sd4))
                print(f"wrote {args.out_sdf}")
        if args.draw_route and hasattr(res.router, "tree"):
            from .utils.draw import write_routing_svg
            write_routing_svg(args.draw_route, g, arch, res.router,
                              net_ids=net_ids)
            print(f"wrote {args.draw_route}")
        if args.power_report and hasattr(res.router, "tree"):
            from .utils.power import (estimate_power, read_activity_file,
                                      write_power_report)
            act = (read_activity_file(args.activity_file, netlist)
                   if args.activity_file else None)
            pw = estimate_power(netlist, arch, g, res.router,
                                activities=act)
            write_power_report(args.power_report, pw)
            print(f"wrote {args.power_report} "
                  f"(total {pw['total_W']*1e3:.3f} mW)")
        if args.timing_report and sta is not None:
            from .route.router import ConnMap
            from .timing.report import write_timing_report
            _, src_rr2, sink_ptr2, sink_rr2, ci2 = net_rr_terminals(
                netlist, placement, g, arch)
            cmap = ConnMap(ci2, sink_ptr2, netlist.num_conns, len(sink_rr2))
            sd = np.asarray(res.router.sink_delays()) \
                if hasattr(res.router, "sink_delays") \
                else res.router.t_sink_delay.cpu().numpy()
            conn_delay = cmap.conn_delays(sd)
            write_timing_report(args.timing_report, netlist, sta, conn_delay)
            print(f"wrote {args.timing_report}")
    print(f"entire flow took {time.perf_counter()-t_start:.2f}s")
    return 0


if __name__ == "__main__":
    sys.exit(main())
