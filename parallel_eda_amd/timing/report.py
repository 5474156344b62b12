"""Timing reports: critical-path extraction + echo files.

Reference surface: the timing echo files and critical-path printout
(vpr/SRC/timing/path_delay.c print_critical_path / echo system,
base/place_and_route.c:184-198). Works over the block-level timing graph.
"""
import numpy as np


def critical_paths(netlist, sta, conn_delay, k=5):
    """Extract the k most critical register-to-register paths.

    Returns a list of paths; each path is a list of
    (block, arrival_time, via_conn_delay) hops ending at a timing endpoint.
    """
    cpd, slack, crit = sta.analyze(conn_delay)
    nd = netlist.net_driver
    sp = netlist.net_sink_ptr
    ss = netlist.net_sinks
    nconn = netlist.num_conns
    net_of_conn = np.repeat(np.arange(netlist.num_nets), np.diff(sp))

    # arrival times per block: recompute forward (same as STA)
    is_seq = netlist.block_is_seq.astype(bool)
    # per-block best incoming conn (the argmax arrival feeding it)
    order = np.argsort(slack, kind="stable")
    paths = []
    seen_endpoints = set()
    for c in order[: 50 * k]:
        snk = int(ss[c])
        if not is_seq[snk]:
            continue
        if snk in seen_endpoints:
            continue
        seen_endpoints.add(snk)
        # walk backwards along most-critical in-edges
        path = [(snk, None, None)]
        cur = int(nd[net_of_conn[c]])
        path.append((cur, float(conn_delay[c]), int(net_of_conn[c])))
        guard = 0
        while not is_seq[cur] and guard < 10000:
            guard += 1
            # most critical incoming connection of cur
            in_conns = np.nonzero(ss == cur)[0]
            if len(in_conns) == 0:
                break
            best = in_conns[np.argmin(slack[in_conns])]
            cur = int(nd[net_of_conn[best]])
            path.append((cur, float(conn_delay[best]), int(net_of_conn[best])))
        paths.append({"endpoint_slack": float(slack[c]),
                      "hops": list(reversed(path))})
        if len(paths) >= k:
            break
    return cpd, paths


def write_timing_report(path, netlist, sta, conn_delay, k=5):
    cpd, paths = critical_paths(netlist, sta, conn_delay, k=k)
    names = netlist.names or [f"blk_{i}" for i in range(netlist.num_blocks)]
    with open(path, "w") as f:
        f.write(f"Critical path delay: {cpd*1e9:.4f} ns "
                f"(fmax {1e-6/cpd:.2f} MHz)\n\n")
        for i, p in enumerate(paths):
            f.write(f"Path {i}: endpoint slack {p['endpoint_slack']*1e9:.4f} ns\n")
            for blk, d, net in p["hops"]:
                if d is None:
                    f.write(f"  {names[blk]}  (endpoint)\n")
                else:
                    f.write(f"  {names[blk]}  -> net {net} "
                            f"(+{d*1e9:.4f} ns)\n")
            f.write("\n")
    return cpd


def parse_sdc(text):
    """Minimal SDC subset (reference: timing/read_sdc.c:115):
    create_clock -period P [-name N]; returns the (last) target period in
    seconds (ns units in the file, like VPR) or None."""
    clocks = parse_sdc_clocks(text)
    return list(clocks.values())[-1] if clocks else None


def parse_sdc_constraints(text):
    """Full supported SDC subset (reference read_sdc.c):
    create_clock, set_false_path -from/-to [get_clocks ...],
    set_multicycle_path N -from/-to, set_input_delay / set_output_delay
    -clock C V. Returns a dict:
      clocks: {name: period_s}
      false_paths: [(from_clk_or_None, to_clk_or_None)]
      multicycle: [(from_clk_or_None, to_clk_or_None, N)]
      input_delay / output_delay: {clock: seconds}
    None in a from/to slot means 'all clocks' (SDC wildcard)."""
    import re

    def clk_arg(rest, flag):
        m = re.search(flag + r"\s+\[\s*get_clocks\s+\{?\s*(\S+?)\s*\}?\s*\]",
                      rest)
        if m:
            return m.group(1)
        m = re.search(flag + r"\s+(?!\[)(\S+)", rest)
        return m.group(1) if m else None

    out = dict(clocks=parse_sdc_clocks(text), false_paths=[],
               multicycle=[], input_delay={}, output_delay={})
    for line in text.splitlines():
        line = line.split("#", 1)[0].strip()
        if line.startswith("set_false_path"):
            out["false_paths"].append((clk_arg(line, "-from"),
                                       clk_arg(line, "-to")))
        elif line.startswith("set_multicycle_path"):
            m = re.search(r"set_multicycle_path\s+(?:-setup\s+)?(\d+)", line)
            n = int(m.group(1)) if m else 1
            out["multicycle"].append((clk_arg(line, "-from"),
                                      clk_arg(line, "-to"), n))
        elif line.startswith("set_input_delay") or \
                line.startswith("set_output_delay"):
            c = clk_arg(line, "-clock")
            v = re.search(r"(?:-clock\s+\S+\s+|\]\s+)([0-9.eE+-]+)", line)
            key = ("input_delay" if line.startswith("set_input") else
                   "output_delay")
            if c and v:
                out[key][c] = float(v.group(1)) * 1e-9
    return out


def pair_constraints(sdc, clock_names):
    """Build the (pair_skip, pair_mult) KxK arrays for
    STA.analyze_domains from parse_sdc_constraints output. clock_names:
    ordered domain names (index = domain id)."""
    import numpy as np
    K = len(clock_names)
    idx = {c: i for i, c in enumerate(clock_names)}
    skip = np.zeros((K, K), dtype=np.uint8)
    mult = np.ones((K, K), dtype=np.float32)

    def rows(frm):
        return [idx[frm]] if frm in idx else list(range(K)) if frm is None \
            else []

    for (f, t) in sdc.get("false_paths", ()):
        for i in rows(f):
            for j in rows(t):
                skip[i, j] = 1
    for (f, t, n) in sdc.get("multicycle", ()):
        for i in rows(f):
            for j in rows(t):
                mult[i, j] = float(n)
    return skip, mult


def parse_sdc_clocks(text):
    """All create_clock constraints as {name: period_seconds}; unnamed
    clocks get the port expression or 'clk<i>'."""
    import re
    clocks = {}
    for line in text.splitlines():
        line = line.split("#", 1)[0]
        m = re.search(r"create_clock\s+(.*)", line)
        if not m:
            continue
        rest = m.group(1)
        pm = re.search(r"-period\s+([0-9.eE+-]+)", rest)
        if not pm:
            continue
        nm = re.search(r"-name\s+(\S+)", rest)
        gp = re.search(r"\[\s*get_ports\s+\{?\s*(\S+?)\s*\}?\s*\]", rest)
        name = (nm.group(1) if nm else
                (gp.group(1) if gp else f"clk{len(clocks)}"))
        clocks[name] = float(pm.group(1)) * 1e-9
    return clocks
