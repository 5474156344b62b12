"""VPR-compatible .route file writer.

Format (reference: vpr/SRC/route/route_common.c:1322 print_route):
  Array size: <nx> x <ny> logic blocks.
  Routing:
  Net <k> (<name>)
  SOURCE (x,y)  Class: <c>
  OPIN (x,y)  Pin: <p>
  CHANX (x,y) to (x2,y2)  Track: <t>
  IPIN (x,y)  Pin: <p>
  SINK (x,y)  Class: <c>
"""
import numpy as np

from ..arch.archdef import RR_TYPE_NAMES, RR_SOURCE, RR_SINK, RR_OPIN, RR_IPIN


def _node_line(g, v, ty_arr, xl, yl, xh, yh, ptc):
    t = RR_TYPE_NAMES[ty_arr[v]]
    if ty_arr[v] in (RR_SOURCE, RR_SINK):
        return f"{t} ({xl[v]},{yl[v]})  Class: {ptc[v]}  "
    if ty_arr[v] in (RR_OPIN, RR_IPIN):
        return f"{t} ({xl[v]},{yl[v]})  Pin: {ptc[v]}  "
    if xl[v] == xh[v] and yl[v] == yh[v]:
        return f"{t} ({xl[v]},{yl[v]})  Track: {ptc[v]}  "
    return (f"{t} ({xl[v]},{yl[v]}) to ({xh[v]},{yh[v]})  Track: {ptc[v]}  ")


def write_route(path, g, arch, net_ids, trees, netlist=None):
    """trees: callable inet -> (nodes, parents, sws, delays) in tree order.

    The traceback is printed as a preorder walk: each tree node is printed
    after its parent; branch points re-print the parent node (VPR's
    traceback also revisits branch points)."""
    ty_arr = np.asarray(g.type)
    xl = np.asarray(g.xlow); yl = np.asarray(g.ylow)
    xh = np.asarray(g.xhigh); yh = np.asarray(g.yhigh)
    ptc = np.asarray(g.ptc)
    with open(path, "w") as f:
        f.write(f"Array size: {arch.nx} x {arch.ny} logic blocks.\n\n")
        f.write("Routing:\n")
        for k, inet in enumerate(net_ids):
            name = (netlist.names[netlist.net_driver[inet]]
                    if netlist is not None and netlist.names else f"net_{inet}")
            f.write(f"\nNet {k} ({name})\n\n")
            nodes, parents, sws, delays = trees(k)
            prev = -1
            for i in range(len(nodes)):
                if parents[i] != prev and parents[i] >= 0:
                    # branch: re-print the attach node
                    f.write(_node_line(g, nodes[parents[i]], ty_arr, xl, yl,
                                       xh, yh, ptc) + "\n")
                f.write(_node_line(g, nodes[i], ty_arr, xl, yl, xh, yh, ptc)
                        + "\n")
                prev = i


def read_route(path, g, arch):
    """Read a .route traceback back into per-net trees (reference:
    route/read_route-style analysis flow — VPR re-reads its own routing
    for timing analysis without re-routing).

    Returns (net_names, trees) where trees[k] = (nodes, parents) numpy
    arrays in tree order (parents index into the net's own arrays).
    Node identity is recovered from (type, xlow, ylow, ptc), which is
    unique in this fabric."""
    ty_arr = np.asarray(g.type)
    xl = np.asarray(g.xlow); yl = np.asarray(g.ylow)
    ptc = np.asarray(g.ptc)
    lookup = {}
    for v in range(g.num_nodes):
        lookup[(int(ty_arr[v]), int(xl[v]), int(yl[v]), int(ptc[v]))] = v
    tname_id = {n: i for i, n in enumerate(RR_TYPE_NAMES)}

    import re
    line_re = re.compile(
        r"^(\w+) \((\d+),(\d+)\)(?: to \(\d+,\d+\))?\s+"
        r"(?:Class|Pin|Track): (\d+)")
    net_names = []
    trees = []
    nodes = []
    parents = []
    index_of = {}
    prev_idx = -1

    def flush():
        if net_names and len(net_names) > len(trees):
            trees.append((np.asarray(nodes, dtype=np.int32),
                          np.asarray(parents, dtype=np.int32)))

    with open(path) as f:
        for line in f:
            line = line.rstrip("\n")
            m = re.match(r"^Net (\d+) \((.*)\)$", line)
            if m:
                flush()
                net_names.append(m.group(2))
                nodes = []; parents = []; index_of = {}; prev_idx = -1
                continue
            m = line_re.match(line)
            if not m:
                continue
            key = (tname_id[m.group(1)], int(m.group(2)), int(m.group(3)),
                   int(m.group(4)))
            if key not in lookup:
                raise ValueError(f"route references unknown rr node {key}")
            v = lookup[key]
            if v in index_of:
                prev_idx = index_of[v]     # branch attach re-print
                continue
            nodes.append(v)
            parents.append(prev_idx)
            index_of[v] = len(nodes) - 1
            prev_idx = len(nodes) - 1
    flush()
    return net_names, trees


def tree_elmore_delays(g, nodes, parents):
    """Per-tree-node source->node delay via the same per-hop Elmore the
    routers use: Tdel(sw) + C(v) * (R(sw) + R(v)/2)."""
    row_ptr = np.asarray(g.row_ptr); edge_dst = np.asarray(g.edge_dst)
    edge_sw = np.asarray(g.edge_sw)
    sw_R = np.asarray(g.sw_R); sw_T = np.asarray(g.sw_Tdel)
    R = np.asarray(g.node_R); C = np.asarray(g.node_C)
    d = np.zeros(len(nodes), dtype=np.float64)
    for i in range(1, len(nodes)):
        u = nodes[i]; pu = nodes[parents[i]]
        sw = 0
        for e in range(row_ptr[pu], row_ptr[pu + 1]):
            if edge_dst[e] == u:
                sw = int(edge_sw[e])
                break
        d[i] = d[parents[i]] + sw_T[sw] + C[u] * (sw_R[sw] + 0.5 * R[u])
    return d
