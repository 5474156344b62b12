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
