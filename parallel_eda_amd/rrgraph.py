"""Routing-resource graph: build (C++ host) + validation + GPU upload.

Reference semantics: vpr/SRC/route/rr_graph.c:385 build_rr_graph and
check_rr_graph.c:21 (validator). The graph is flat SoA + CSR; `to_torch`
moves the arrays to a device untransformed (HBM-resident rr graph).
"""
import numpy as np

from .arch.archdef import (ArchDef, RR_SOURCE, RR_SINK, RR_OPIN, RR_IPIN,
                           RR_CHANX, RR_CHANY)
from . import ops


class RRGraphError(Exception):
    pass


def build_rr_graph(arch: ArchDef):
    """Build the rr graph for `arch`. Returns the C++ RRGraph handle."""
    cpu = ops.cpu()
    params = arch.to_dict()
    params["base_cost"] = arch.base_costs()
    return cpu.build_rr_graph(params)


def check_rr_graph(g, arch: ArchDef):
    """Structural validator (reference: check_rr_graph.c:21).

    Raises RRGraphError on the first violated invariant.
    """
    ty = np.asarray(g.type)
    xlow, ylow = np.asarray(g.xlow), np.asarray(g.ylow)
    xhigh, yhigh = np.asarray(g.xhigh), np.asarray(g.yhigh)
    cap = np.asarray(g.capacity)
    row_ptr, dst, sw = np.asarray(g.row_ptr), np.asarray(g.edge_dst), np.asarray(g.edge_sw)
    n = g.num_nodes
    if len(ty) != n or len(row_ptr) != n + 1:
        raise RRGraphError("array length mismatch")
    if row_ptr[-1] != g.num_edges or len(dst) != g.num_edges:
        raise RRGraphError("edge count mismatch")
    if (dst < 0).any() or (dst >= n).any():
        raise RRGraphError("edge destination out of range")
    if (xlow > xhigh).any() or (ylow > yhigh).any():
        raise RRGraphError("inverted bounding box")
    if (cap < 1).any():
        raise RRGraphError("node capacity < 1")
    # type-legal edges (SOURCE->OPIN, OPIN->CHAN, CHAN->CHAN|IPIN, IPIN->SINK)
    src_of_edge = np.repeat(np.arange(n), np.diff(row_ptr))
    st, dt = ty[src_of_edge], ty[dst]
    legal = (
        ((st == RR_SOURCE) & (dt == RR_OPIN)) |
        ((st == RR_OPIN) & ((dt == RR_CHANX) | (dt == RR_CHANY))) |
        (((st == RR_CHANX) | (st == RR_CHANY)) &
         ((dt == RR_CHANX) | (dt == RR_CHANY) | (dt == RR_IPIN))) |
        ((st == RR_IPIN) & (dt == RR_SINK))
    )
    if not legal.all():
        bad = np.nonzero(~legal)[0][0]
        raise RRGraphError(
            f"illegal edge type {st[bad]}->{dt[bad]} at edge {bad}")
    # SINK nodes have no out-edges; SOURCE nodes have >=1
    deg = np.diff(row_ptr)
    if (deg[ty == RR_SINK] != 0).any():
        raise RRGraphError("SINK with out-edges")
    if (deg[ty == RR_SOURCE] == 0).any():
        raise RRGraphError("SOURCE with no OPINs")
    # every logic/IO tile has source+sink
    ts, tk = np.asarray(g.tile_source), np.asarray(g.tile_sink)
    nx, ny = arch.nx, arch.ny
    gy = ny + 2
    for x in range(1, nx + 1):
        for y in range(1, ny + 1):
            if ts[x * gy + y] < 0 or tk[x * gy + y] < 0:
                raise RRGraphError(f"CLB tile ({x},{y}) missing source/sink")
    # wires: every CHAN node reachable-from-something (has in-edges) —
    # check via in-degree
    indeg = np.zeros(n, dtype=np.int64)
    np.add.at(indeg, dst, 1)
    chan = (ty == RR_CHANX) | (ty == RR_CHANY)
    frac_orphan = float((indeg[chan] == 0).mean()) if chan.any() else 0.0
    if frac_orphan > 0.02:
        raise RRGraphError(f"{frac_orphan:.1%} of wires undrivable")
    if (indeg[ty == RR_SINK] == 0).any():
        raise RRGraphError("SINK with no in-edges")
    if (indeg[ty == RR_IPIN] == 0).any():
        raise RRGraphError("IPIN with no in-edges")
    # sampled source->all-sinks connectivity (scipy BFS; catches fabrics
    # whose W/L ratio leaves the Fs=3 switch digraph reducible — seen at
    # W=8,L=4 where a stagger class has a single track)
    check_connectivity(g, arch, samples=3)
    (sw)
    return True


def check_connectivity(g, arch, samples=3):
    from scipy.sparse import csr_matrix
    from scipy.sparse.csgraph import breadth_first_order
    n = g.num_nodes
    row_ptr = np.asarray(g.row_ptr)
    dst = np.asarray(g.edge_dst)
    ty = np.asarray(g.type)
    m = csr_matrix((np.ones(len(dst), dtype=np.int8), dst, row_ptr),
                   shape=(n, n))
    ts = np.asarray(g.tile_source)
    gy = arch.ny + 2
    sinks = np.nonzero(ty == RR_SINK)[0]
    probes = [(1, 1), (arch.nx, arch.ny),
              (max(1, arch.nx // 2), max(1, arch.ny // 2))][:samples]
    for (x, y) in probes:
        src = ts[x * gy + y]
        if src < 0:
            continue
        order = breadth_first_order(m, int(src), return_predecessors=False)
        seen = np.zeros(n, dtype=bool)
        seen[order] = True
        frac = float(seen[sinks].mean())
        if frac < 1.0:
            raise RRGraphError(
                f"fabric not fully connected: SOURCE({x},{y}) reaches "
                f"{frac:.1%} of sinks (W/L ratio too small for Fs=3?)")
    return True


def graph_summary(g):
    ty = np.asarray(g.type)
    counts = {name: int((ty == i).sum()) for i, name in enumerate(
        ["SOURCE", "SINK", "OPIN", "IPIN", "CHANX", "CHANY"])}
    return {
        "num_nodes": g.num_nodes,
        "num_edges": int(g.num_edges),
        "degree_max": g.degree_max,
        **counts,
    }


def to_torch(g, device):
    """Upload the SoA arrays to `device` as torch tensors (dict)."""
    import torch
    out = {}
    for name in ("type", "xlow", "ylow", "xhigh", "yhigh", "ptc", "capacity",
                 "node_R", "node_C", "row_ptr", "edge_dst", "edge_sw",
                 "tile_source", "tile_sink", "sw_R", "sw_Cin", "sw_Tdel",
                 "base_cost"):
        arr = np.asarray(getattr(g, name))
        out[name] = torch.from_numpy(arr.copy()).to(device)
    out["num_nodes"] = g.num_nodes
    out["num_edges"] = int(g.num_edges)
    out["degree_max"] = int(g.degree_max)
    return out
