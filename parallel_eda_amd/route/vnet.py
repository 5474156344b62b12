"""Virtual nets: split wide (high-fanout) nets into per-sink-cluster
routing units.

Reference: create_virtual_nets partitioning_multi_sink_delta_stepping_
route.cxx:3465 (+ add_new_merge_vnet:4452) — a net with many sinks is
split into spatially-clustered sink groups so independent workers (there
TBB tasks, here GPU workgroups in round 2) can route one net's sinks
concurrently; small clusters are merged. Sinks in the same cluster still
share the incremental tree, so spatial clustering bounds the wirelength
loss of giving up cross-cluster tree reuse (measured: tools/vnet_sim.py).
"""
import numpy as np


def cluster_sinks(xs, ys, max_cluster):
    """Greedy spatial clustering: recursively split the sink set along
    the wider bb axis at the median until every cluster has at most
    max_cluster sinks. Deterministic; returns a list of index arrays."""
    idx = np.arange(len(xs))
    out = []
    stack = [idx]
    while stack:
        cur = stack.pop()
        if len(cur) <= max_cluster:
            out.append(np.sort(cur))
            continue
        w = xs[cur].max() - xs[cur].min()
        h = ys[cur].max() - ys[cur].min()
        key = xs[cur] if w >= h else ys[cur]
        order = cur[np.argsort(key, kind="stable")]
        mid = len(order) // 2
        stack.append(order[mid:])
        stack.append(order[:mid])
    out.sort(key=lambda a: int(a[0]))
    return out


def split_virtual_nets(sink_ptr, sink_rr, xl, yl, max_sinks=8):
    """Split every net with more than max_sinks sinks into spatial sink
    clusters. Returns (vnet_parent, vnet_sink_ptr, vnet_sinks):
    vnet i routes sinks vnet_sinks[vnet_sink_ptr[i]:vnet_sink_ptr[i+1]]
    (indices into sink_rr's flat array) and belongs to original net
    vnet_parent[i]. Nets at or under the threshold become one vnet."""
    sink_ptr = np.asarray(sink_ptr)
    n_nets = len(sink_ptr) - 1
    parents, ptr, flat = [], [0], []
    for n in range(n_nets):
        lo, hi = int(sink_ptr[n]), int(sink_ptr[n + 1])
        conns = np.arange(lo, hi)
        if hi - lo <= max_sinks:
            groups = [np.arange(hi - lo)]
        else:
            sx = xl[sink_rr[lo:hi]].astype(np.int32)
            sy = yl[sink_rr[lo:hi]].astype(np.int32)
            groups = cluster_sinks(sx, sy, max_sinks)
        for gidx in groups:
            parents.append(n)
            flat.extend(conns[gidx].tolist())
            ptr.append(len(flat))
    return (np.asarray(parents, dtype=np.int32),
            np.asarray(ptr, dtype=np.int64),
            np.asarray(flat, dtype=np.int64))
