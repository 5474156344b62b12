"""Bit-packed routed-path codec.

Reference: parallel_route/path_codec.h — routed paths are broadcast as
edge-index sequences packed at ``w = ceil(log2(max_out_degree + 1))`` bits
per hop, with the all-ones code ``2**w - 1`` as the trailer
(mpi_route_load_balanced_...cxx:454-456 sizes w from the graph). An edge
index (position within the source node's adjacency row) is much smaller
than a node id (our CSR degree is < 64 vs 12.7M nodes at bitcoin scale),
so a path costs ~6 bits/hop instead of 32 — the same trick applies to any
compact route exchange or on-disk traceback (write_routes_packed below).

Trees are encoded as their root-to-sink paths sharing the codec; branch
structure is reconstructed by the consumer walking from the root (shared
prefixes re-traverse existing tree nodes, exactly how the reference's
incremental path broadcasts re-walk the receiver's route tree).
"""
import numpy as np


class PathCodec:
    def __init__(self, row_ptr, edge_dst):
        self.row_ptr = np.asarray(row_ptr, dtype=np.int64)
        self.edge_dst = np.asarray(edge_dst, dtype=np.int32)
        max_deg = int(np.diff(self.row_ptr).max()) if len(self.row_ptr) > 1 else 0
        self.width = max(1, int(np.ceil(np.log2(max_deg + 2))))
        self.trailer = (1 << self.width) - 1
        if max_deg >= self.trailer:
            self.width += 1
            self.trailer = (1 << self.width) - 1

    def _edge_index(self, u, v):
        lo, hi = self.row_ptr[u], self.row_ptr[u + 1]
        idx = np.nonzero(self.edge_dst[lo:hi] == v)[0]
        if not len(idx):
            raise ValueError(f"no edge {u}->{v}")
        return int(idx[0])

    def encode(self, path_nodes):
        """Pack a node path [n0, n1, ... nk] into a uint64 array of
        edge-index codes (w bits/hop) terminated by the trailer code."""
        codes = [self._edge_index(path_nodes[i], path_nodes[i + 1])
                 for i in range(len(path_nodes) - 1)]
        codes.append(self.trailer)
        w = self.width
        nbits = len(codes) * w
        out = np.zeros((nbits + 63) // 64, dtype=np.uint64)
        for i, c in enumerate(codes):
            bit = i * w
            word, off = bit // 64, bit % 64
            out[word] |= np.uint64(c) << np.uint64(off)
            if off + w > 64:
                out[word + 1] |= np.uint64(c) >> np.uint64(64 - off)
        return out

    def decode(self, start_node, packed):
        """Inverse of encode: walk edge indices from start_node until the
        trailer. Returns the node path including start_node."""
        w = self.width
        mask = np.uint64(self.trailer)
        packed = np.asarray(packed, dtype=np.uint64)
        path = [int(start_node)]
        i = 0
        while True:
            bit = i * w
            word, off = bit // 64, bit % 64
            if word >= len(packed):
                raise ValueError("missing trailer")
            c = int(packed[word] >> np.uint64(off))
            if off + w > 64 and word + 1 < len(packed):
                c |= int(packed[word + 1]) << (64 - off)
            c &= int(mask)
            if c == self.trailer:
                return path
            u = path[-1]
            lo = self.row_ptr[u]
            if lo + c >= self.row_ptr[u + 1]:
                raise ValueError(f"edge index {c} out of range at node {u}")
            path.append(int(self.edge_dst[lo + c]))
            i += 1


def encode_tree_paths(codec, nodes, parents, sink_mask):
    """Encode a route tree as its root-to-sink paths (one packed array per
    sink). nodes/parents: tree arrays (parent index -1 for root);
    sink_mask: bool per tree position marking sinks."""
    out = []
    for k in np.nonzero(sink_mask)[0]:
        rev = []
        i = int(k)
        while i >= 0:
            rev.append(int(nodes[i]))
            i = int(parents[i])
        out.append(codec.encode(rev[::-1]))
    return out
