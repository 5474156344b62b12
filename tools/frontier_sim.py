"""Algorithm simulator for the router's frontier structures.

Validates the ROUND-2 calendar-bucket frontier design against the current
ping-pong design and an exact Dijkstra reference, on the real rr graph,
BEFORE spending GPU time on the HIP port. Simulates one workgroup's
per-sink search (sequential semantics — races are validated on-device;
this pins down termination, bucket-wrap, overflow and reopen behavior
and measures the scan-work ratio the bucket queue is supposed to win).

Run: python tools/frontier_sim.py [arch] [n_sinks]
"""
import heapq
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import numpy as np

from parallel_eda_amd.arch.archdef import get_arch
from parallel_eda_amd import rrgraph


class Cost:
    """Mirrors the kernel's edge relaxation (crit=0 congestion-only)."""

    def __init__(self, g, arch, occ=None, pres_fac=0.5):
        self.row_ptr = np.asarray(g.row_ptr)
        self.dst = np.asarray(g.edge_dst)
        self.sw = np.asarray(g.edge_sw)
        self.ty = np.asarray(g.type)
        self.base = np.asarray(g.base_cost)
        self.cap = np.asarray(g.capacity)
        self.occ = occ if occ is not None else np.zeros(g.num_nodes, np.int32)
        self.pres_fac = pres_fac
        self.xl = np.asarray(g.xlow); self.yl = np.asarray(g.ylow)
        self.xh = np.asarray(g.xhigh); self.yh = np.asarray(g.yhigh)
        self.L = arch.L
        self.seg_base = float(self.base[4])
        self.astar = 1.2

    def edge_cost(self, w):
        over = self.occ[w] + 1 - self.cap[w]
        pres = 1.0 + over * self.pres_fac if over > 0 else 1.0
        return np.float32(self.base[self.ty[w]] * pres)

    def h(self, v, sink):
        tx, ty2 = self.xl[sink], self.yl[sink]
        dx = max(self.xl[v] - tx, tx - self.xh[v], 0)
        dy = max(self.yl[v] - ty2, ty2 - self.yh[v], 0)
        nseg = -(-(dx + dy) // self.L)
        return np.float32(self.astar * (nseg * self.seg_base + 0.95))

    def edges(self, v, sink):
        for e in range(self.row_ptr[v], self.row_ptr[v + 1]):
            w = self.dst[e]
            t = self.ty[w]
            if t == 1 and w != sink:
                continue
            if t == 3 and (self.xl[w] != self.xl[sink] or
                           self.yl[w] != self.yl[sink]):
                continue
            yield int(w)


def dijkstra_ref(c, src, sink):
    dist = {src: np.float32(0.0)}
    pq = [(0.0, src)]
    while pq:
        d, v = heapq.heappop(pq)
        if v == sink:
            return d
        if d > dist.get(v, np.inf):
            continue
        for w in c.edges(v, sink):
            nd = np.float32(d + c.edge_cost(w))
            if nd < dist.get(w, np.inf):
                dist[w] = nd
                heapq.heappush(pq, (float(nd), w))
    return None


def pingpong(c, src, sink, delta, f_cap=1 << 16):
    """Current kernel structure: ping-pong buffers, kept-entry rescans."""
    INF = np.float32(np.inf)
    best = {}
    cur = [(np.float32(c.h(src, sink)), np.float32(0.0), src, src)]
    best[src] = (np.float32(0.0), src)
    best_sink = INF
    scans = appends = rounds = 0
    fmin = cur[0][0]
    while cur:
        if best_sink <= fmin:
            break
        rounds += 1
        thr = fmin + delta
        nxt = []
        fmin_next = INF
        for (tot, back, v, prev) in cur:
            scans += 1
            if best.get(v, (INF, -1)) != (back, prev):
                continue
            if tot > thr:
                nxt.append((tot, back, v, prev))
                fmin_next = min(fmin_next, tot)
                continue
            if v == sink:
                continue
            for w in c.edges(v, sink):
                nb = np.float32(back + c.edge_cost(w))
                old = best.get(w, (INF, -1))
                if (nb, v) < old:
                    best[w] = (nb, v)
                    ntot = np.float32(nb + c.h(w, sink))
                    nxt.append((ntot, nb, w, v))
                    appends += 1
                    fmin_next = min(fmin_next, ntot)
                    if w == sink:
                        best_sink = min(best_sink, nb)
        if len(nxt) > f_cap:
            raise RuntimeError("frontier overflow")
        cur = nxt
        fmin = fmin_next
    return best_sink, dict(scans=scans, appends=appends, rounds=rounds)


def calendar(c, src, sink, delta, nb=64, bucket_cap=1 << 12):
    """Round-2 design: circular calendar buckets + overflow bucket.

    bucket index = floor((tot - f0) / delta) relative to the calendar base
    f0; entries beyond nb-2 go to the overflow bucket (index nb-1), which
    is redistributed when the base catches up. Each entry is touched
    O(1 + redistributions) instead of O(rounds)."""
    INF = np.float32(np.inf)
    best = {src: (np.float32(0.0), src)}
    f0 = float(c.h(src, sink))
    buckets = [[] for _ in range(nb)]
    over = []

    def push(tot, back, v, prev):
        k = int((tot - f0) / delta)
        if k < 0:
            k = 0
        if k >= nb - 1:
            over.append((tot, back, v, prev))
        else:
            buckets[(base + k) % nb].append((tot, back, v, prev))

    base = 0
    best_sink = INF
    scans = appends = rounds = redist = 0
    push(np.float32(f0), np.float32(0.0), src, src)
    steps = 0
    while True:
        steps += 1
        if steps > 10 ** 7:
            raise RuntimeError("no termination")
        # current bucket
        cur = buckets[base % nb]
        if not cur:
            if best_sink <= f0 + delta:
                break
            # advance the calendar
            empty = all(not b for b in buckets)
            if empty:
                if not over:
                    break
                # redistribute overflow against the new base
                redist += 1
                f0 = float(min(t for (t, *_rest) in over))
                items, over = over, []
                for it in items:
                    scans += 1
                    push(*it)
                continue
            base += 1
            f0 += delta
            continue
        rounds += 1
        buckets[base % nb] = []
        for (tot, back, v, prev) in cur:
            scans += 1
            if best.get(v, (INF, -1)) != (back, prev):
                continue
            if v == sink:
                best_sink = min(best_sink, back)
                continue
            for w in c.edges(v, sink):
                nbk = np.float32(back + c.edge_cost(w))
                old = best.get(w, (INF, -1))
                if (nbk, v) < old:
                    best[w] = (nbk, v)
                    ntot = np.float32(nbk + c.h(w, sink))
                    push(ntot, nbk, w, v)
                    appends += 1
                    if w == sink:
                        best_sink = min(best_sink, nbk)
        if best_sink <= f0:
            break
    return best_sink, dict(scans=scans, appends=appends, rounds=rounds,
                           redist=redist)


def main():
    arch_name = sys.argv[1] if len(sys.argv) > 1 else "tseng"
    n_pairs = int(sys.argv[2]) if len(sys.argv) > 2 else 40
    arch = get_arch(arch_name)
    g = rrgraph.build_rr_graph(arch)
    rng = np.random.default_rng(3)
    ts = np.asarray(g.tile_source)
    tk = np.asarray(g.tile_sink)
    gy = arch.ny + 2
    occ = rng.integers(0, 3, g.num_nodes).astype(np.int32)  # fake congestion
    c = Cost(g, arch, occ=occ, pres_fac=1.5)
    delta = 3.0 * c.seg_base
    tot_pp = dict(scans=0, appends=0, rounds=0)
    tot_cal = dict(scans=0, appends=0, rounds=0, redist=0)
    mism = 0
    for _ in range(n_pairs):
        sx, sy = 1 + rng.integers(arch.nx), 1 + rng.integers(arch.ny)
        tx, ty2 = 1 + rng.integers(arch.nx), 1 + rng.integers(arch.ny)
        if (sx, sy) == (tx, ty2):
            continue
        src = int(ts[sx * gy + sy])
        sink = int(tk[tx * gy + ty2])
        ref = dijkstra_ref(c, src, sink)
        bp, sp = pingpong(c, src, sink, delta)
        bc, sc = calendar(c, src, sink, delta)
        for k in tot_pp:
            tot_pp[k] += sp[k]
        for k in tot_cal:
            tot_cal[k] += sc[k]
        # delta-stepping is inexact by up to ~delta of ordering slack;
        # both structures must land within that envelope of the optimum
        for name, got in (("pingpong", bp), ("calendar", bc)):
            if got > ref + delta + 1e-4:
                print(f"MISMATCH {name}: got {got} ref {ref}")
                mism += 1
    print(f"pairs={n_pairs} mismatches={mism}")
    print(f"ping-pong: {tot_pp}")
    print(f"calendar:  {tot_cal}")
    if tot_cal['scans']:
        print(f"scan ratio pingpong/calendar = "
              f"{tot_pp['scans'] / tot_cal['scans']:.2f}x")
    return 0 if mism == 0 else 1


if __name__ == "__main__":
    sys.exit(main())
