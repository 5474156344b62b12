"""Failure-forensics dumps (reference: check_route_tree's graphviz dump
to error.dot / duplicate.dot on assert failure, router.cxx:145-200)."""


def write_tree_dot(path, nodes, parents, g=None, label=""):
    """Write one route tree as graphviz; node labels carry rr type/coords
    when the graph is provided."""
    import numpy as np
    ty_names = ["SRC", "SINK", "OPIN", "IPIN", "CHANX", "CHANY"]
    ty = xl = yl = None
    if g is not None:
        ty = np.asarray(g.type)
        xl = np.asarray(g.xlow)
        yl = np.asarray(g.ylow)
    with open(path, "w") as f:
        f.write(f'digraph tree {{\n  label="{label}";\n')
        for k, v in enumerate(nodes):
            if ty is not None:
                f.write(f'  n{k} [label="{int(v)}\\n'
                        f'{ty_names[ty[v]]}({xl[v]},{yl[v]})"];\n')
            else:
                f.write(f'  n{k} [label="{int(v)}"];\n')
        for k, p in enumerate(parents):
            if p >= 0:
                f.write(f"  n{int(p)} -> n{k};\n")
        f.write("}\n")
