"""parallel_eda_amd — MI355X-native FPGA place-and-route engine.

A from-scratch GPU-native rebuild of the capabilities of the reference
parallel-VPR codebase (chinhau5/parallel_eda): simulated-annealing placement,
PathFinder negotiated-congestion timing-driven routing, and static timing
analysis over a routing-resource graph, with the hot engines as hand-written
CDNA4 HIP kernels and multi-GPU scaling via RCCL over xGMI.

Layer map (mirrors SURVEY.md section 1 of the reference analysis):
  arch/     device model (architecture params, grid)           [ref: libarchfpga]
  rrgraph   routing-resource graph builder (C++ host)          [ref: vpr/SRC/route/rr_graph.c]
  io/       .blif/.net/.place/.route + synthetic netlists      [ref: vpr/SRC/base/read_blif.c etc.]
  place/    SA placer (CPU oracle + GPU batched engine)        [ref: vpr/SRC/place/place.c]
  route/    PathFinder router (CPU oracle + GPU wavefront)     [ref: vpr/SRC/route/, parallel_route/]
  timing/   STA: levelized slack/criticality sweeps            [ref: vpr/SRC/timing/path_delay.c]
  parallel/ multi-GPU decomposition + RCCL collectives         [ref: vpr/SRC/parallel_route/mpi_*]
"""

__version__ = "0.1.0"
