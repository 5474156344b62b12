#!/usr/bin/env python3
"""Regenerate the golden flow outputs (tests/golden/golden.{net,place,route})
after a DELIBERATE fabric/packer/placer/router change. The golden test
byte-compares against these; regenerating is an explicit act recorded in
the commit that changes behavior."""
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))

from parallel_eda_amd.__main__ import main

G = ROOT / "tests" / "golden"


def run():
    rc = main([str(G / "golden.blif"), str(G / "golden_arch.xml"),
               "--route_chan_width", "12", "--seed", "3",
               "--timing_tradeoff", "0.5",
               "--out_net", str(G / "golden.net"),
               "--out_place", str(G / "golden.place"),
               "--out_route", str(G / "golden.route")])
    if rc != 0:
        raise SystemExit(f"golden flow failed rc={rc}")
    print("goldens regenerated under", G)


if __name__ == "__main__":
    run()
