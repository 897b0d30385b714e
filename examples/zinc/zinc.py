"""ZINC-style example (reference examples/zinc): positionless
molecular topology, penalized-logP graph regression on GIN."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = multihead_config("GIN", ["graph"], [1],
                              input_features=4,
                              output_names=["logP"])
    ds = topology_graphs(args.num_samples, n_range=(9, 37), p=0.12,
                         seed=107)
    run_flow(config, ds, "zinc", args.num_epoch)

if __name__ == "__main__":
    main()
