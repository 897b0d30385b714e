"""CSCE-style example (reference examples/csce): GAP band-gap
regression over molecular graphs, single graph head on GAT."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = multihead_config("GAT", ["graph"], [1],
                              input_features=4,
                              output_names=["GAP"],
                              extra_arch={"edge_dim": 3})
    ds = topology_graphs(args.num_samples, n_range=(12, 28), p=0.15,
                         seed=103)
    run_flow(config, ds, "csce", args.num_epoch)

if __name__ == "__main__":
    main()
