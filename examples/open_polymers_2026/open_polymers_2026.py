"""OPoly-style example (reference examples/open_polymers_2026):
polymer property regression — graph-level multi-dim targets over
chain-topology molecules on PNA-free stacks (GIN here)."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = multihead_config("GIN", ["graph"], [3],
                              input_features=4,
                              output_names=["Tg_density_ffv"])
    ds = topology_graphs(args.num_samples, n_range=(16, 40), p=0.08,
                         seed=83, out_dim=3)
    run_flow(config, ds, "open_polymers_2026", args.num_epoch)

if __name__ == "__main__":
    main()
