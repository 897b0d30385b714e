"""Transition1x-style example (reference examples/transition1x):
near-transition-state conformers, energy+forces on EGNN."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = mlip_config("EGNN", radius=5.0,
                         extra_arch={"equivariance": True})
    ds = mlip_molecules(args.num_samples, n_range=(5, 14), seed=43,
                        min_dist=0.8)
    run_flow(config, ds, "transition1x", args.num_epoch)

if __name__ == "__main__":
    main()
