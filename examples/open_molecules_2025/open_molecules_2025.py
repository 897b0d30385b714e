"""OMol25-style example (reference examples/open_molecules_2025):
large organic molecules (up to ~50 atoms here), MACE MLIP."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = mlip_config("MACE", radius=5.0)
    ds = mlip_molecules(args.num_samples, n_range=(20, 48), seed=79)
    run_flow(config, ds, "open_molecules_2025", args.num_epoch)

if __name__ == "__main__":
    main()
