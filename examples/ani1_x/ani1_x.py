"""ANI-1x-style example (reference examples/ani1_x): mixed-species
variable-size molecules, energy+force MLIP on MACE; synthetic LJ
surrogate data (no network)."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = mlip_config("MACE", radius=5.0)
    ds = mlip_molecules(args.num_samples, n_range=(6, 20), seed=41)
    run_flow(config, ds, "ani1_x", args.num_epoch)

if __name__ == "__main__":
    main()
