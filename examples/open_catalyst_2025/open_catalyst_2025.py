"""OC25-style example (reference examples/open_catalyst_2025):
larger adsorbate+slab systems, MACE MLIP."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = mlip_config("MACE", radius=4.0, extra_arch={
        "periodic_boundary_conditions": True})
    ds = lj_dataset(num_samples=args.num_samples, num_atoms=64,
                    cell_size=10.0, radius=4.0, pbc=True, seed=67)
    run_flow(config, ds, "open_catalyst_2025", args.num_epoch)

if __name__ == "__main__":
    main()
