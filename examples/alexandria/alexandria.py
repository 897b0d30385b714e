"""Alexandria-style example (reference examples/alexandria): periodic
inorganic materials, energy+forces on PaiNN."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = mlip_config("PAINN", radius=3.0, extra_arch={
        "num_radial": 12, "equivariance": True,
        "periodic_boundary_conditions": True})
    ds = lj_dataset(num_samples=args.num_samples, num_atoms=27,
                    cell_size=6.0, radius=3.0, pbc=True, seed=47)
    run_flow(config, ds, "alexandria", args.num_epoch)

if __name__ == "__main__":
    main()
