"""OC22-style total-energy example (reference
examples/open_catalyst_2022): oxide slabs, total energy + forces on
SchNet."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = mlip_config("SchNet", radius=4.0, extra_arch={
        "num_gaussians": 32, "num_filters": 32,
        "periodic_boundary_conditions": True})
    ds = lj_dataset(num_samples=args.num_samples, num_atoms=48,
                    cell_size=9.0, radius=4.0, pbc=True, seed=61)
    run_flow(config, ds, "open_catalyst_2022", args.num_epoch)

if __name__ == "__main__":
    main()
