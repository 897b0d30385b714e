"""MPtrj-style example (reference examples/mptrj): Materials Project
relaxation-trajectory frames (periodic), MACE MLIP."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = mlip_config("MACE", radius=3.0, extra_arch={
        "periodic_boundary_conditions": True})
    ds = lj_dataset(num_samples=args.num_samples, num_atoms=27,
                    cell_size=6.2, radius=3.0, pbc=True, seed=53)
    run_flow(config, ds, "mptrj", args.num_epoch)

if __name__ == "__main__":
    main()
