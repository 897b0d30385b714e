"""OMat24-style example (reference examples/open_materials_2024):
non-equilibrium inorganic crystal structures, MACE MLIP fp32."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = mlip_config("MACE", radius=3.2, extra_arch={
        "periodic_boundary_conditions": True})
    ds = lj_dataset(num_samples=args.num_samples, num_atoms=32,
                    cell_size=6.6, radius=3.2, pbc=True, seed=73)
    run_flow(config, ds, "open_materials_2024", args.num_epoch)

if __name__ == "__main__":
    main()
