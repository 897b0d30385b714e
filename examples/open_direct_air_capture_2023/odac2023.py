"""ODAC23-style example (reference
examples/open_direct_air_capture_2023): sparse MOF-like periodic
frameworks with guest molecules, energy+forces on EGNN."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = mlip_config("EGNN", radius=4.5, extra_arch={
        "periodic_boundary_conditions": True, "equivariance": False})
    ds = lj_dataset(num_samples=args.num_samples, num_atoms=64,
                    cell_size=14.0, radius=4.5, pbc=True, seed=71)
    run_flow(config, ds, "odac2023", args.num_epoch)

if __name__ == "__main__":
    main()
