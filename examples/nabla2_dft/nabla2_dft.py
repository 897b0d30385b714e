"""nabla2-DFT-style example (reference examples/nabla2_dft): drug-like
conformers, energy+forces MLIP on PaiNN."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = mlip_config("PAINN", radius=5.0, extra_arch={
        "num_radial": 12, "equivariance": True})
    ds = mlip_molecules(args.num_samples, n_range=(12, 30), seed=97)
    run_flow(config, ds, "nabla2_dft", args.num_epoch)

if __name__ == "__main__":
    main()
