"""QCML-style example (reference examples/qcml): broad quantum
chemistry sweep — graph energy + node charge heads."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = multihead_config("EGNN", ["graph", "node"], [1, 1],
                              output_names=["energy", "charge"],
                              extra_arch={"equivariance": False})
    ds = multihead_molecules(args.num_samples, graph_dims=(1,),
                             node_dims=(1,), seed=101)
    run_flow(config, ds, "qcml", args.num_epoch)

if __name__ == "__main__":
    main()
