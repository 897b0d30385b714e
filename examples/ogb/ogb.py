"""OGB-style example (reference examples/ogb): ogbg-mol* graph
property prediction with edge features on CGCNN."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = multihead_config("CGCNN", ["graph"], [2],
                              input_features=4,
                              output_names=["mol_props"],
                              extra_arch={"edge_dim": 3})
    ds = topology_graphs(args.num_samples, n_range=(10, 32), p=0.15,
                         seed=109, out_dim=2)
    run_flow(config, ds, "ogb", args.num_epoch)

if __name__ == "__main__":
    main()
