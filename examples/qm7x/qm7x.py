"""QM7-X-style example (reference examples/qm7x): five-task training
with mixed head types — graph HLGAP + node targets [forces(3), hCHG,
hVDIP, hRAT] — over the concatenated-y multihead layout (reference
qm7x.json output_dim [1,3,1,1,1])."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa

def main():
    args = standard_args()
    config = multihead_config(
        "SchNet", ["graph", "node", "node", "node", "node"],
        [1, 3, 1, 1, 1],
        output_names=["HLGAP", "forces", "hCHG", "hVDIP", "hRAT"],
        extra_arch={"num_gaussians": 16, "num_filters": 32,
                    "equivariance": False})
    ds = multihead_molecules(args.num_samples, graph_dims=(1,),
                             node_dims=(3, 1, 1, 1), seed=89)
    run_flow(config, ds, "qm7x", args.num_epoch)

if __name__ == "__main__":
    main()
