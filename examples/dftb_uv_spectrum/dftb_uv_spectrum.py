"""DFTB UV-spectrum example (reference examples/dftb_uv_spectrum,
train_smooth_uv_spectrum.py / train_discrete_uv_spectrum.py): ONE
graph head predicting a whole spectrum vector (reference smooth
output_dim [37500]; scaled down here)."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _example_lib import *  # noqa
import argparse

def main():
    p = argparse.ArgumentParser()
    p.add_argument("--num_epoch", type=int, default=None)
    p.add_argument("--num_samples", type=int, default=32)
    p.add_argument("--mode", choices=["smooth", "discrete"],
                   default="smooth")
    args = p.parse_args()
    n_freq = 64
    config = multihead_config("GIN", ["graph"], [n_freq],
                              output_names=["spectrum"],
                              hidden_dim=48)
    ds = spectrum_molecules(args.num_samples, n_freq=n_freq,
                            smooth=args.mode == "smooth", seed=113)
    run_flow(config, ds, f"dftb_uv_{args.mode}", args.num_epoch)

if __name__ == "__main__":
    main()
