"""Fetch the AdK equilibrium MD trajectory (the protein dataset).

Re-owned counterpart of the reference's dataset_generation/protein/
mdanalysis.py: downloads the MDAnalysisData AdK equilibrium dataset into
``--data-home`` and prints its shape summary. The same loader backs the
training preprocessing (distegnn_amd/data/readers/protein.py). Requires
MDAnalysis + MDAnalysisData and network access — neither exists in this
offline image, so the import is guarded with a clear message.

Usage: python fetch_adk.py --data-home protein
"""

import argparse


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--data-home", type=str, default="protein")
    ap.add_argument("--backbone", action="store_true", default=True)
    args = ap.parse_args()

    import os
    import sys

    sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(
        __file__)), "..", ".."))
    from distegnn_amd.data.readers.protein import load_adk

    universe, atom_ix, charges, n_frames, dims = load_adk(
        args.data_home, backbone=args.backbone)
    print(f"AdK equilibrium: {n_frames} frames, {len(atom_ix)} backbone "
          f"atoms, charges {tuple(charges.shape)}, box {dims[:3]}")


if __name__ == "__main__":
    main()
