"""Permutation variable importances
(reference cli/compute_variable_importances.cc)."""
import argparse

import ydf_amd as ydf
from ydf_amd.cli._common import read_dataset


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--model", required=True)
    ap.add_argument("--dataset", required=True)
    ap.add_argument("--num_repetitions", type=int, default=1)
    args = ap.parse_args()
    model = ydf.load_model(args.model)
    an = model.analyze(read_dataset(args.dataset))
    for name, ranks in an.variable_importances.items():
        print(f"Variable importance ({name}):")
        for i, (s, f) in enumerate(ranks, 1):
            print(f"  {i:3d}. {f:30s} {s:.6g}")


if __name__ == "__main__":
    main()
