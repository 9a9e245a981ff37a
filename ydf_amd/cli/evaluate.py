"""Evaluates a model on a dataset (reference cli/evaluate.cc)."""
import argparse

import ydf_amd as ydf
from ydf_amd.cli._common import read_dataset


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--model", required=True)
    ap.add_argument("--dataset", required=True)
    args = ap.parse_args()
    model = ydf.load_model(args.model)
    print(model.evaluate(read_dataset(args.dataset)))


if __name__ == "__main__":
    main()
