"""Infers and prints a dataspec from a dataset
(reference cli/infer_dataspec.cc)."""
import argparse
import json

from ydf_amd.cli._common import read_dataset
from ydf_amd.dataset.dataset import _to_column_dict, infer_dataspec


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--dataset", required=True)
    ap.add_argument("--label", default=None)
    ap.add_argument("--output", default=None, help="write dataspec.json")
    args = ap.parse_args()
    spec = infer_dataspec(_to_column_dict(read_dataset(args.dataset)),
                          label=args.label)
    if args.output:
        with open(args.output, "w") as f:
            json.dump(spec.to_json(), f, indent=1)
    for c in spec.columns:
        print(f"  {c.name:30s} {c.semantic.name}")


if __name__ == "__main__":
    main()
