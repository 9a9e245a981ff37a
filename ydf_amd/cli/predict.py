"""Generates predictions (reference cli/predict.cc)."""
import argparse

import numpy as np

import ydf_amd as ydf
from ydf_amd.cli._common import read_dataset


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--model", required=True)
    ap.add_argument("--dataset", required=True)
    ap.add_argument("--output", required=True, help="csv output path")
    args = ap.parse_args()
    model = ydf.load_model(args.model)
    preds = model.predict(read_dataset(args.dataset))
    header = ",".join(model.label_classes[1:]) if (
        model.label_classes and preds.ndim == 1) else "prediction"
    if preds.ndim == 2 and model.label_classes:
        header = ",".join(model.label_classes)
    np.savetxt(args.output, preds, delimiter=",", header=header, comments="")
    print(f"wrote {len(preds)} predictions to {args.output}")


if __name__ == "__main__":
    main()
