"""Prints a model's data specification (reference cli/show_dataspec.cc)."""
import argparse
import json
import os

from ydf_amd.dataset.dataspec import DataSpecification, Semantic


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--model", help="model dir (reads its dataspec)")
    ap.add_argument("--dataspec", help="dataspec.json path")
    args = ap.parse_args()
    path = args.dataspec or os.path.join(args.model, "dataspec.json")
    if os.path.exists(path):
        with open(path) as f:
            spec = DataSpecification.from_json(json.load(f))
    else:
        # reference-format model dir (the default save() layout)
        from ydf_amd.model.model_lib import load_model

        spec = load_model(args.model).dataspec
    print(f"{len(spec.columns)} column(s); label: {spec.label!r}")
    for c in spec.columns:
        extra = ""
        if c.semantic == Semantic.CATEGORICAL:
            extra = f" vocab={c.vocab_size}"
        elif c.boundaries is not None:
            extra = (f" mean={c.mean:.4g} min={c.min_value:.4g} "
                     f"max={c.max_value:.4g} cuts={len(c.boundaries)}")
        print(f"  {c.name:30s} {c.semantic.name}{extra}")


if __name__ == "__main__":
    main()
