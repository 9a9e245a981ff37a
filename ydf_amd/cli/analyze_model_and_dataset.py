"""Model analysis report (reference cli/analyze_model_and_dataset.cc)."""
import argparse

import ydf_amd as ydf
from ydf_amd.cli._common import read_dataset


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--model", required=True)
    ap.add_argument("--dataset", required=True)
    ap.add_argument("--output", default=None, help="write HTML report")
    args = ap.parse_args()
    model = ydf.load_model(args.model)
    an = model.analyze(read_dataset(args.dataset))
    print(an.to_text())
    if args.output:
        with open(args.output, "w") as f:
            f.write("<html><body>" + an._repr_html_() + "</body></html>")
        print(f"wrote {args.output}")


if __name__ == "__main__":
    main()
