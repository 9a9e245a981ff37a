"""Generates a synthetic dataset (reference cli/utils/synthesize_dataset.cc)."""
import argparse


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--output", required=True,
                    help="typed path, e.g. csv:/tmp/train.csv")
    ap.add_argument("--num_examples", type=int, default=10000)
    ap.add_argument("--num_numerical", type=int, default=8)
    ap.add_argument("--num_categorical", type=int, default=2)
    ap.add_argument("--num_boolean", type=int, default=1)
    ap.add_argument("--task", default="classification",
                    choices=["classification", "regression", "ranking"])
    ap.add_argument("--num_classes", type=int, default=2)
    ap.add_argument("--missing_ratio", type=float, default=0.0)
    ap.add_argument("--seed", type=int, default=1234)
    args = ap.parse_args()
    from ydf_amd.cli._common import write_dataset
    from ydf_amd.dataset.synthetic import generate_synthetic_dataset

    cols = generate_synthetic_dataset(
        num_examples=args.num_examples, num_numerical=args.num_numerical,
        num_categorical=args.num_categorical,
        num_boolean=args.num_boolean, task=args.task,
        num_classes=args.num_classes, missing_ratio=args.missing_ratio,
        seed=args.seed)
    write_dataset(args.output, cols)
    print(f"wrote {args.num_examples} examples to {args.output}")


if __name__ == "__main__":
    main()
