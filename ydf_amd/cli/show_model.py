"""Prints a model description (reference cli/show_model.cc)."""
import argparse

import ydf_amd as ydf


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--model", required=True)
    ap.add_argument("--full_definition", action="store_true",
                    help="also print the first tree")
    args = ap.parse_args()
    model = ydf.load_model(args.model)
    print(model.describe())
    vi = model.variable_importances()
    for name, ranks in vi.items():
        print(f"\nVariable importance ({name}):")
        for i, (s, f) in enumerate(ranks[:20], 1):
            print(f"  {i:3d}. {f:30s} {s:.6g}")
    if args.full_definition and model.num_trees():
        print("\nTree 0:")
        print(model.print_tree(0))


if __name__ == "__main__":
    main()
