"""Converts a dataset between formats (reference cli/convert_dataset.cc):
csv: <-> tfrecord: <-> avro:."""
import argparse


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--input", required=True, help="typed input path")
    ap.add_argument("--output", required=True, help="typed output path")
    args = ap.parse_args()
    from ydf_amd.cli._common import read_dataset, write_dataset
    from ydf_amd.dataset.dataset import _to_column_dict

    cols = _to_column_dict(read_dataset(args.input))
    write_dataset(args.output, cols)
    n = len(next(iter(cols.values())))
    print(f"converted {n} examples -> {args.output}")


if __name__ == "__main__":
    main()
