"""Benchmarks model inference speed (reference cli/benchmark_inference.cc;
defaults mirror its batch_size=100, num_runs=20)."""
import argparse

import ydf_amd as ydf
from ydf_amd.cli._common import read_dataset


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--model", required=True)
    ap.add_argument("--dataset", required=True)
    ap.add_argument("--num_runs", type=int, default=20)
    ap.add_argument("--warmup_runs", type=int, default=1)
    ap.add_argument("--device", default=None)
    args = ap.parse_args()
    model = ydf.load_model(args.model)
    res = model.benchmark(read_dataset(args.dataset),
                          benchmark_duration=args.num_runs * 0.05,
                          warmup_duration=args.warmup_runs * 0.05,
                          device=args.device)
    print(res)


if __name__ == "__main__":
    main()
