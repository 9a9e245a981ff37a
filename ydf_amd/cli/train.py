"""Trains a model (reference cli/train.cc)."""
import argparse
import json

from ydf_amd.cli._common import LEARNERS, TASKS, read_dataset


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--dataset", required=True, help="csv:<path> train data")
    ap.add_argument("--output", required=True, help="model directory")
    ap.add_argument("--label", required=True)
    ap.add_argument("--learner", default="GRADIENT_BOOSTED_TREES",
                    choices=sorted(LEARNERS))
    ap.add_argument("--task", default="CLASSIFICATION", choices=sorted(TASKS))
    ap.add_argument("--weights", default=None)
    ap.add_argument("--hparams", default="{}",
                    help='JSON generic hyper-parameters, e.g. '
                         '\'{"num_trees": 500}\'')
    args = ap.parse_args()
    hp = json.loads(args.hparams)
    learner = LEARNERS[args.learner](label=args.label, task=TASKS[args.task],
                                     weights=args.weights, **hp)
    model = learner.train(read_dataset(args.dataset))
    model.save(args.output)
    print(model.describe())


if __name__ == "__main__":
    main()
