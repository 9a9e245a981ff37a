"""Edits a saved model (reference cli/edit_model.cc): rename the
label/weights columns, or strip training-only payloads
(pure_serving)."""
import argparse

import ydf_amd as ydf


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--input", required=True, help="input model dir")
    ap.add_argument("--output", required=True, help="output model dir")
    ap.add_argument("--new_label_name", default=None)
    ap.add_argument("--new_weights_name", default=None)
    ap.add_argument("--pure_serving", default=None,
                    help="true/false: strip training-only data "
                         "(reference MakePureServing)")
    args = ap.parse_args()

    model = ydf.load_model(args.input)
    if args.new_label_name is not None:
        spec = model.dataspec
        if spec.label is not None:
            spec.label_column.name = args.new_label_name
        spec.label = args.new_label_name
    if args.new_weights_name is not None:
        model.metadata["weights_column"] = args.new_weights_name
    if args.pure_serving is not None and \
            args.pure_serving.lower() in ("true", "1", "yes"):
        import numpy as np

        model.training_logs = None
        model.tuner_logs = None
        model._self_evaluation = None
        if model.metadata:
            model.metadata.pop("feature_gains", None)
        model.forest.cover = np.zeros_like(model.forest.cover)
    model.save(args.output)
    print(f"model written to {args.output}")


if __name__ == "__main__":
    main()
