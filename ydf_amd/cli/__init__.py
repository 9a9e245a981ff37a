"""Command-line tools (capability analogue of the reference's CLI layer,
yggdrasil_decision_forests/cli/: train, predict, evaluate, show_model,
show_dataspec, infer_dataspec, benchmark_inference,
analyze_model_and_dataset, compute_variable_importances).

Usage: python -m ydf_amd.cli.<tool> --help
"""
