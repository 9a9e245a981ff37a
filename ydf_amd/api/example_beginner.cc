// C++ API walkthrough (reference examples/beginner.cc analogue):
// synthesize a dataset, train a GBT, evaluate, save the reference-
// format model directory, reload it and check prediction parity.
//
// Build (no ROCm/Python/protobuf needed):
//   g++ -O2 -std=c++17 example_beginner.cc api.cc ../ops/cc/cpu_ops.cpp \
//       -o beginner -lpthread
#include <cmath>
#include <cstdio>
#include <random>

#include "ydf_amd_api.h"

int main(int argc, char** argv) {
  const char* out_dir = argc > 1 ? argv[1] : "/tmp/ydfa_cpp_model";
  std::mt19937 rng(7);
  std::normal_distribution<float> g;
  const int n = 20000;
  std::vector<float> x1(n), x2(n);
  std::vector<std::string> color(n), label(n);
  const char* colors[] = {"red", "green", "blue", "yellow"};
  for (int i = 0; i < n; ++i) {
    x1[i] = g(rng);
    x2[i] = g(rng);
    color[i] = colors[rng() % 4];
    const bool pos = 2 * x1[i] - x2[i] + (color[i] == "red" ? 1.5f : 0.f)
                     + 0.3f * g(rng) > 0;
    label[i] = pos ? "yes" : "no";
  }
  ydfa::api::Dataset ds;
  ds.AddNumerical("x1", x1);
  ds.AddNumerical("x2", x2);
  ds.AddCategorical("color", color);
  ds.AddCategorical("label", label);

  ydfa::api::GbtConfig cfg;
  cfg.num_trees = 50;
  auto model = ydfa::api::TrainGradientBoostedTrees(cfg, ds, "label");

  const auto preds = model->Predict(ds);
  int correct = 0;
  for (int i = 0; i < n; ++i)
    if ((preds[i] > 0.5f) ==
        (label[i] == model->label_classes()[1]))
      ++correct;
  const double acc = (double)correct / n;
  std::printf("trees=%d nodes=%d accuracy=%.4f\n", model->num_trees(),
              model->num_nodes(), acc);
  if (acc < 0.9) {
    std::printf("FAIL: accuracy too low\n");
    return 1;
  }

  model->Save(out_dir);
  auto re = ydfa::api::Model::Load(out_dir);
  const auto preds2 = re->Predict(ds);
  float max_diff = 0.f;
  for (int i = 0; i < n; ++i)
    max_diff = std::max(max_diff, std::fabs(preds[i] - preds2[i]));
  std::printf("save/load max prediction diff: %g\n", max_diff);
  if (max_diff > 1e-5f) {
    std::printf("FAIL: reload mismatch\n");
    return 1;
  }
  std::printf("OK\n");
  return 0;
}
