// ydf_amd C++ user API — training and serving without Python.
//
// Capability analogue of the reference's C++ API alias headers
// (api/training.h:38-60, api/serving.h) and examples/beginner.cc:
// build a Dataset, train a GradientBoostedTrees model, save/load the
// REFERENCE on-disk model directory (header.pb / data_spec.pb /
// gradient_boosted_trees_header.pb / nodes-00000-of-00001 / done,
// model/model_library.cc:92-107), and predict.
//
// The training hot path runs the same C++ ops (cpu_ops.cpp) that twin
// the HIP kernels: quantile binning, level-wise histogram build,
// split-gain scan (hessian gain, categorical CART ordering), row
// routing, leaf values — so a model trained here matches the Python
// CPU path's semantics.
//
// Build: link api.cc + ../ops/cc/cpu_ops.cpp with any C++17 compiler
// (no HIP/ROCm, Python or protobuf dependency).
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <string>
#include <vector>

namespace ydfa {
namespace api {

enum class ColumnType { kNumerical, kCategorical, kBoolean };
enum class Task { kClassification, kRegression };

struct ColumnSpec {
  std::string name;
  ColumnType type = ColumnType::kNumerical;
  std::vector<std::string> vocab;  // categorical: index 0 = OOV
  double mean = 0.0;               // numerical: imputation value
  std::vector<float> boundaries;   // numerical: quantile cuts
};

// Column-major in-memory dataset (the serving ExampleSet / training
// VerticalDataset analogue).
class Dataset {
 public:
  void AddNumerical(const std::string& name,
                    const std::vector<float>& values);
  void AddCategorical(const std::string& name,
                      const std::vector<std::string>& values);
  size_t num_rows() const { return num_rows_; }

  const std::map<std::string, std::vector<float>>& numericals() const {
    return numericals_;
  }
  const std::map<std::string, std::vector<std::string>>& categoricals()
      const {
    return categoricals_;
  }

 private:
  size_t num_rows_ = 0;
  std::map<std::string, std::vector<float>> numericals_;
  std::map<std::string, std::vector<std::string>> categoricals_;
};

// Flat decision-forest model (SoA node arrays, the serving layout).
class Model {
 public:
  // Loads a reference-format model directory (GBT or RF;
  // classification probability / regression value output).
  static std::unique_ptr<Model> Load(const std::string& directory);

  // Writes the reference-format model directory.
  void Save(const std::string& directory) const;

  // Predictions for every row: binary classification -> P(positive
  // class); regression -> value.
  std::vector<float> Predict(const Dataset& dataset) const;

  // Single example, features in data-spec feature order (categorical
  // = vocabulary index as float).
  float PredictRow(const float* features) const;

  int num_trees() const { return (int)roots_.size(); }
  int num_nodes() const { return (int)feat_.size(); }
  Task task() const { return task_; }
  const std::vector<ColumnSpec>& features() const { return features_; }
  const std::string& label() const { return label_; }
  const std::vector<std::string>& label_classes() const {
    return label_classes_;
  }

  // --- internal state (filled by the trainer / loader) ---
  std::vector<int32_t> feat_;   // -1 = leaf
  std::vector<float> thr_;      // threshold (x > thr -> right) / leaf
  std::vector<int32_t> left_;   // right = left + 1
  std::vector<int32_t> roots_;
  std::vector<int32_t> cat_idx_;          // -1 or mask index
  std::vector<uint64_t> masks_;           // [n_masks][4]
  std::vector<float> cover_;              // training example count
  std::vector<ColumnSpec> features_;      // data-spec feature order
  std::string label_;
  std::vector<std::string> label_classes_;
  Task task_ = Task::kClassification;
  float init_prediction_ = 0.0f;
  float leaf_scale_ = 1.0f;   // GBT: shrinkage baked at build; RF: 1/T
  bool sigmoid_ = false;
  bool is_rf_ = false;
  bool winner_take_all_ = false;
};

// GBT hyper-parameters (reference names/defaults).
struct GbtConfig {
  int num_trees = 300;
  int max_depth = 6;
  float shrinkage = 0.1f;
  int min_examples = 5;
  float min_sum_hessian_in_leaf = 1e-3f;
  float l2_regularization = 0.0f;
  float l1_regularization = 0.0f;
  float l2_categorical_regularization = 1.0f;
  int random_seed = 123456;  // reserved (no stochastic paths yet)
};

// Trains a gradient-boosted-trees model (binary classification when
// the label column is categorical with two classes; regression when
// numerical). No validation split / early stopping in the C++ API —
// train exactly num_trees trees.
std::unique_ptr<Model> TrainGradientBoostedTrees(
    const GbtConfig& config, const Dataset& dataset,
    const std::string& label);

}  // namespace api
}  // namespace ydfa
