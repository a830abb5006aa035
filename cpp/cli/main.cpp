/*! migbm CLI — `migbm config=train.conf [key=value ...]`.
 *  Capability parity target: reference src/main.cpp + src/application/application.cpp
 *  (tasks train / predict / refit; config-file + argv key=value parsing). */
#include "migbm/boosting.h"
#include "migbm/common.h"
#include "migbm/config.h"
#include "migbm/dataset.h"
#include "migbm/metric.h"
#include "migbm/objective.h"

#include "migbm/network.h"

#include <cstdio>
#include <fstream>
#include <memory>
#include <string>

namespace migbm {

class Application {
 public:
  explicit Application(int argc, char** argv) {
    std::unordered_map<std::string, std::string> params;
    for (int i = 1; i < argc; ++i) {
      // split on the FIRST '=' only: values may themselves contain '='
      std::string a(argv[i]);
      auto eq = a.find('=');
      if (eq == std::string::npos || eq == 0) continue;
      params[Config::ResolveAlias(Common::ToLower(Common::Trim(a.substr(0, eq))))] =
          Common::Trim(a.substr(eq + 1));
    }
    // config file first, argv overrides
    auto it = params.find("config");
    if (it != params.end()) {
      std::ifstream f(it->second);
      if (!f.good()) Log::Fatal("Cannot open config file %s", it->second.c_str());
      std::string line;
      std::unordered_map<std::string, std::string> file_params;
      while (std::getline(f, line)) {
        auto hash = line.find('#');
        if (hash != std::string::npos) line = line.substr(0, hash);
        line = Common::Trim(line);
        if (line.empty()) continue;
        auto eq = line.find('=');
        if (eq == std::string::npos) continue;
        file_params[Config::ResolveAlias(Common::ToLower(Common::Trim(line.substr(0, eq))))] =
            Common::Trim(line.substr(eq + 1));
      }
      for (auto& kv : params) file_params[kv.first] = kv.second;  // argv wins
      params = std::move(file_params);
    }
    config_.Set(params);
  }

  void Run() {
    // distributed CPU training over the standalone TCP mesh (reference
    // application.cpp Network::Init parity): machines= or machine_list_filename
    if (config_.num_machines > 1 &&
        (config_.tree_learner == "data" || config_.tree_learner == "feature" ||
         config_.tree_learner == "voting")) {
      std::string machines = config_.machines;
      if (machines.empty() && !config_.machine_list_filename.empty()) {
        std::ifstream mf(config_.machine_list_filename);
        if (!mf.good())
          Log::Fatal("Cannot open machine list %s", config_.machine_list_filename.c_str());
        std::string line;
        while (std::getline(mf, line)) {
          line = Common::Trim(line);
          if (line.empty()) continue;
          if (!machines.empty()) machines += ",";
          machines += line;
        }
      }
      if (machines.empty())
        Log::Fatal("num_machines>1 requires machines= or machine_list_filename=");
      NetworkInitSockets(machines, config_.local_listen_port, config_.time_out,
                         config_.num_machines);
    }
    if (config_.task == "train") Train();
    else if (config_.task == "refit" || config_.task == "refit_tree") Refit();
    else if (config_.task == "predict" || config_.task == "prediction" ||
             config_.task == "test")
      Predict();
    else if (config_.task == "convert_model")
      ConvertModel();
    else
      Log::Fatal("Unknown task %s", config_.task.c_str());
  }

 private:
  void LoadData() {
    if (config_.data.empty()) Log::Fatal("No training data (data=...) specified");
    DatasetLoader loader(config_);
    if (Dataset::IsBinFile(config_.data.c_str())) {
      train_data_ = Dataset::LoadFromBinFile(config_.data.c_str());
    } else if (Network::is_distributed() && !config_.pre_partition) {
      // round-robin row sharding by rank (each machine reads the shared file)
      train_data_ = loader.LoadFromFile(config_.data.c_str(), Network::rank(),
                                        Network::num_machines());
    } else {
      train_data_ = loader.LoadFromFile(config_.data.c_str());
    }
    Log::Info("Loaded train data: %d rows, %d features", train_data_->num_data(),
              train_data_->num_total_features());
    for (auto& vf : config_.valid) {
      if (vf.empty()) continue;
      valid_data_.push_back(loader.LoadFromFileAlignWithOtherDataset(vf.c_str(),
                                                                     train_data_.get()));
      Log::Info("Loaded valid data %s: %d rows", vf.c_str(),
                valid_data_.back()->num_data());
    }
    if (config_.save_binary) {
      std::string out = config_.data + ".bin";
      train_data_->SaveBinaryFile(out.c_str());
      Log::Info("Saved binary dataset to %s", out.c_str());
    }
  }

  void Train() {
    LoadData();
    objective_.reset(ObjectiveFunction::Create(config_.objective, config_));
    if (objective_) objective_->Init(train_data_->metadata(), train_data_->num_data());
    auto metric_names = config_.metric;
    if (metric_names.empty())
      metric_names.push_back(config_.objective == "regression" ? "l2" : config_.objective);
    for (auto& name : metric_names) {
      std::unique_ptr<Metric> m(Metric::Create(name, config_));
      if (m) {
        m->Init(train_data_->metadata(), train_data_->num_data());
        train_metrics_.push_back(std::move(m));
      }
    }
    boosting_.reset(GBDT::CreateBoosting(
        config_.boosting, config_.input_model.empty() ? nullptr
                                                      : config_.input_model.c_str()));
    std::vector<const Metric*> tm;
    for (auto& m : train_metrics_) tm.push_back(m.get());
    boosting_->Init(&config_, train_data_.get(), objective_.get(), tm);
    for (auto& vd : valid_data_) {
      std::vector<std::unique_ptr<Metric>> vms;
      for (auto& name : metric_names) {
        std::unique_ptr<Metric> m(Metric::Create(name, config_));
        if (m) {
          m->Init(vd->metadata(), vd->num_data());
          vms.push_back(std::move(m));
        }
      }
      std::vector<const Metric*> vmp;
      for (auto& m : vms) vmp.push_back(m.get());
      valid_metrics_.push_back(std::move(vms));
      boosting_->AddValidDataset(vd.get(), vmp);
    }
    Log::Info("Started training for %d iterations...", config_.num_iterations);
    double t0 = Timer::Now();
    boosting_->Train(config_.snapshot_freq, config_.output_model);
    Log::Info("Finished training in %.3f seconds", Timer::Now() - t0);
    boosting_->SaveModelToFile(0, -1, config_.saved_feature_importance_type,
                               config_.output_model.c_str());
    Log::Info("Model saved to %s", config_.output_model.c_str());
  }

  void Predict() {
    if (config_.input_model.empty()) Log::Fatal("task=predict requires input_model=");
    boosting_.reset(GBDT::CreateBoosting("gbdt", config_.input_model.c_str()));
    auto rows = LoadRawRowsForPredict(config_.data.c_str(), config_,
                                      boosting_->MaxFeatureIdx() + 1);
    int ptype = 0;
    if (config_.predict_raw_score) ptype = 1;
    if (config_.predict_leaf_index) ptype = 2;
    if (config_.predict_contrib) ptype = 3;
    const int ncol = boosting_->MaxFeatureIdx() + 1;
    const int per_row = boosting_->NumPredictOneRow(config_.start_iteration_predict,
                                                    config_.num_iteration_predict,
                                                    ptype == 2, ptype == 3);
    FILE* fp = fopen(config_.output_result.c_str(), "w");
    if (!fp) Log::Fatal("Cannot open %s", config_.output_result.c_str());
    std::vector<double> out(per_row);
    std::vector<double> feats(ncol);
    for (auto& r : rows) {
      for (int c = 0; c < ncol; ++c)
        feats[c] = c < static_cast<int>(r.size()) ? r[c] : 0.0;
      switch (ptype) {
        case 0: boosting_->Predict(feats.data(), out.data(),
                                   config_.start_iteration_predict,
                                   config_.num_iteration_predict); break;
        case 1: boosting_->PredictRaw(feats.data(), out.data(),
                                      config_.start_iteration_predict,
                                      config_.num_iteration_predict); break;
        case 2: boosting_->PredictLeafIndex(feats.data(), out.data(),
                                            config_.start_iteration_predict,
                                            config_.num_iteration_predict); break;
        case 3: boosting_->PredictContrib(feats.data(), out.data(),
                                          config_.start_iteration_predict,
                                          config_.num_iteration_predict); break;
      }
      for (int k = 0; k < per_row; ++k) {
        if (k) fputc('\t', fp);
        fprintf(fp, "%.17g", out[k]);
      }
      fputc('\n', fp);
    }
    fclose(fp);
    Log::Info("Predictions written to %s", config_.output_result.c_str());
  }

  void Refit() {
    if (config_.input_model.empty()) Log::Fatal("task=refit requires input_model=");
    LoadData();
    boosting_.reset(GBDT::CreateBoosting("gbdt", config_.input_model.c_str()));
    boosting_->ResetConfig(&config_);  // file-loaded boosters carry no config
    objective_.reset(ObjectiveFunction::Create(boosting_->ObjectiveName(), config_));
    if (objective_) objective_->Init(train_data_->metadata(), train_data_->num_data());
    std::vector<const Metric*> none;
    boosting_->ResetTrainingData(train_data_.get(), objective_.get(), none);
    // leaf assignments of the current model on the new data
    const int ntrees = boosting_->NumberOfTotalModel();
    const data_size_t n = train_data_->num_data();
    std::vector<int32_t> leaf_preds(static_cast<size_t>(n) * ntrees);
    const int ncol = boosting_->MaxFeatureIdx() + 1;
    auto rows = LoadRawRowsForPredict(config_.data.c_str(), config_, ncol);
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < n; ++i) {
      std::vector<double> feats(ncol, 0.0);
      for (int c = 0; c < ncol && c < static_cast<int>(rows[i].size()); ++c)
        feats[c] = rows[i][c];
      std::vector<double> out(ntrees);
      boosting_->PredictLeafIndex(feats.data(), out.data(), 0, -1);
      for (int t = 0; t < ntrees; ++t)
        leaf_preds[static_cast<size_t>(i) * ntrees + t] = static_cast<int32_t>(out[t]);
    }
    boosting_->RefitTree(leaf_preds.data(), n, ntrees);
    boosting_->SaveModelToFile(0, -1, config_.saved_feature_importance_type,
                               config_.output_model.c_str());
    Log::Info("Refitted model saved to %s", config_.output_model.c_str());
  }

  void ConvertModel() {
    if (config_.input_model.empty()) Log::Fatal("convert_model requires input_model=");
    boosting_.reset(GBDT::CreateBoosting("gbdt", config_.input_model.c_str()));
    std::string text;
    std::string out;
    if (config_.convert_model_language == "cpp") {
      text = boosting_->ModelToIfElse(-1);
      out = config_.convert_model.empty() ? "gbdt_prediction.cpp" : config_.convert_model;
    } else {
      text = boosting_->DumpModel(0, -1, 0);
      out = config_.output_model.empty() ? "model.json" : config_.output_model;
    }
    FILE* fp = fopen(out.c_str(), "w");
    if (!fp) Log::Fatal("Cannot open %s", out.c_str());
    fwrite(text.data(), 1, text.size(), fp);
    fclose(fp);
    Log::Info("Converted model written to %s", out.c_str());
  }

  Config config_;
  std::unique_ptr<Dataset> train_data_;
  std::vector<std::unique_ptr<Dataset>> valid_data_;
  std::unique_ptr<ObjectiveFunction> objective_;
  std::vector<std::unique_ptr<Metric>> train_metrics_;
  std::vector<std::vector<std::unique_ptr<Metric>>> valid_metrics_;
  std::unique_ptr<GBDT> boosting_;
};

}  // namespace migbm

int main(int argc, char** argv) {
  try {
    migbm::Application app(argc, argv);
    app.Run();
  } catch (const std::exception& ex) {
    fprintf(stderr, "%s\n", ex.what());
    return 1;
  }
  return 0;
}
