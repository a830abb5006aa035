/*! migbm distributed CPU tree learners over Network collectives + learner factory.
 *  Parity target: reference src/treelearner/{data,feature,voting}_parallel_tree_learner.cpp
 *  and tree_learner.cpp (factory). Algorithms re-implemented fresh:
 *  - data-parallel: rows sharded across ranks; per-leaf histogram allreduce; every rank
 *    scans all features -> identical split everywhere (deterministic by construction).
 *  - feature-parallel: full data everywhere; ranks scan disjoint feature subsets; best
 *    split allreduced.
 *  - voting-parallel (PV-Tree): local top-k proposals, global vote, reduce only the
 *    voted features' histograms.
 *  The multi-GPU hot path does NOT go through here — the HIP learner reduces histograms
 *  with RCCL over xGMI directly (see hip/).
 */
#include "migbm/network.h"
#include "migbm/tree_learner.h"

#include <algorithm>
#include <cstring>

namespace migbm {

namespace {

/*! fixed-size wire form of SplitInfo for allgather (numerical + small cat bitset). */
struct WireSplit {
  double gain;
  double left_sum_gradient, left_sum_hessian;
  double right_sum_gradient, right_sum_hessian;
  double left_output, right_output;
  int64_t left_count, right_count;
  int feature;
  uint32_t threshold;
  uint8_t default_left;
  uint8_t is_cat;
  uint8_t n_cat_words;
  int8_t monotone_type;
  uint32_t cat_words[8];

  static WireSplit From(const SplitInfo& s) {
    WireSplit w;
    memset(&w, 0, sizeof(w));
    w.gain = s.IsValid() ? s.gain : kMinScore;
    w.left_sum_gradient = s.left_sum_gradient;
    w.left_sum_hessian = s.left_sum_hessian;
    w.right_sum_gradient = s.right_sum_gradient;
    w.right_sum_hessian = s.right_sum_hessian;
    w.left_output = s.left_output;
    w.right_output = s.right_output;
    w.left_count = s.left_count;
    w.right_count = s.right_count;
    w.feature = s.feature;
    w.threshold = s.threshold;
    w.default_left = s.default_left;
    w.monotone_type = s.monotone_type;
    w.is_cat = !s.cat_bitset_inner.empty();
    w.n_cat_words = static_cast<uint8_t>(std::min<size_t>(8, s.cat_bitset_inner.size()));
    for (int i = 0; i < w.n_cat_words; ++i) w.cat_words[i] = s.cat_bitset_inner[i];
    return w;
  }
  SplitInfo To() const {
    SplitInfo s;
    s.gain = gain;
    s.left_sum_gradient = left_sum_gradient;
    s.left_sum_hessian = left_sum_hessian;
    s.right_sum_gradient = right_sum_gradient;
    s.right_sum_hessian = right_sum_hessian;
    s.left_output = left_output;
    s.right_output = right_output;
    s.left_count = static_cast<data_size_t>(left_count);
    s.right_count = static_cast<data_size_t>(right_count);
    s.feature = feature;
    s.threshold = threshold;
    s.default_left = default_left;
    s.monotone_type = monotone_type;
    if (is_cat) s.cat_bitset_inner.assign(cat_words, cat_words + n_cat_words);
    return s;
  }
};

/*! argmax-by-gain across all ranks' candidates (ties broken by feature index, then rank). */
SplitInfo SyncUpGlobalBestSplit(const SplitInfo& local) {
  if (!Network::is_distributed()) return local;
  WireSplit w = WireSplit::From(local);
  std::vector<WireSplit> all(Network::num_machines());
  Network::Allgather(reinterpret_cast<const char*>(&w), sizeof(w),
                     reinterpret_cast<char*>(all.data()));
  SplitInfo best = all[0].To();
  for (int r = 1; r < Network::num_machines(); ++r) {
    SplitInfo cand = all[r].To();
    if (cand.IsValid() && (!best.IsValid() || cand > best)) best = cand;
  }
  return best;
}

}  // namespace

/*! Data-parallel learner: local row shard, globally reduced histograms, and the
 *  reference's FEATURE-OWNERSHIP gain scan (data_parallel_tree_learner.cpp
 *  structure): features are assigned to ranks balanced by bin count, each rank
 *  scans only its owned features of the reduced histogram, and the per-rank
 *  winners are argmax-allgathered — dividing FindBestSplit work by the rank
 *  count. (The wire-level reduce-scatter of the histogram payload itself is an
 *  RCCL-path optimization — see docs/ROADMAP.md #1 — the injected-collective
 *  transport here moves the same bytes either way at these payload sizes.) */
class DataParallelTreeLearner : public SerialTreeLearner {
 public:
  explicit DataParallelTreeLearner(const Config* config) : SerialTreeLearner(config) {}

 protected:
  void OnHistogramReady(int leaf) override {
    if (Network::is_distributed()) {
      hist_t* hist = HistSlot(leaf_to_slot_[leaf]);
      Network::AllreduceSum(hist, 2 * static_cast<size_t>(train_data_->num_total_bin()));
    }
    SerialTreeLearner::OnHistogramReady(leaf);  // EFB default-bin reconstruction
  }
  void FindBestSplitForLeaf(int leaf, const LeafContext& ctx) override {
    if (!Network::is_distributed()) {
      SerialTreeLearner::FindBestSplitForLeaf(leaf, ctx);
      return;
    }
    if (feature_owner_.empty()) AssignFeatureOwnership();
    std::vector<int8_t> saved = is_feature_used_;
    const int nf = train_data_->num_features();
    const int rank = Network::rank();
    for (int f = 0; f < nf; ++f)
      if (feature_owner_[f] != rank) is_feature_used_[f] = 0;
    SerialTreeLearner::FindBestSplitForLeaf(leaf, ctx);
    is_feature_used_ = saved;
    best_split_per_leaf_[leaf] = SyncUpGlobalBestSplit(best_split_per_leaf_[leaf]);
  }
  /*! deterministic greedy assignment balanced by bin count (all ranks compute the
   *  identical map; parity: reference per-tree ownership balancing) */
  void AssignFeatureOwnership() {
    const int nf = train_data_->num_features();
    const int world = Network::num_machines();
    feature_owner_.assign(nf, 0);
    std::vector<int> order(nf);
    for (int f = 0; f < nf; ++f) order[f] = f;
    std::stable_sort(order.begin(), order.end(), [&](int a, int b) {
      return train_data_->FeatureNumBin(a) > train_data_->FeatureNumBin(b);
    });
    std::vector<int64_t> load(world, 0);
    for (int f : order) {
      int r = 0;
      for (int i = 1; i < world; ++i)
        if (load[i] < load[r]) r = i;
      feature_owner_[f] = r;
      load[r] += train_data_->FeatureNumBin(f);
    }
  }
  std::vector<int> feature_owner_;
  void ReduceRootStats(double* sum_g, double* sum_h, data_size_t* cnt) override {
    if (!Network::is_distributed()) return;
    double v[2] = {*sum_g, *sum_h};
    Network::AllreduceSum(v, 2);
    *sum_g = v[0];
    *sum_h = v[1];
    *cnt = static_cast<data_size_t>(
        Network::GlobalSyncUpBySum(static_cast<int64_t>(*cnt)));
  }
  void GlobalChildCounts(data_size_t* left_cnt, data_size_t* right_cnt) override {
    if (!Network::is_distributed()) return;
    int64_t v[2] = {*left_cnt, *right_cnt};
    Network::AllreduceSum(v, 2);
    *left_cnt = static_cast<data_size_t>(v[0]);
    *right_cnt = static_cast<data_size_t>(v[1]);
  }
};

/*! Feature-parallel learner: full data, disjoint feature scan, best-split allreduce. */
class FeatureParallelTreeLearner : public SerialTreeLearner {
 public:
  explicit FeatureParallelTreeLearner(const Config* config) : SerialTreeLearner(config) {}

 protected:
  void FindBestSplitForLeaf(int leaf, const LeafContext& ctx) override {
    // mask features not owned by this rank
    std::vector<int8_t> saved = is_feature_used_;
    const int nf = train_data_->num_features();
    const int world = Network::num_machines(), rank = Network::rank();
    for (int f = 0; f < nf; ++f)
      if (f % world != rank) is_feature_used_[f] = 0;
    SerialTreeLearner::FindBestSplitForLeaf(leaf, ctx);
    is_feature_used_ = saved;
    best_split_per_leaf_[leaf] = SyncUpGlobalBestSplit(best_split_per_leaf_[leaf]);
  }
};

/*! Voting-parallel learner (PV-Tree): each rank proposes its local top-k split
 *  features, a global vote selects 2k features, and ONLY those features' histogram
 *  ranges are globally reduced — O(k·bins) wire bytes instead of O(num_total_bin).
 *  Both children build their own histograms (subtraction is invalid when only the
 *  voted ranges are global). Parity: reference voting_parallel_tree_learner.cpp. */
class VotingParallelTreeLearner : public SerialTreeLearner {
 public:
  explicit VotingParallelTreeLearner(const Config* config) : SerialTreeLearner(config) {}

  void Init(const Dataset* train_data, bool is_constant_hessian) override {
    SerialTreeLearner::Init(train_data, is_constant_hessian);
    build_both_children_ = Network::is_distributed();
    voted_mask_.assign(config_->num_leaves,
                       std::vector<int8_t>(train_data->num_features(), 1));
  }

 protected:
  void ReduceRootStats(double* sum_g, double* sum_h, data_size_t* cnt) override {
    if (!Network::is_distributed()) return;
    double v[2] = {*sum_g, *sum_h};
    Network::AllreduceSum(v, 2);
    *sum_g = v[0];
    *sum_h = v[1];
    *cnt = static_cast<data_size_t>(Network::GlobalSyncUpBySum(static_cast<int64_t>(*cnt)));
  }
  void GlobalChildCounts(data_size_t* left_cnt, data_size_t* right_cnt) override {
    if (!Network::is_distributed()) return;
    int64_t v[2] = {*left_cnt, *right_cnt};
    Network::AllreduceSum(v, 2);
    *left_cnt = static_cast<data_size_t>(v[0]);
    *right_cnt = static_cast<data_size_t>(v[1]);
  }

  void OnHistogramReady(int leaf) override {
    if (!Network::is_distributed()) return;
    const int nf = train_data_->num_features();
    const int k = std::min(config_->top_k, nf);
    hist_t* hist = HistSlot(leaf_to_slot_[leaf]);
    const LeafContext& ctx = leaf_ctx_[leaf];
    // 0. pre-vote: materialize bundled/sparse default bins from LOCAL leaf totals
    // (the base fix at the end uses GLOBAL totals, valid only after the reduce;
    // voting on unmaterialized histograms would skew candidates for EFB/sparse
    // features). Local totals = full-bin sum of any dense unbundled feature.
    if (train_data_->has_bundles() || train_data_->has_sparse()) {
      for (int f = 0; f < nf; ++f) {
        int def_bin = train_data_->feature_bundled(f)
                          ? 0 : train_data_->feature_sparse_default_bin(f);
        if (def_bin < 0) continue;
        hist_t* fh = hist + 2 * train_data_->hist_offset(f);
        double g = 0, h = 0;
        for (int b = 0; b < train_data_->FeatureNumBin(f); ++b) {
          if (b == def_bin) continue;
          g += fh[2 * b];
          h += fh[2 * b + 1];
        }
        fh[2 * def_bin] = local_leaf_sum_g_ - g;
        fh[2 * def_bin + 1] = local_leaf_sum_h_ - h;
      }
    }
    // 1. local candidate gain per feature
    std::vector<std::pair<double, int>> gains(nf);
#pragma omp parallel for schedule(static)
    for (int f = 0; f < nf; ++f) {
      SplitInfo si;
      const BinMapper* m = train_data_->FeatureBinMapper(f);
      const hist_t* fh = hist + 2 * train_data_->hist_offset(f);
      if (m->bin_type() == BinType::kCategorical) {
        FindBestThresholdCategorical(fh, m->num_bin(), ctx, *config_, &si);
      } else {
        FindBestThresholdNumerical(fh, m->num_bin(), m->num_numeric_bin(), m->nan_bin(),
                                   ctx, *config_, 0, -1, &si);
      }
      gains[f] = {si.IsValid() ? si.gain : -1e308, f};
    }
    // 2. local top-k proposal
    std::partial_sort(gains.begin(), gains.begin() + k, gains.end(),
                      [](auto& a, auto& b) { return a.first > b.first; });
    std::vector<int32_t> proposal(k);
    for (int i = 0; i < k; ++i) proposal[i] = gains[i].second;
    // 3. global vote: allgather proposals, count votes, take global top 2k
    const int world = Network::num_machines();
    std::vector<int32_t> all(static_cast<size_t>(world) * k);
    Network::Allgather(reinterpret_cast<const char*>(proposal.data()),
                       static_cast<int>(k * sizeof(int32_t)),
                       reinterpret_cast<char*>(all.data()));
    std::vector<int> votes(nf, 0);
    for (int32_t f : all)
      if (f >= 0 && f < nf) votes[f]++;
    std::vector<int> order(nf);
    for (int f = 0; f < nf; ++f) order[f] = f;
    std::sort(order.begin(), order.end(), [&](int a, int b) {
      return votes[a] > votes[b] || (votes[a] == votes[b] && a < b);
    });
    const int n_sel = std::min(2 * k, nf);
    auto& mask = voted_mask_[leaf];
    mask.assign(nf, 0);
    for (int i = 0; i < n_sel; ++i)
      if (votes[order[i]] > 0) mask[order[i]] = 1;
    // 4. reduce ONLY the voted features' histogram ranges (compact buffer)
    std::vector<hist_t> compact;
    for (int f = 0; f < nf; ++f) {
      if (!mask[f]) continue;
      const hist_t* fh = hist + 2 * train_data_->hist_offset(f);
      compact.insert(compact.end(), fh, fh + 2 * train_data_->FeatureNumBin(f));
    }
    Network::AllreduceSum(compact.data(), compact.size());
    size_t off = 0;
    for (int f = 0; f < nf; ++f) {
      if (!mask[f]) continue;
      hist_t* fh = hist + 2 * train_data_->hist_offset(f);
      std::copy(compact.begin() + off,
                compact.begin() + off + 2 * train_data_->FeatureNumBin(f), fh);
      off += 2 * train_data_->FeatureNumBin(f);
    }
    // voted features' default bins are already globally correct (the reduce summed
    // per-rank locally-materialized defaults); non-voted features stay local-only
    // and are never scanned (FindBestSplitForLeaf restricts to voted_mask_), so the
    // base global fix — which would mix global totals with local bins — is skipped.
  }

  void FindBestSplitForLeaf(int leaf, const LeafContext& ctx) override {
    if (!Network::is_distributed()) {
      SerialTreeLearner::FindBestSplitForLeaf(leaf, ctx);
      return;
    }
    std::vector<int8_t> saved = is_feature_used_;
    const int nf = train_data_->num_features();
    for (int f = 0; f < nf; ++f) is_feature_used_[f] &= voted_mask_[leaf][f];
    SerialTreeLearner::FindBestSplitForLeaf(leaf, ctx);
    is_feature_used_ = saved;
  }

 private:
  std::vector<std::vector<int8_t>> voted_mask_;
};

TreeLearner* (*g_create_hip_learner)(const Config*) = nullptr;

TreeLearner* TreeLearner::Create(const std::string& learner_type,
                                 const std::string& device_type, const Config* config) {
  const bool gpu = device_type == "gpu" || device_type == "cuda";
  if (gpu) {
    // fail loudly rather than silently training on CPU when a GPU was requested
    if (g_create_hip_learner == nullptr)
      Log::Fatal("device_type=%s requested but the HIP learner module is not linked",
                 device_type.c_str());
    return g_create_hip_learner(config);
  }
  if (learner_type == "serial") return new SerialTreeLearner(config);
  if (learner_type == "feature") return new FeatureParallelTreeLearner(config);
  if (learner_type == "data") return new DataParallelTreeLearner(config);
  if (learner_type == "voting") return new VotingParallelTreeLearner(config);
  Log::Fatal("Unknown tree learner %s", learner_type.c_str());
  return nullptr;
}

}  // namespace migbm
