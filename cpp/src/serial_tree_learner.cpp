/*! migbm SerialTreeLearner — CPU leaf-wise (best-first) histogram learner with histogram
 *  subtraction. Parity target: reference src/treelearner/serial_tree_learner.cpp (Train,
 *  FindBestSplits, Split) — algorithm re-implemented fresh. */
#include "migbm/tree_learner.h"
#include "migbm/network.h"

#include <chrono>
#include "migbm/objective.h"

#include <algorithm>
#include <fstream>
#include <numeric>
#include <set>

namespace migbm {

// ------------------------------------------------------------------ DataPartition
void DataPartition::Split(int leaf, int right_leaf,
                          const std::function<bool(data_size_t)>& go_left) {
  const data_size_t begin = leaf_begin_[leaf];
  const data_size_t cnt = leaf_count_[leaf];
  data_size_t* idx = indices_.data() + begin;
  data_size_t* tmp = temp_.data() + begin;

  const int max_threads = omp_get_max_threads();
  int nblock;
  data_size_t bsize;
  Threading::BlockInfo<data_size_t>(max_threads, cnt, 1024, &nblock, &bsize);
  std::vector<data_size_t> left_cnts(nblock, 0);
#pragma omp parallel for schedule(static, 1)
  for (int b = 0; b < nblock; ++b) {
    data_size_t s = b * bsize, e = std::min(cnt, s + bsize);
    data_size_t lc = 0;
    for (data_size_t i = s; i < e; ++i) lc += go_left(idx[i]) ? 1 : 0;
    left_cnts[b] = lc;
  }
  std::vector<data_size_t> loff(nblock + 1, 0), roff(nblock + 1, 0);
  for (int b = 0; b < nblock; ++b) loff[b + 1] = loff[b] + left_cnts[b];
  const data_size_t total_left = loff[nblock];
  roff[0] = total_left;
  for (int b = 0; b < nblock; ++b) {
    data_size_t s = b * bsize, e = std::min(cnt, s + bsize);
    roff[b + 1] = roff[b] + (e - s) - left_cnts[b];
  }
#pragma omp parallel for schedule(static, 1)
  for (int b = 0; b < nblock; ++b) {
    data_size_t s = b * bsize, e = std::min(cnt, s + bsize);
    data_size_t lp = loff[b], rp = roff[b];
    for (data_size_t i = s; i < e; ++i) {
      if (go_left(idx[i])) tmp[lp++] = idx[i];
      else tmp[rp++] = idx[i];
    }
  }
  std::copy(tmp, tmp + cnt, idx);
  leaf_count_[leaf] = total_left;
  leaf_begin_[right_leaf] = begin + total_left;
  leaf_count_[right_leaf] = cnt - total_left;
}

void DataPartition::SplitDenseU8(int leaf, int right_leaf, const uint8_t* col, uint32_t thr,
                                 int nan_bin, bool default_left) {
  const data_size_t begin = leaf_begin_[leaf];
  const data_size_t cnt = leaf_count_[leaf];
  data_size_t* idx = indices_.data() + begin;
  data_size_t* tmp = temp_.data() + begin;
  const int max_threads = omp_get_max_threads();
  int nblock;
  data_size_t bsize;
  Threading::BlockInfo<data_size_t>(max_threads, cnt, 1024, &nblock, &bsize);
  std::vector<data_size_t> left_cnts(nblock, 0);
  const uint8_t nb = nan_bin >= 0 ? static_cast<uint8_t>(nan_bin) : 255;
  const bool has_nan = nan_bin >= 0;
#pragma omp parallel for schedule(static, 1)
  for (int b = 0; b < nblock; ++b) {
    data_size_t s = b * bsize, e = std::min(cnt, s + bsize);
    data_size_t lc = 0;
    if (!has_nan) {
      for (data_size_t i = s; i < e; ++i) lc += col[idx[i]] <= thr ? 1 : 0;
    } else {
      for (data_size_t i = s; i < e; ++i) {
        const uint8_t v = col[idx[i]];
        lc += (v == nb ? default_left : v <= thr) ? 1 : 0;
      }
    }
    left_cnts[b] = lc;
  }
  std::vector<data_size_t> loff(nblock + 1, 0), roff(nblock + 1, 0);
  for (int b = 0; b < nblock; ++b) loff[b + 1] = loff[b] + left_cnts[b];
  const data_size_t total_left = loff[nblock];
  roff[0] = total_left;
  for (int b = 0; b < nblock; ++b) {
    data_size_t s = b * bsize, e = std::min(cnt, s + bsize);
    roff[b + 1] = roff[b] + (e - s) - left_cnts[b];
  }
#pragma omp parallel for schedule(static, 1)
  for (int b = 0; b < nblock; ++b) {
    data_size_t s = b * bsize, e = std::min(cnt, s + bsize);
    data_size_t lw = loff[b], rw = roff[b];
    if (!has_nan) {
      for (data_size_t i = s; i < e; ++i) {
        const data_size_t r = idx[i];
        if (col[r] <= thr) tmp[lw++] = r;
        else tmp[rw++] = r;
      }
    } else {
      for (data_size_t i = s; i < e; ++i) {
        const data_size_t r = idx[i];
        const uint8_t v = col[r];
        if (v == nb ? default_left : v <= thr) tmp[lw++] = r;
        else tmp[rw++] = r;
      }
    }
  }
  std::copy(tmp, tmp + cnt, idx);
  leaf_count_[leaf] = total_left;
  leaf_begin_[right_leaf] = begin + total_left;
  leaf_count_[right_leaf] = cnt - total_left;
}

std::unique_ptr<ForcedNode> ParseForcedSplits(const std::string& path) {
  if (path.empty()) return nullptr;
  std::ifstream jf(path);
  if (!jf.good()) {
    Log::Warning("Cannot open forced splits file %s", path.c_str());
    return nullptr;
  }
  std::string content((std::istreambuf_iterator<char>(jf)),
                      std::istreambuf_iterator<char>());
  size_t pos = 0;
  std::function<std::unique_ptr<ForcedNode>()> parse = [&]() -> std::unique_ptr<ForcedNode> {
    auto skip = [&]() { while (pos < content.size() && isspace(content[pos])) ++pos; };
    skip();
    if (pos >= content.size() || content[pos] != '{') return nullptr;
    ++pos;
    auto node = std::make_unique<ForcedNode>();
    while (pos < content.size() && content[pos] != '}') {
      skip();
      if (content[pos] == ',') { ++pos; continue; }
      if (content[pos] != '"') break;
      size_t kend = content.find('"', pos + 1);
      std::string key = content.substr(pos + 1, kend - pos - 1);
      pos = content.find(':', kend) + 1;
      skip();
      if (key == "feature") node->feature = atoi(content.c_str() + pos);
      else if (key == "threshold") node->threshold = atof(content.c_str() + pos);
      if (key == "left") node->left = parse();
      else if (key == "right") node->right = parse();
      else {  // skip number token
        while (pos < content.size() && content[pos] != ',' && content[pos] != '}') ++pos;
        continue;
      }
      skip();
    }
    if (pos < content.size() && content[pos] == '}') ++pos;
    return node;
  };
  auto root = parse();
  if (root) Log::Info("Loaded forced splits from %s", path.c_str());
  return root;
}

// ------------------------------------------------------------------ SerialTreeLearner
void SerialTreeLearner::Init(const Dataset* train_data, bool is_constant_hessian) {
  train_data_ = train_data;
  is_constant_hessian_ = is_constant_hessian;
  const data_size_t n = train_data_->num_data();
  partition_.Init(n, config_->num_leaves);
  ordered_grad_.resize(n);
  ordered_hess_.resize(n);
  {
    // histogram_pool_size (MB) caps the slot count (reference HistogramPool);
    // <= 0 means unlimited = one slot per leaf
    const double cap_mb = config_->histogram_pool_size;
    const size_t slot_bytes =
        static_cast<size_t>(2) * train_data_->num_total_bin() * sizeof(hist_t);
    int slots = config_->num_leaves;
    if (cap_mb > 0) {
      const int64_t fit = static_cast<int64_t>(cap_mb * 1024.0 * 1024.0 / slot_bytes);
      slots = static_cast<int>(std::max<int64_t>(3, std::min<int64_t>(slots, fit)));
    }
    pool_slots_ = slots;
    hist_store_.resize(static_cast<size_t>(slots) * 2 * train_data_->num_total_bin());
    slot_owner_.assign(slots, -1);
    slot_used_.assign(slots, 0);
  }
  leaf_to_slot_.resize(config_->num_leaves);
  best_split_per_leaf_.resize(config_->num_leaves);
  leaf_ctx_.resize(config_->num_leaves);
  feature_rng_ = Random(config_->feature_fraction_seed);
  extra_rng_ = Random(config_->extra_seed);
  cegb_feature_used_.assign(train_data_->num_total_features(), 0);
  forced_of_leaf_.assign(config_->num_leaves, nullptr);
  forced_root_.reset();
  forced_root_ = ParseForcedSplits(config_->forcedsplits_filename);
  leaf_branch_features_.assign(config_->num_leaves, {});
  interaction_groups_.clear();
  if (!config_->interaction_constraints.empty()) {
    // format: "[0,1,2],[2,3]"
    std::string sdef = config_->interaction_constraints;
    size_t pos = 0;
    while ((pos = sdef.find('[', pos)) != std::string::npos) {
      size_t end = sdef.find(']', pos);
      if (end == std::string::npos) break;
      std::set<int> grp;
      for (auto& tok : Common::Split(sdef.substr(pos + 1, end - pos - 1).c_str(), ',')) {
        auto t = Common::Trim(tok);
        if (!t.empty()) grp.insert(atoi(t.c_str()));
      }
      if (!grp.empty()) interaction_groups_.push_back(std::move(grp));
      pos = end + 1;
    }
  }
}

void SerialTreeLearner::ResetTrainingData(const Dataset* train_data) {
  Init(train_data, is_constant_hessian_);
}

void SerialTreeLearner::SetBaggingData(const Dataset* subset, const data_size_t* used_indices,
                                       data_size_t num_data) {
  if (subset != nullptr) {
    ResetTrainingData(subset);
    bag_indices_ = nullptr;
    bag_cnt_ = 0;
  } else {
    bag_indices_ = used_indices;
    bag_cnt_ = num_data;
  }
}

std::vector<int8_t> SerialTreeLearner::SampleFeatures(bool per_node) {
  const int nf = train_data_->num_features();
  std::vector<int8_t> used(nf, 1);
  double frac = per_node ? config_->feature_fraction_bynode : config_->feature_fraction;
  if (frac >= 1.0) return used;
  int k = std::max(1, static_cast<int>(nf * frac));
  std::fill(used.begin(), used.end(), 0);
  auto sel = feature_rng_.Sample(nf, k);
  for (int f : sel) used[f] = 1;
  return used;
}

static bool MonoDebug() {
  static const bool v = getenv("MIGBM_MONO_DEBUG") != nullptr;
  return v;
}

int SerialTreeLearner::AcquireSlot(int leaf, int pin_a, int pin_b) {
  if (leaf_to_slot_[leaf] >= 0) {
    slot_used_[leaf_to_slot_[leaf]] = ++slot_clock_;
    return leaf_to_slot_[leaf];
  }
  int victim = -1;
  int64_t oldest = std::numeric_limits<int64_t>::max();
  for (int s2 = 0; s2 < pool_slots_; ++s2) {
    if (slot_owner_[s2] < 0) {
      victim = s2;
      break;
    }
    if (slot_owner_[s2] == pin_a || slot_owner_[s2] == pin_b) continue;
    if (slot_used_[s2] < oldest) {
      oldest = slot_used_[s2];
      victim = s2;
    }
  }
  MIGBM_CHECK(victim >= 0);
  if (slot_owner_[victim] >= 0) leaf_to_slot_[slot_owner_[victim]] = -1;
  slot_owner_[victim] = leaf;
  leaf_to_slot_[leaf] = victim;
  slot_used_[victim] = ++slot_clock_;
  return victim;
}

void SerialTreeLearner::ComputeHistogram(int leaf, data_size_t cnt,
                                         const data_size_t* indices) {
  // ordered-gradient gather, deferred: the row-wise histogram reads gradients by
  // row id directly and skips this pass entirely
  bool gathered = false;
  auto gather = [&]() {
    if (gathered) return;
    gathered = true;
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < cnt; ++i) {
      ordered_grad_[i] = gradients_[indices[i]];
      ordered_hess_[i] = hessians_[indices[i]];
    }
  };
  Timer::Global().Start("hist");
  struct HistPhaseGuard {
    ~HistPhaseGuard() { Timer::Global().Stop("hist"); }
  } hist_phase_guard;
  hist_t* hist = HistSlot(AcquireSlot(leaf));
  std::fill(hist, hist + 2 * train_data_->num_total_bin(), 0.0);
  // ---- histogram mode: empirical col-wise vs row-wise choice, reference
  // TrainingShareStates-style. Row-wise is only a candidate when every feature
  // is in play and the dataset is dense (the row view materializes all bins).
  const bool row_possible = !train_data_->has_sparse() &&
                            config_->feature_fraction >= 1.0 &&
                            !config_->force_col_wise &&
                            train_data_->num_features() >= 2;
  if (hist_mode_ < 0) {
    if (config_->force_row_wise && row_possible) hist_mode_ = 1;
    else if (!row_possible) hist_mode_ = 0;
  }
  const bool big_leaf = cnt * 2 >= train_data_->num_data();
  if (hist_mode_ < 0 && big_leaf) {
    // trial: time this (root) histogram in the mode not yet measured
    const int mode = hist_trials_done_;  // 0 = col first, 1 = row second
    const auto t0 = std::chrono::steady_clock::now();
    if (mode == 0) {
      gather();
      train_data_->ConstructHistograms(is_feature_used_, indices, cnt,
                                       ordered_grad_.data(), ordered_hess_.data(), hist);
    } else {
      train_data_->ConstructHistogramsRowWise(indices, cnt, gradients_, hessians_, hist,
                                              /*row_indexed=*/true);
    }
    hist_trial_time_[mode] =
        std::chrono::duration<double>(std::chrono::steady_clock::now() - t0).count();
    if (++hist_trials_done_ >= 2)
      hist_mode_ = hist_trial_time_[1] < hist_trial_time_[0] ? 1 : 0;
    return;
  }
  if (hist_mode_ == 1 || (hist_mode_ < 0 && hist_trials_done_ == 1)) {
    // locked row-wise — but SMALL leaves pay row-wise's fixed per-call cost
    // (per-thread private-histogram zero + merge) without amortizing it, so they
    // drop to the col-wise path (hybrid, like the reference's per-leaf choice)
    static const int kRowWiseMinRows = [] {
      const char* e = getenv("MIGBM_ROWWISE_MIN_ROWS");
      return e ? atoi(e) : -1;
    }();
    const data_size_t min_rows = kRowWiseMinRows >= 0
                                     ? kRowWiseMinRows
                                     : train_data_->num_total_bin();
    if (cnt >= min_rows) {
      if (tree_const_hess_ < 0) {
        // first row-wise call of this tree is the root: classify the hessians
        // once (constant -> count mode) and, if varying, build the interleaved
        // (g,h) pair array so the hot loop does ONE 8B load per row
        const score_t h0 = hessians_[indices[0]];
        bool ok = true;
#pragma omp parallel for schedule(static) reduction(&& : ok)
        for (data_size_t i = 0; i < cnt; ++i) ok = ok && hessians_[indices[i]] == h0;
        tree_const_hess_ = ok ? 1 : 0;
        if (!ok) {
          // interleave over ALL rows: the first row-wise call of a tree may be
          // a leaf subset (histogram trial phase), and later leaves read gh_ at
          // their own row ids
          const data_size_t n_all = train_data_->num_data();
          gh_.resize(2 * static_cast<size_t>(n_all));
#pragma omp parallel for schedule(static)
          for (data_size_t r = 0; r < n_all; ++r) {
            gh_[2 * static_cast<size_t>(r)] = gradients_[r];
            gh_[2 * static_cast<size_t>(r) + 1] = hessians_[r];
          }
        }
      }
      if (tree_const_hess_ == 1) {
        train_data_->ConstructHistogramsRowWise(indices, cnt, gradients_, hessians_, hist,
                                                /*row_indexed=*/true);
      } else {
        train_data_->ConstructHistogramsRowWiseGH(indices, cnt, gh_.data(), hist);
      }
      return;
    }
  }
  if ((train_data_->has_bundles() || train_data_->has_sparse()) &&
      Network::is_distributed()) {
    double sg = 0.0, sh = 0.0;
#pragma omp parallel for schedule(static) reduction(+ : sg, sh)
    for (data_size_t i = 0; i < cnt; ++i) {
      sg += gradients_[indices[i]];
      sh += hessians_[indices[i]];
    }
    local_leaf_sum_g_ = sg;
    local_leaf_sum_h_ = sh;
  }
  gather();
  if (train_data_->has_sparse()) {
    // membership mask for the sparse-column nonzero scan; cleared after use so
    // the buffer never needs a full memset
    if (in_leaf_mask_.size() != static_cast<size_t>(train_data_->num_data()))
      in_leaf_mask_.assign(train_data_->num_data(), 0);
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < cnt; ++i) in_leaf_mask_[indices[i]] = 1;
    train_data_->ConstructHistograms(is_feature_used_, indices, cnt, ordered_grad_.data(),
                                     ordered_hess_.data(), gradients_, hessians_,
                                     in_leaf_mask_.data(), hist);
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < cnt; ++i) in_leaf_mask_[indices[i]] = 0;
  } else {
    train_data_->ConstructHistograms(is_feature_used_, indices, cnt, ordered_grad_.data(),
                                     ordered_hess_.data(), hist);
  }
}

void SerialTreeLearner::SubtractHistogram(int /*dst_leaf*/, int parent_slot, int sibling_slot) {
  hist_t* parent = HistSlot(parent_slot);
  const hist_t* sib = HistSlot(sibling_slot);
  const size_t n = 2 * static_cast<size_t>(train_data_->num_total_bin());
#pragma omp parallel for schedule(static)
  for (int64_t i = 0; i < static_cast<int64_t>(n); ++i) parent[i] -= sib[i];
}

void SerialTreeLearner::FindBestSplitForLeaf(int leaf, const LeafContext& ctx) {
  Timer::Global().Start("find_best_split");
  struct SplitPhaseGuard {
    ~SplitPhaseGuard() { Timer::Global().Stop("find_best_split"); }
  } split_phase_guard;
  SplitInfo& best = best_split_per_leaf_[leaf];
  best.Reset();
  if (config_->max_depth > 0 && ctx.depth >= config_->max_depth) return;
  if (ctx.num_data < 2 * config_->min_data_in_leaf) return;
  const hist_t* hist = HistSlot(leaf_to_slot_[leaf]);
  const int nf = train_data_->num_features();
  std::vector<int8_t> node_mask;
  const std::vector<int8_t>* mask = &is_feature_used_;
  if (config_->feature_fraction_bynode < 1.0 || !interaction_groups_.empty()) {
    node_mask = config_->feature_fraction_bynode < 1.0 ? SampleFeatures(true)
                                                       : std::vector<int8_t>(nf, 1);
    for (int f = 0; f < nf; ++f) node_mask[f] &= is_feature_used_[f];
    if (!interaction_groups_.empty()) {
      // interaction constraints: feature f allowed iff some group contains the
      // leaf's branch features plus f (parity: reference ColSampler constraints)
      const auto& used = leaf_branch_features_[leaf];
      for (int f = 0; f < nf; ++f) {
        if (!node_mask[f]) continue;
        const int orig = train_data_->RealFeatureIndex(f);
        bool ok = false;
        for (const auto& grp : interaction_groups_) {
          if (!grp.count(orig)) continue;
          bool covers = true;
          for (int u : used) {
            if (!grp.count(u)) { covers = false; break; }
          }
          if (covers) { ok = true; break; }
        }
        node_mask[f] = ok ? 1 : 0;
      }
    }
    mask = &node_mask;
  }
  std::vector<SplitInfo> cand(nf);
#pragma omp parallel for schedule(static)
  for (int f = 0; f < nf; ++f) {
    cand[f].Reset();
    if (!(*mask)[f]) continue;
    const BinMapper* m = train_data_->FeatureBinMapper(f);
    const hist_t* fh = hist + 2 * train_data_->hist_offset(f);
    if (m->bin_type() == BinType::kCategorical) {
      FindBestThresholdCategorical(fh, m->num_bin(), ctx, *config_, &cand[f]);
    } else {
      int rand_t = -1;
      if (config_->extra_trees) {
        // extra-trees: a single random threshold per feature per node
        rand_t = extra_rng_.NextInt(0, std::max(1, m->num_numeric_bin() - 1));
      }
      int8_t mono = 0;
      if (!config_->monotone_constraints.empty()) {
        int orig = train_data_->RealFeatureIndex(f);
        if (orig < static_cast<int>(config_->monotone_constraints.size()))
          mono = static_cast<int8_t>(config_->monotone_constraints[orig]);
      }
      MonoAdvBounds adv;
      if (mono_advanced_ && leaf < static_cast<int>(mono_leaf_in_subtree_.size()) &&
          mono_leaf_in_subtree_[leaf])
        MonoAdvBoundsForFeature(leaf, f, m->num_numeric_bin(), &adv);
      FindBestThresholdNumerical(fh, m->num_bin(), m->num_numeric_bin(), m->nan_bin(), ctx,
                                 *config_, mono, rand_t, &cand[f],
                                 adv.empty() ? nullptr : &adv);
    }
    cand[f].feature = f;
  }
  for (int f = 0; f < nf; ++f) {
    if (!cand[f].IsValid()) continue;
    const int orig = train_data_->RealFeatureIndex(f);
    // monotone_penalty: depth-decaying multiplicative penalty on monotone splits
    // (parity: reference ComputeMonotoneSplitGainPenalty)
    if (cand[f].monotone_type != 0 && config_->monotone_penalty > 0.0) {
      const double pen = config_->monotone_penalty;
      const int depth = ctx.depth;
      double factor;
      if (pen >= depth + 1.0) factor = kEpsilon;
      else if (pen <= 1.0) factor = 1.0 - pen / std::pow(2.0, depth) + kEpsilon;
      else factor = 1.0 - std::pow(2.0, pen - 1.0 - depth) + kEpsilon;
      cand[f].gain *= factor;
    }
    // feature_contri: multiplicative per-feature gain reweighting
    if (!config_->feature_contri.empty() &&
        orig < static_cast<int>(config_->feature_contri.size())) {
      cand[f].gain *= config_->feature_contri[orig];
    }
    // CEGB: cost-efficient gradient boosting gain penalty
    if (config_->cegb_tradeoff > 0.0 &&
        (config_->cegb_penalty_split > 0.0 ||
         !config_->cegb_penalty_feature_coupled.empty() ||
         !config_->cegb_penalty_feature_lazy.empty())) {
      double penalty = config_->cegb_penalty_split;
      if (orig < static_cast<int>(config_->cegb_penalty_feature_coupled.size()) &&
          !cegb_feature_used_[orig])
        penalty += config_->cegb_penalty_feature_coupled[orig];
      if (orig < static_cast<int>(config_->cegb_penalty_feature_lazy.size()))
        penalty += config_->cegb_penalty_feature_lazy[orig] * ctx.num_data;
      cand[f].gain -= config_->cegb_tradeoff * penalty;
      if (cand[f].gain <= 0) continue;
    }
    if (cand[f] > best) best = cand[f];
  }
  if (MonoDebug() && best.IsValid())
    fprintf(stderr, "[mono]   fbs leaf=%d -> f=%d thr=%u mono=%d gain=%.4g b=[%.4g,%.4g]\n",
            leaf, best.feature, best.threshold, (int)best.monotone_type, best.gain,
            ctx.out_lo, ctx.out_hi);
}

bool SerialTreeLearner::MakeForcedSplit(int leaf, const LeafContext& ctx,
                                        const ForcedNode* node, SplitInfo* out) {
  const int inner = train_data_->InnerFeatureIndex(node->feature);
  if (inner < 0) return false;
  const BinMapper* m = train_data_->FeatureBinMapper(inner);
  if (m->bin_type() != BinType::kNumerical) return false;
  int bin = static_cast<int>(m->ValueToBin(node->threshold));
  bin = std::min(bin, m->num_numeric_bin() - 2);
  if (bin < 0) return false;
  if (leaf_to_slot_[leaf] < 0) {
    // pool-evicted histogram: rebuild it for this leaf before reading
    data_size_t cnt2;
    const data_size_t* idx2 = partition_.GetIndexOnLeaf(leaf, &cnt2);
    ComputeHistogram(leaf, cnt2, idx2);
    OnHistogramReady(leaf);
  }
  const hist_t* fh = HistSlot(leaf_to_slot_[leaf]) + 2 * train_data_->hist_offset(inner);
  double gl = 0, hl = 0;
  for (int b = 0; b <= bin; ++b) {
    gl += fh[2 * b];
    hl += fh[2 * b + 1];
  }
  out->Reset();
  out->feature = inner;
  out->threshold = static_cast<uint32_t>(bin);
  out->default_left = false;
  out->left_sum_gradient = gl;
  out->left_sum_hessian = hl;
  out->right_sum_gradient = ctx.sum_gradient - gl;
  out->right_sum_hessian = ctx.sum_hessian - hl;
  const double cf = ctx.num_data > 0 && ctx.sum_hessian > 0
                        ? ctx.num_data / ctx.sum_hessian : 1.0;
  out->left_count = static_cast<data_size_t>(Common::RoundInt(hl * cf));
  out->right_count = ctx.num_data - out->left_count;
  out->left_output = GainMath::CalculateSplittedLeafOutput(
      gl, hl, config_->lambda_l1, config_->lambda_l2, config_->max_delta_step);
  out->right_output = GainMath::CalculateSplittedLeafOutput(
      out->right_sum_gradient, out->right_sum_hessian, config_->lambda_l1,
      config_->lambda_l2, config_->max_delta_step);
  // monotone constraints apply to forced splits too: clamp into the leaf's
  // inherited bounds and tag the direction so the Train loop propagates bounds
  out->left_output = std::min(std::max(out->left_output, ctx.out_lo), ctx.out_hi);
  out->right_output = std::min(std::max(out->right_output, ctx.out_lo), ctx.out_hi);
  if (!config_->monotone_constraints.empty()) {
    const int orig = train_data_->RealFeatureIndex(inner);
    if (orig < static_cast<int>(config_->monotone_constraints.size()))
      out->monotone_type = static_cast<int8_t>(config_->monotone_constraints[orig]);
    if (out->monotone_type > 0 && out->left_output > out->right_output) {
      out->left_output = out->right_output =
          (out->left_output + out->right_output) / 2.0;
    } else if (out->monotone_type < 0 && out->left_output < out->right_output) {
      out->left_output = out->right_output =
          (out->left_output + out->right_output) / 2.0;
    }
  }
  out->gain = 1e30;  // forced splits take precedence over gain selection
  return true;
}

std::function<bool(data_size_t)> SerialTreeLearner::MakeGoLeft(const SplitInfo& s) const {
  const int f = s.feature;
  const Dataset* data = train_data_;
  const BinMapper* m = train_data_->FeatureBinMapper(f);
  if (!s.cat_bitset_inner.empty()) {
    std::vector<uint32_t> bits = s.cat_bitset_inner;
    const int nwords = static_cast<int>(bits.size());
    return [data, f, bits, nwords](data_size_t row) {
      uint32_t b = data->GetBin(row, f);
      return (b >> 5) < static_cast<uint32_t>(nwords) && ((bits[b >> 5] >> (b & 31)) & 1);
    };
  }
  const uint32_t thr = s.threshold;
  const int nanb = m->nan_bin();
  const bool default_left = s.default_left;
  if (nanb < 0) {
    return [data, f, thr](data_size_t row) { return data->GetBin(row, f) <= thr; };
  }
  const uint32_t nb = static_cast<uint32_t>(nanb);
  return [data, f, thr, nb, default_left](data_size_t row) {
    uint32_t b = data->GetBin(row, f);
    if (b == nb) return default_left;
    return b <= thr;
  };
}

Tree* SerialTreeLearner::Train(const score_t* gradients, const score_t* hessians,
                               bool /*is_first_tree*/) {
  tree_const_hess_ = -1;  // re-classify hessians for this tree's gradients
  mono_intermediate_ = !config_->monotone_constraints.empty() &&
                       config_->monotone_constraints_method != "basic";
  mono_advanced_ = mono_intermediate_ &&
                   config_->monotone_constraints_method == "advanced";
  if (mono_intermediate_) {
    mono_node_parent_.assign(config_->num_leaves, -1);
    mono_leaf_in_subtree_.assign(config_->num_leaves, 0);
  }
  gradients_ = gradients;
  hessians_ = hessians;
  if (config_->use_quantized_grad) {
    // CPU gradient discretization (reference GradientDiscretizer semantics):
    // stochastic rounding onto a num_grad_quant_bins grid scaled by absmax.
    // Values stay score_t so the whole histogram pipeline is unchanged; sums of
    // grid multiples are exact in fp64, so subtraction keeps its guarantee.
    const data_size_t n = train_data_->num_data();
    quant_grad_.resize(n);
    quant_hess_.resize(n);
    double gmax = 0.0, hmax = 0.0;
#pragma omp parallel for schedule(static) reduction(max : gmax, hmax)
    for (data_size_t i = 0; i < n; ++i) {
      gmax = std::max(gmax, std::fabs(static_cast<double>(gradients[i])));
      hmax = std::max(hmax, static_cast<double>(hessians[i]));
    }
    const int levels = std::max(1, config_->num_grad_quant_bins / 2);
    const double gs = gmax > 0 ? gmax / levels : 1.0;
    const double hs = hmax > 0 ? hmax / (2.0 * levels) : 1.0;
    quant_seed_ = quant_seed_ * 1664525u + 1013904223u;
    const uint32_t seed = quant_seed_;
#pragma omp parallel for schedule(static)
    for (data_size_t i = 0; i < n; ++i) {
      uint32_t x = (static_cast<uint32_t>(i) * 2654435761u) ^ seed;
      x ^= x >> 16; x *= 2246822519u; x ^= x >> 13;
      const double rg = (x & 0xFFFF) * (1.0 / 65536.0);
      const double rh = ((x >> 16) & 0xFFFF) * (1.0 / 65536.0);
      int gq = static_cast<int>(std::floor(gradients[i] / gs + rg));
      int hq = static_cast<int>(std::floor(hessians[i] / hs + rh));
      gq = std::max(-levels, std::min(levels, gq));
      hq = std::max(0, std::min(2 * levels, hq));
      quant_grad_[i] = static_cast<score_t>(gq * gs);
      quant_hess_[i] = static_cast<score_t>(hq * hs);
    }
    gradients_ = quant_grad_.data();
    hessians_ = quant_hess_.data();
  }
  ++iter_counter_;
  is_feature_used_ = SampleFeatures(false);

  auto tree = std::make_unique<Tree>(config_->num_leaves);
  mono_tree_ = tree.get();  // advanced monotone bounds walk the growing tree
  partition_.ResetToRoot(bag_indices_, bag_cnt_);
  std::fill(leaf_to_slot_.begin(), leaf_to_slot_.end(), -1);
  std::fill(slot_owner_.begin(), slot_owner_.end(), -1);
  std::fill(slot_used_.begin(), slot_used_.end(), 0);
  slot_clock_ = 0;

  // root stats
  data_size_t root_cnt;
  const data_size_t* root_idx = partition_.GetIndexOnLeaf(0, &root_cnt);
  double sum_g = 0.0, sum_h = 0.0;
  if (config_->deterministic) {
    // fixed 1024-row blocks summed in order: invariant to the thread count
    const int nblk = static_cast<int>((root_cnt + 1023) / 1024);
    std::vector<double> bg(nblk, 0.0), bh(nblk, 0.0);
#pragma omp parallel for schedule(static)
    for (int b = 0; b < nblk; ++b) {
      const data_size_t s0 = static_cast<data_size_t>(b) * 1024;
      const data_size_t e0 = std::min<data_size_t>(root_cnt, s0 + 1024);
      double g = 0.0, h = 0.0;
      for (data_size_t i = s0; i < e0; ++i) {
        g += gradients_[root_idx[i]];
        h += hessians_[root_idx[i]];
      }
      bg[b] = g;
      bh[b] = h;
    }
    for (int b = 0; b < nblk; ++b) {
      sum_g += bg[b];
      sum_h += bh[b];
    }
  } else {
#pragma omp parallel for schedule(static) reduction(+ : sum_g, sum_h)
    for (data_size_t i = 0; i < root_cnt; ++i) {
      sum_g += gradients_[root_idx[i]];
      sum_h += hessians_[root_idx[i]];
    }
  }
  data_size_t global_root_cnt = root_cnt;
  ReduceRootStats(&sum_g, &sum_h, &global_root_cnt);
  leaf_ctx_[0] = {sum_g, sum_h, global_root_cnt,
                  GainMath::CalculateSplittedLeafOutput(sum_g, sum_h, config_->lambda_l1,
                                                        config_->lambda_l2,
                                                        config_->max_delta_step), 0};
  ComputeHistogram(0, root_cnt, root_idx);
  OnHistogramReady(0);
  FindBestSplitForLeaf(0, leaf_ctx_[0]);

  int num_leaves = 1;
  std::fill(forced_of_leaf_.begin(), forced_of_leaf_.end(), nullptr);
  forced_of_leaf_[0] = forced_root_.get();
  for (int split_i = 0; split_i < config_->num_leaves - 1; ++split_i) {
    // forced splits first (in discovery order), then best-gain leaf
    int best_leaf = -1;
    SplitInfo forced_split;
    for (int l = 0; l < num_leaves; ++l) {
      if (forced_of_leaf_[l] != nullptr &&
          MakeForcedSplit(l, leaf_ctx_[l], forced_of_leaf_[l], &forced_split)) {
        best_leaf = l;
        best_split_per_leaf_[l] = forced_split;
        break;
      }
      if (forced_of_leaf_[l] != nullptr) forced_of_leaf_[l] = nullptr;  // unusable
    }
    if (best_leaf < 0) {
      double best_gain = 0.0;
      for (int l = 0; l < num_leaves; ++l) {
        if (best_split_per_leaf_[l].IsValid() && best_split_per_leaf_[l].gain > best_gain) {
          best_gain = best_split_per_leaf_[l].gain;
          best_leaf = l;
        }
      }
    }
    if (best_leaf < 0) break;
    SplitInfo& s = best_split_per_leaf_[best_leaf];
    const int f = s.feature;
    const int orig_f = train_data_->RealFeatureIndex(f);
    const BinMapper* m = train_data_->FeatureBinMapper(f);
    const int right_leaf = num_leaves;

    // partition first so the tree records exact child counts
    {
      // fast path: plain dense-uint8 numeric split (the dominant case)
      const BinMapper* sm = train_data_->FeatureBinMapper(s.feature);
      const auto& scol = train_data_->column(train_data_->feature_column(s.feature));
      if (s.cat_bitset_inner.empty() && !train_data_->feature_bundled(s.feature) &&
          !scol.is_sparse() && !scol.is4() && !scol.is16()) {
        partition_.SplitDenseU8(best_leaf, right_leaf, scol.data8(), s.threshold,
                                sm->nan_bin(), s.default_left);
      } else {
        partition_.Split(best_leaf, right_leaf, MakeGoLeft(s));
      }
    }
    data_size_t left_cnt_actual = partition_.leaf_count(best_leaf);
    data_size_t right_cnt_actual = partition_.leaf_count(right_leaf);
    GlobalChildCounts(&left_cnt_actual, &right_cnt_actual);
    s.left_count = left_cnt_actual;
    s.right_count = right_cnt_actual;

    // tree structure update
    int split_parent_node = -1;
    if (!s.cat_bitset_inner.empty()) {
      // map bin-level bitset to category-value bitset for prediction on raw values
      std::vector<uint32_t> cat_bits;
      for (int w = 0; w < static_cast<int>(s.cat_bitset_inner.size()); ++w) {
        uint32_t word = s.cat_bitset_inner[w];
        while (word) {
          int bit = __builtin_ctz(word);
          word &= word - 1;
          int bin = w * 32 + bit;
          int cat = static_cast<int>(m->BinToValue(bin));
          if (cat >= 0) {
            if ((cat >> 5) >= static_cast<int>(cat_bits.size())) cat_bits.resize((cat >> 5) + 1, 0);
            cat_bits[cat >> 5] |= 1u << (cat & 31);
          }
        }
      }
      split_parent_node = tree->leaf_parent(best_leaf);
      tree->SplitCategorical(best_leaf, f, orig_f, cat_bits.data(),
                             static_cast<int>(cat_bits.size()), s.left_output, s.right_output,
                             s.left_count, s.right_count, s.left_sum_hessian,
                             s.right_sum_hessian, static_cast<float>(s.gain),
                             m->missing_type());
      // store bin-level bitset on the tree node for training-time partition? partition uses s directly
    } else {
      split_parent_node = tree->leaf_parent(best_leaf);
      tree->Split(best_leaf, f, orig_f, s.threshold,
                  train_data_->RealThreshold(f, s.threshold), s.left_output, s.right_output,
                  s.left_count, s.right_count, s.left_sum_hessian, s.right_sum_hessian,
                  static_cast<float>(s.gain), m->missing_type(), s.default_left);
    }

    // propagate forced-split children
    {
      const ForcedNode* fn = forced_of_leaf_[best_leaf];
      forced_of_leaf_[best_leaf] = fn != nullptr ? fn->left.get() : nullptr;
      forced_of_leaf_[right_leaf] = fn != nullptr ? fn->right.get() : nullptr;
    }

    // bookkeeping for CEGB / interaction constraints
    cegb_feature_used_[orig_f] = 1;
    if (!interaction_groups_.empty()) {
      leaf_branch_features_[right_leaf] = leaf_branch_features_[best_leaf];
      leaf_branch_features_[best_leaf].insert(orig_f);
      leaf_branch_features_[right_leaf].insert(orig_f);
    }

    // child contexts
    const double parent_out = (s.left_output * s.left_sum_hessian +
                               s.right_output * s.right_sum_hessian) /
                              std::max(s.left_sum_hessian + s.right_sum_hessian, kEpsilon);
    int parent_depth = leaf_ctx_[best_leaf].depth;
    const double b_lo = leaf_ctx_[best_leaf].out_lo;
    const double b_hi = leaf_ctx_[best_leaf].out_hi;
    leaf_ctx_[best_leaf] = {s.left_sum_gradient, s.left_sum_hessian, left_cnt_actual,
                            parent_out, parent_depth + 1, b_lo, b_hi};
    leaf_ctx_[right_leaf] = {s.right_sum_gradient, s.right_sum_hessian, right_cnt_actual,
                             parent_out, parent_depth + 1, b_lo, b_hi};
    if (!mono_intermediate_ && s.monotone_type != 0) {
      // BasicLeafConstraints: descendants of the low side may not exceed the split
      // midpoint, and vice versa — monotonicity holds for the whole subtree
      const double mid = (s.left_output + s.right_output) / 2.0;
      if (s.monotone_type > 0) {
        leaf_ctx_[best_leaf].out_hi = std::min(b_hi, mid);
        leaf_ctx_[right_leaf].out_lo = std::max(b_lo, mid);
      } else {
        leaf_ctx_[best_leaf].out_lo = std::max(b_lo, mid);
        leaf_ctx_[right_leaf].out_hi = std::min(b_hi, mid);
      }
    }
    if (mono_intermediate_) {
      const int new_node = tree->leaf_parent(best_leaf);
      mono_node_parent_[new_node] = split_parent_node;
    }
    ++num_leaves;

    // histograms: build smaller child, subtract for larger.
    // parent hist currently lives in slot leaf_to_slot_[best_leaf] (may have
    // been evicted by the pool, in which case the larger child is recomputed).
    const int parent_slot = leaf_to_slot_[best_leaf];
    bool left_smaller = left_cnt_actual <= right_cnt_actual;
    int small_leaf = left_smaller ? best_leaf : right_leaf;
    int large_leaf = left_smaller ? right_leaf : best_leaf;
    // detach the parent mapping, then give the smaller child a fresh slot and
    // (when valid) hand the parent's slot to the larger child for subtraction
    leaf_to_slot_[best_leaf] = -1;
    if (parent_slot >= 0) {
      slot_owner_[parent_slot] = large_leaf;
      leaf_to_slot_[large_leaf] = parent_slot;
      slot_used_[parent_slot] = ++slot_clock_;
    }
    const int spare_slot = AcquireSlot(small_leaf, large_leaf);
    data_size_t small_cnt;
    const data_size_t* small_idx = partition_.GetIndexOnLeaf(small_leaf, &small_cnt);
    ComputeHistogram(small_leaf, small_cnt, small_idx);
    OnHistogramReady(small_leaf);
    if (build_both_children_ || leaf_to_slot_[large_leaf] < 0) {
      // voting-parallel builds both children explicitly (subtraction is invalid
      // when only voted ranges are global); a pool-evicted parent forces the
      // same recompute fallback (reference HistogramPool semantics)
      AcquireSlot(large_leaf, small_leaf);
      data_size_t large_cnt;
      const data_size_t* large_idx = partition_.GetIndexOnLeaf(large_leaf, &large_cnt);
      ComputeHistogram(large_leaf, large_cnt, large_idx);
      OnHistogramReady(large_leaf);
    } else {
      // in-place: parent_slot -= small_slot -> becomes large hist
      SubtractHistogram(large_leaf, leaf_to_slot_[large_leaf], spare_slot);
    }

    if (mono_intermediate_) {
      // NOTE: s references best_split_per_leaf_[best_leaf], which the child
      // FindBestSplitForLeaf calls below overwrite — snapshot the APPLIED split
      const SplitInfo applied = s;
      FindBestSplitForLeaf(small_leaf, leaf_ctx_[small_leaf]);
      FindBestSplitForLeaf(large_leaf, leaf_ctx_[large_leaf]);
      if (MonoDebug())
        fprintf(stderr,
                "[mono] split leaf=%d->(%d,%d) inner=%d thr=%u mono=%d "
                "out=(%.6g,%.6g) bounds=[%.4g,%.4g]\n",
                best_leaf, best_leaf, right_leaf, applied.feature, applied.threshold,
                (int)applied.monotone_type, applied.left_output, applied.right_output,
                b_lo, b_hi);
      // tighten contiguous leaves' bounds against the new outputs and recompute
      // their best splits (reference IntermediateLeafConstraints::Update)
      MonotoneIntermediateUpdate(tree.get(), best_leaf, right_leaf, applied,
                                 tree->leaf_parent(best_leaf),
                                 applied.cat_bitset_inner.empty());
    } else {
      FindBestSplitForLeaf(small_leaf, leaf_ctx_[small_leaf]);
      FindBestSplitForLeaf(large_leaf, leaf_ctx_[large_leaf]);
    }
  }
  if (config_->use_quantized_grad && config_->quant_train_renew_leaf) {
    // renew leaf outputs from the UNquantized gradients (reference
    // quant_train_renew_leaf): removes discretization bias from the values
    // while splits stay those chosen on the quantized histograms.
    for (int l = 0; l < tree->num_leaves(); ++l) {
      data_size_t cnt;
      const data_size_t* idx = partition_.GetIndexOnLeaf(l, &cnt);
      double sg = 0.0, sh = 0.0;
#pragma omp parallel for schedule(static) reduction(+ : sg, sh)
      for (data_size_t i = 0; i < cnt; ++i) {
        sg += gradients[idx[i]];
        sh += hessians[idx[i]];
      }
      tree->SetLeafOutput(l, GainMath::CalculateSplittedLeafOutput(
          sg, sh, config_->lambda_l1, config_->lambda_l2, config_->max_delta_step));
    }
  }
  if (config_->linear_tree) CalculateLinear(tree.get());
  mono_tree_ = nullptr;
  return tree.release();
}

/*! Fit a ridge-regularized weighted linear model in every leaf over the numerical
 *  features on the leaf's branch path (parity: reference LinearTreeLearner, Eigen-free:
 *  small Cholesky solve; targets -g/h weighted by h). */
void SerialTreeLearner::CalculateLinear(Tree* tree) {
  if (!train_data_->has_raw()) {
    Log::Warning("linear_tree requires raw values; dataset was built without them");
    return;
  }
  tree->SetLinear(true);
  const int nl = tree->num_leaves();
  if (nl <= 1) return;
  // collect per-leaf branch features (numerical, unique, inner indices)
  std::vector<std::vector<int>> leaf_feats(nl);
  std::function<void(int, std::vector<int>&)> walk = [&](int node, std::vector<int>& path) {
    if (node < 0) {
      leaf_feats[~node] = path;
      return;
    }
    const int fi = tree->split_feature_inner(node);
    const bool is_num = !tree->IsCategoricalSplit(node);
    bool added = false;
    if (is_num && std::find(path.begin(), path.end(), fi) == path.end()) {
      path.push_back(fi);
      added = true;
    }
    walk(tree->left_child(node), path);
    walk(tree->right_child(node), path);
    if (added) path.pop_back();
  };
  std::vector<int> path;
  walk(0, path);

#pragma omp parallel for schedule(dynamic)
  for (int l = 0; l < nl; ++l) {
    const auto& feats = leaf_feats[l];
    const int k = static_cast<int>(feats.size());
    data_size_t cnt;
    const data_size_t* idx = partition_.GetIndexOnLeaf(l, &cnt);
    if (k == 0 || cnt < static_cast<data_size_t>(k + 2)) continue;
    const int dim = k + 1;  // coefficients + intercept
    std::vector<double> A(dim * dim, 0.0), b(dim, 0.0), z(dim);
    bool has_nan = false;
    for (data_size_t i = 0; i < cnt; ++i) {
      const data_size_t r = idx[i];
      for (int j = 0; j < k; ++j) {
        z[j] = train_data_->raw_value(feats[j], r);
        if (std::isnan(z[j])) { has_nan = true; break; }
      }
      if (has_nan) break;
      z[k] = 1.0;
      const double h = hessians_[r];
      const double g = gradients_[r];
      for (int a = 0; a < dim; ++a) {
        for (int c2 = a; c2 < dim; ++c2) A[a * dim + c2] += h * z[a] * z[c2];
        b[a] += -g * z[a];
      }
    }
    if (has_nan) continue;  // leaves with missing values stay piecewise-constant
    for (int a = 0; a < k; ++a) A[a * dim + a] += config_->linear_lambda + config_->lambda_l2;
    A[k * dim + k] += config_->lambda_l2;
    // Cholesky solve (upper triangle filled)
    std::vector<double> Lm(dim * dim, 0.0);
    bool ok = true;
    for (int a = 0; a < dim && ok; ++a) {
      for (int c2 = 0; c2 <= a; ++c2) {
        double sum = A[std::min(a, c2) * dim + std::max(a, c2)];
        for (int t = 0; t < c2; ++t) sum -= Lm[a * dim + t] * Lm[c2 * dim + t];
        if (a == c2) {
          if (sum <= 1e-12) { ok = false; break; }
          Lm[a * dim + a] = std::sqrt(sum);
        } else {
          Lm[a * dim + c2] = sum / Lm[c2 * dim + c2];
        }
      }
    }
    if (!ok) continue;
    std::vector<double> y(dim), beta(dim);
    for (int a = 0; a < dim; ++a) {
      double sum = b[a];
      for (int t = 0; t < a; ++t) sum -= Lm[a * dim + t] * y[t];
      y[a] = sum / Lm[a * dim + a];
    }
    for (int a = dim - 1; a >= 0; --a) {
      double sum = y[a];
      for (int t = a + 1; t < dim; ++t) sum -= Lm[t * dim + a] * beta[t];
      beta[a] = sum / Lm[a * dim + a];
    }
    bool finite = true;
    for (double v : beta) finite &= std::isfinite(v);
    if (!finite) continue;
    std::vector<int> feats_real(k);
    for (int j = 0; j < k; ++j) feats_real[j] = train_data_->RealFeatureIndex(feats[j]);
    std::vector<double> coeffs(beta.begin(), beta.begin() + k);
    tree->SetLeafLinear(l, beta[k], feats_real, feats, coeffs);
  }
}


void SerialTreeLearner::EnsureLeafHistogram(int leaf) {
  if (leaf_to_slot_[leaf] >= 0) return;
  data_size_t cnt;
  const data_size_t* idx = partition_.GetIndexOnLeaf(leaf, &cnt);
  ComputeHistogram(leaf, cnt, idx);
  OnHistogramReady(leaf);
}

/*! contiguity-aware descent into the subtree OPPOSITE a monotone ancestor
 *  split: finds the leaves whose regions touch the freshly split leaves and
 *  tightens their output bounds against the new outputs. */
void SerialTreeLearner::MonoGoDown(const Tree* tree, int node,
                                   const std::vector<int>& up_feats,
                                   const std::vector<uint32_t>& up_thresholds,
                                   const std::vector<uint8_t>& up_was_right,
                                   bool update_max, int split_feature,
                                   const SplitInfo& s, bool use_left, bool use_right,
                                   uint32_t split_threshold,
                                   std::vector<int>* leaves_to_update) {
  if (node < 0) {
    const int leaf = ~node;
    // leaves that cannot split anymore are still bound-tracked (their OUTPUT is
    // already monotone-consistent; only future-split bounds matter)
    double lo_c, hi_c;
    if (use_left && use_right) {
      lo_c = std::min(s.left_output, s.right_output);
      hi_c = std::max(s.left_output, s.right_output);
    } else if (use_right) {
      lo_c = hi_c = s.right_output;
    } else {
      lo_c = hi_c = s.left_output;
    }
    bool changed = false;
    if (MonoDebug()) {
      const double out = tree->LeafOutput(leaf);
      if (update_max ? (lo_c < out - 1e-12) : (hi_c > out + 1e-12))
        fprintf(stderr,
                "[mono] INVARIANT BROKEN leaf=%d out=%.6g update_max=%d c=[%.6g,%.6g]\n",
                leaf, out, update_max ? 1 : 0, lo_c, hi_c);
    }
    if (update_max) {
      if (lo_c < leaf_ctx_[leaf].out_hi) {
        leaf_ctx_[leaf].out_hi = lo_c;
        changed = true;
      }
    } else {
      if (hi_c > leaf_ctx_[leaf].out_lo) {
        leaf_ctx_[leaf].out_lo = hi_c;
        changed = true;
      }
    }
    // advanced mode recomputes bounds fresh at scan time, so a visited leaf's
    // constraints may have changed (even RELAXED) regardless of the scalar
    // tightening above (reference AdvancedConstraintEntry always reports change)
    if (changed || mono_advanced_) {
      leaves_to_update->push_back(leaf);
      if (MonoDebug())
        fprintf(stderr, "[mono]   update leaf=%d out=%.6g -> bounds=[%.4g,%.4g]\n", leaf,
                tree->LeafOutput(leaf), leaf_ctx_[leaf].out_lo, leaf_ctx_[leaf].out_hi);
    }
    return;
  }
  const int feat = tree->split_feature_inner(node);
  const uint32_t thr = tree->threshold_in_bin(node);
  const bool numerical = !tree->IsCategoricalSplit(node);
  // prune subtrees that cannot touch the original leaves' region: a split on a
  // feature already crossed on the way up bounds the region on one side
  bool go_left = true, go_right = true;
  if (numerical) {
    for (size_t i = 0; i < up_feats.size(); ++i) {
      if (up_feats[i] != feat) continue;
      if (thr >= up_thresholds[i] && !up_was_right[i]) go_right = false;
      if (thr <= up_thresholds[i] && up_was_right[i]) go_left = false;
      if (!go_left && !go_right) break;
    }
  }
  // a same-feature split separates one child's region from one of the two new
  // leaves (reference use_left_leaf_for_update_right / use_right_..._left)
  bool right_child_sees_left = true;  // right descent stays contiguous w/ LEFT leaf
  bool left_child_sees_right = true;  // left descent stays contiguous w/ RIGHT leaf
  if (numerical && feat == split_feature) {
    if (thr >= split_threshold) right_child_sees_left = false;
    if (thr <= split_threshold) left_child_sees_right = false;
  }
  if (go_left) {
    MonoGoDown(tree, tree->left_child(node), up_feats, up_thresholds, up_was_right,
               update_max, split_feature, s, use_left,
               use_right && left_child_sees_right, split_threshold, leaves_to_update);
  }
  if (go_right) {
    MonoGoDown(tree, tree->right_child(node), up_feats, up_thresholds, up_was_right,
               update_max, split_feature, s, use_left && right_child_sees_left,
               use_right, split_threshold, leaves_to_update);
  }
}

/*! advanced ("monotone precise") mode — descend the subtree OPPOSITE a monotone
 *  ancestor and fold every contiguous leaf's output into the per-bin bound
 *  arrays over the bin interval [s, e) of the scanned feature that the leaf can
 *  actually constrain. Splits on the scanned feature narrow the interval (an
 *  empty interval prunes the branch — this is what keeps only boundary-adjacent
 *  regions alive when the monotone feature IS the scanned feature); splits on
 *  other features use the same contiguity fences as MonoGoDown.
 *  (reference GoDownToFindConstrainingLeaves, monotone_constraints.hpp:1002) */
void SerialTreeLearner::MonoAdvGoDown(const Tree* tree, int node, int f, bool want_min,
                                      int s, int e, const std::vector<int>& up_feats,
                                      const std::vector<uint32_t>& up_thresholds,
                                      const std::vector<uint8_t>& up_was_right,
                                      MonoAdvBounds* out) const {
  if (s >= e) return;
  if (node < 0) {
    const double v = tree->LeafOutput(~node);
    if (want_min) {
      for (int b = s; b < e; ++b) out->lo[b] = std::max(out->lo[b], v);
    } else {
      for (int b = s; b < e; ++b) out->hi[b] = std::min(out->hi[b], v);
    }
    return;
  }
  const int feat = tree->split_feature_inner(node);
  const uint32_t thr = tree->threshold_in_bin(node);
  const bool numerical = !tree->IsCategoricalSplit(node);
  if (numerical && feat == f) {
    MonoAdvGoDown(tree, tree->left_child(node), f, want_min, s,
                  std::min<int>(static_cast<int>(thr) + 1, e), up_feats, up_thresholds,
                  up_was_right, out);
    MonoAdvGoDown(tree, tree->right_child(node), f, want_min,
                  std::max<int>(static_cast<int>(thr) + 1, s), e, up_feats, up_thresholds,
                  up_was_right, out);
    return;
  }
  bool go_left = true, go_right = true;
  if (numerical) {
    for (size_t i = 0; i < up_feats.size(); ++i) {
      if (up_feats[i] != feat) continue;
      if (thr >= up_thresholds[i] && !up_was_right[i]) go_right = false;
      if (thr <= up_thresholds[i] && up_was_right[i]) go_left = false;
      if (!go_left && !go_right) break;
    }
  }
  if (go_left)
    MonoAdvGoDown(tree, tree->left_child(node), f, want_min, s, e, up_feats,
                  up_thresholds, up_was_right, out);
  if (go_right)
    MonoAdvGoDown(tree, tree->right_child(node), f, want_min, s, e, up_feats,
                  up_thresholds, up_was_right, out);
}

/*! per-(leaf, feature) bound computation for the advanced monotone mode: walk UP
 *  from the leaf; at every monotone ancestor whose opposite subtree lies on the
 *  constraining side, walk DOWN it collecting contiguous leaves' outputs into
 *  dense per-bin min/max arrays. Recomputed fresh from the current tree at every
 *  scan, so bounds RELAX when a constraining leaf is split into finer regions —
 *  the point of the precise mode. (reference AdvancedConstraintEntry::
 *  RecomputeConstraintsIfNeeded + GoUpToFindConstrainingLeaves) */
void SerialTreeLearner::MonoAdvBoundsForFeature(int leaf, int f, int num_numeric_bin,
                                                MonoAdvBounds* out) const {
  out->lo.clear();
  out->hi.clear();
  const Tree* tree = mono_tree_;
  if (tree == nullptr || tree->leaf_parent(leaf) < 0) return;
  out->lo.assign(num_numeric_bin, -std::numeric_limits<double>::infinity());
  out->hi.assign(num_numeric_bin, std::numeric_limits<double>::infinity());
  for (int pass = 0; pass < 2; ++pass) {
    const bool want_min = (pass == 0);
    int it_start = 0, it_end = num_numeric_bin;
    std::vector<int> up_feats;
    std::vector<uint32_t> up_thr;
    std::vector<uint8_t> up_right;
    int node = ~leaf;  // child handle as stored in the parent's child slots
    int parent = tree->leaf_parent(leaf);
    while (parent >= 0) {
      const int feat = tree->split_feature_inner(parent);
      const bool p_num = !tree->IsCategoricalSplit(parent);
      const bool is_right = tree->right_child(parent) == node;
      const uint32_t thr = tree->threshold_in_bin(parent);
      if (p_num && feat == f) {
        // narrow to the leaf's own bin range; the right-child case keeps bin
        // `thr` itself so boundary-adjacent opposite regions stay non-empty
        if (is_right) it_start = std::max<int>(it_start, static_cast<int>(thr));
        else it_end = std::min<int>(it_end, static_cast<int>(thr) + 1);
      }
      bool should_descend = p_num;
      if (p_num) {
        for (size_t i = 0; i < up_feats.size(); ++i) {
          if (up_feats[i] == feat && (up_right[i] != 0) == is_right) {
            should_descend = false;
            break;
          }
        }
      }
      if (should_descend) {
        const int orig_f = tree->split_feature(parent);
        int8_t mono = 0;
        if (orig_f >= 0 && orig_f < static_cast<int>(config_->monotone_constraints.size()))
          mono = static_cast<int8_t>(config_->monotone_constraints[orig_f]);
        if (mono != 0) {
          // increasing (mono>0) with the leaf on the right: the opposite (left)
          // subtree lies BELOW -> it contributes lower bounds (want_min pass)
          const bool opposite_is_lower = (mono > 0) == is_right;
          if (opposite_is_lower == want_min) {
            const int opposite =
                is_right ? tree->left_child(parent) : tree->right_child(parent);
            MonoAdvGoDown(tree, opposite, f, want_min, it_start, it_end, up_feats,
                          up_thr, up_right, out);
          }
        }
        up_right.push_back(is_right ? 1 : 0);
        up_thr.push_back(thr);
        up_feats.push_back(feat);
      }
      node = parent;
      parent = mono_node_parent_[parent];
    }
  }
}

void SerialTreeLearner::MonotoneIntermediateUpdate(const Tree* tree, int left_leaf,
                                                   int right_leaf, const SplitInfo& s,
                                                   int split_node, bool is_numerical) {
  const bool was_mono_subtree = mono_leaf_in_subtree_[left_leaf] != 0;
  if (s.monotone_type != 0 || was_mono_subtree) {
    mono_leaf_in_subtree_[left_leaf] = 1;
    mono_leaf_in_subtree_[right_leaf] = 1;
  }
  if (!mono_leaf_in_subtree_[left_leaf]) return;
  // children bound the SIBLING's actual output (tighter than Basic's midpoint
  // clamp on one side, looser on the other — reference UpdateConstraintsWithOutputs)
  if (is_numerical && s.monotone_type != 0) {
    if (s.monotone_type < 0) {
      leaf_ctx_[left_leaf].out_lo = std::max(leaf_ctx_[left_leaf].out_lo, s.right_output);
      leaf_ctx_[right_leaf].out_hi =
          std::min(leaf_ctx_[right_leaf].out_hi, s.left_output);
    } else {
      leaf_ctx_[left_leaf].out_hi = std::min(leaf_ctx_[left_leaf].out_hi, s.right_output);
      leaf_ctx_[right_leaf].out_lo =
          std::max(leaf_ctx_[right_leaf].out_lo, s.left_output);
    }
  }
  // walk UP from the new split; at every monotone ancestor whose opposite
  // subtree is contiguous, walk DOWN it to tighten leaf bounds
  std::vector<int> leaves_to_update;
  std::vector<int> up_feats;
  std::vector<uint32_t> up_thresholds;
  std::vector<uint8_t> up_was_right;
  const int split_feature = is_numerical ? tree->split_feature_inner(split_node) : -1;
  const uint32_t split_threshold = is_numerical ? tree->threshold_in_bin(split_node) : 0;
  int node = split_node;
  int parent = mono_node_parent_[node];
  while (parent >= 0) {
    const int feat = tree->split_feature_inner(parent);
    const bool p_numerical = !tree->IsCategoricalSplit(parent);
    const int orig_f = tree->split_feature(parent);
    int8_t mono = 0;
    if (orig_f >= 0 && orig_f < static_cast<int>(config_->monotone_constraints.size()))
      mono = static_cast<int8_t>(config_->monotone_constraints[orig_f]);
    const bool is_right_child = tree->right_child(parent) == node;
    // skip descents that cannot contain contiguous leaves: the same feature
    // crossed in the same direction earlier already fences the region
    bool should_descend = p_numerical;
    if (p_numerical) {
      for (size_t i = 0; i < up_feats.size(); ++i) {
        if (up_feats[i] == feat &&
            (up_was_right[i] != 0) == is_right_child) {
          should_descend = false;
          break;
        }
      }
    }
    if (should_descend) {
      if (mono != 0) {
        const int opposite = is_right_child ? tree->left_child(parent)
                                            : tree->right_child(parent);
        // mono<0 (decreasing): the LEFT subtree must stay ABOVE the right side;
        // coming up from the left child, the opposite (right) side gets a MAX cap
        const bool update_max = (mono < 0) == !is_right_child;
        MonoGoDown(tree, opposite, up_feats, up_thresholds, up_was_right, update_max,
                   split_feature, s, true, true, split_threshold, &leaves_to_update);
      }
      up_was_right.push_back(is_right_child ? 1 : 0);
      up_thresholds.push_back(tree->threshold_in_bin(parent));
      up_feats.push_back(feat);
    }
    node = parent;
    parent = mono_node_parent_[node];
  }
  // constraints changed: those leaves' cached best splits may now be invalid
  std::sort(leaves_to_update.begin(), leaves_to_update.end());
  leaves_to_update.erase(std::unique(leaves_to_update.begin(), leaves_to_update.end()),
                         leaves_to_update.end());
  for (int l : leaves_to_update) {
    if (l == left_leaf || l == right_leaf) continue;
    EnsureLeafHistogram(l);
    FindBestSplitForLeaf(l, leaf_ctx_[l]);
  }
}

void SerialTreeLearner::AddPredictionToScore(const Tree* tree, double* out_score) {
  if (tree->is_linear()) {
    // linear leaves read raw feature values per row: use the tree-walk path
    if (bag_indices_ != nullptr)
      tree->AddPredictionToScore(train_data_, bag_indices_, bag_cnt_, out_score);
    else
      tree->AddPredictionToScore(train_data_, train_data_->num_data(), out_score);
    return;
  }
  // the partition already maps every trained row to its leaf: O(n) scatter-add
  // instead of a per-row tree walk (reference ScoreUpdater::AddScore(tree_learner))
  const int nl = tree->num_leaves();
#pragma omp parallel for schedule(dynamic, 1)
  for (int l = 0; l < nl; ++l) {
    data_size_t cnt;
    const data_size_t* idx = partition_.GetIndexOnLeaf(l, &cnt);
    const double out = tree->LeafOutput(l);
    for (data_size_t i = 0; i < cnt; ++i) out_score[idx[i]] += out;
  }
}

void SerialTreeLearner::RenewTreeOutput(Tree* tree, const ObjectiveFunction* obj,
                                        std::function<double(const label_t*, int)>,
                                        data_size_t, const data_size_t*, data_size_t,
                                        const double* train_score) {
  if (obj == nullptr || !obj->NeedRenewTreeOutput()) return;
  // generic renewal: per leaf, objective-specific output from the rows in the leaf
  const int nl = tree->num_leaves();
  data_size_t cnt;
  std::vector<double> acc;  // distributed: (out*cnt, cnt) pairs for the global mean
  if (Network::is_distributed()) acc.assign(2 * nl, 0.0);
  for (int l = 0; l < nl; ++l) {
    const data_size_t* idx = partition_.GetIndexOnLeaf(l, &cnt);
    if (cnt == 0) continue;
    double new_out = obj->RenewTreeOutput(tree->LeafOutput(l), idx, cnt, train_score);
    if (Network::is_distributed()) {
      acc[2 * l] = new_out * cnt;
      acc[2 * l + 1] = cnt;
    } else {
      tree->SetLeafOutput(l, new_out);
    }
  }
  if (Network::is_distributed()) {
    // each rank renewed from its local shard; sync to the count-weighted mean so
    // every rank keeps the identical model (the reference leaves the outputs
    // rank-local, silently diverging the machines' models)
    Network::AllreduceSum(acc.data(), 2 * nl);
    for (int l = 0; l < nl; ++l)
      if (acc[2 * l + 1] > 0) tree->SetLeafOutput(l, acc[2 * l] / acc[2 * l + 1]);
  }
}

// ------------------------------------------------------------------ base fallbacks
Tree* TreeLearner::FitByExistingTree(const Tree*, const score_t*, const score_t*) {
  Log::Fatal("FitByExistingTree not supported by this learner");
  return nullptr;
}
Tree* TreeLearner::FitByExistingTree(const Tree* t, const std::vector<int>&, const score_t* g,
                                     const score_t* h) {
  return FitByExistingTree(t, g, h);
}

}  // namespace migbm
