/*!
 * migbm TreeLearner interface + SerialTreeLearner (CPU leaf-wise histogram learner).
 * Capability parity target: reference include/LightGBM/tree_learner.h,
 * src/treelearner/serial_tree_learner.{h,cpp}, data_partition.hpp. Fresh implementation.
 */
#ifndef MIGBM_TREE_LEARNER_H_
#define MIGBM_TREE_LEARNER_H_

#include "config.h"
#include "dataset.h"
#include "feature_histogram.h"
#include "tree.h"

#include <memory>
#include <set>
#include <vector>

namespace migbm {

/*! forced-splits JSON node (forcedsplits_filename; reference format). */
struct ForcedNode {
  int feature = -1;
  double threshold = 0.0;
  std::unique_ptr<ForcedNode> left, right;
};
/*! parse the forced-splits JSON file (empty path / bad file -> nullptr). */
std::unique_ptr<ForcedNode> ParseForcedSplits(const std::string& path);

class TreeLearner {
 public:
  virtual ~TreeLearner() = default;
  virtual void Init(const Dataset* train_data, bool is_constant_hessian) = 0;
  virtual void ResetTrainingData(const Dataset* train_data) = 0;
  virtual void ResetConfig(const Config* config) = 0;
  /*! Train one tree from gradients/hessians (already bagging-masked via SetBaggingData). */
  virtual Tree* Train(const score_t* gradients, const score_t* hessians, bool is_first_tree) = 0;
  virtual void SetBaggingData(const Dataset* subset, const data_size_t* used_indices,
                              data_size_t num_data) = 0;
  /*! Refit an existing tree structure to new gradients. */
  virtual Tree* FitByExistingTree(const Tree* old_tree, const score_t* g, const score_t* h);
  virtual Tree* FitByExistingTree(const Tree* old_tree, const std::vector<int>& leaf_pred,
                                  const score_t* g, const score_t* h);
  virtual void AddPredictionToScore(const Tree* tree, double* out_score) = 0;
  virtual void RenewTreeOutput(Tree* tree, const class ObjectiveFunction* obj,
                               std::function<double(const label_t*, int)> residual_getter,
                               data_size_t total_num_data, const data_size_t* bag_indices,
                               data_size_t bag_cnt, const double* train_score) = 0;

  // ---- device-resident boosting hooks (HIP learner; no-ops on CPU learners).
  // The GBDT engine keeps one code path: when IsHIPLearner() is true the learner owns
  // device score/grad/hess buffers and these hooks keep them in sync.
  virtual bool IsHIPLearner() const { return false; }
  /*! true if grad/hess for this objective are computed by a device kernel */
  virtual bool DeviceObjectiveSupported(const std::string& objective_name) const {
    (void)objective_name;
    return false;
  }
  /*! compute grad/hess on device from device scores (device objectives only) */
  virtual void DeviceBoosting(const class ObjectiveFunction* obj) { (void)obj; }
  /*! add a constant to the device score vector (boost_from_average) */
  virtual void DeviceAddInitScore(double v) { (void)v; }
  /*! multiclass: select which class's device score buffer subsequent
   *  Train/UpdateScore/DeviceAddInitScore calls operate on. */
  virtual void SetClassOffset(int class_id) { (void)class_id; }
  /*! download the device train scores into a host buffer (for metrics / custom obj) */
  virtual void DownloadTrainScore(double* dst) { (void)dst; }
  /*! push host-modified train scores back to the device (DART drop/renormalize). */
  virtual void UploadTrainScore(const double* src) { (void)src; }
  /*! device pointwise train-metric eval: reduce (Σ w·loss, Σ w) on the GPU so
   *  eval does not download the score vector. `convert_kind` selects the
   *  objective's output transform (0 identity, 1 exp, 2 sigmoid(param),
   *  3 logistic, 4 log1p(exp), 5 signed-square). Returns false when the learner
   *  cannot evaluate this loss on device (host fallback). */
  virtual bool DeviceEvalPointwise(int loss_kind, double loss_a, int convert_kind,
                                   double convert_param, double* out_sum,
                                   double* out_wsum) {
    (void)loss_kind; (void)loss_a; (void)convert_kind; (void)convert_param;
    (void)out_sum; (void)out_wsum;
    return false;
  }

  static TreeLearner* Create(const std::string& learner_type, const std::string& device_type,
                             const Config* config);
};

/*! Leaf -> row-index partition with multithreaded stable split. */
class DataPartition {
 public:
  void Init(data_size_t num_data, int num_leaves) {
    num_data_ = num_data;
    indices_.resize(num_data);
    temp_.resize(num_data);
    leaf_begin_.assign(num_leaves, 0);
    leaf_count_.assign(num_leaves, 0);
    row_to_leaf_.assign(num_data, 0);
  }
  /*! reset to a single root leaf holding all (or bagged) rows */
  void ResetToRoot(const data_size_t* used_indices, data_size_t cnt) {
    if (used_indices != nullptr) {
      std::copy(used_indices, used_indices + cnt, indices_.begin());
      used_cnt_ = cnt;
    } else {
      used_cnt_ = num_data_;
#pragma omp parallel for schedule(static)
      for (data_size_t i = 0; i < num_data_; ++i) indices_[i] = i;
    }
    std::fill(leaf_begin_.begin(), leaf_begin_.end(), 0);
    std::fill(leaf_count_.begin(), leaf_count_.end(), 0);
    leaf_count_[0] = used_cnt_;
  }
  const data_size_t* GetIndexOnLeaf(int leaf, data_size_t* out_cnt) const {
    *out_cnt = leaf_count_[leaf];
    return indices_.data() + leaf_begin_[leaf];
  }
  data_size_t leaf_count(int leaf) const { return leaf_count_[leaf]; }

  /*! Stable-partition rows of `leaf` into (leaf, right_leaf) by predicate go_left(row). */
  void Split(int leaf, int right_leaf, const std::function<bool(data_size_t)>& go_left);
  /*! fast path for plain dense-uint8 numeric splits: the per-row decision is
   *  inlined (no std::function indirection) — the dominant case. */
  void SplitDenseU8(int leaf, int right_leaf, const uint8_t* col, uint32_t thr,
                    int nan_bin, bool default_left);

  data_size_t used_cnt() const { return used_cnt_; }

 private:
  data_size_t num_data_ = 0;
  data_size_t used_cnt_ = 0;
  std::vector<data_size_t> indices_, temp_;
  std::vector<data_size_t> leaf_begin_, leaf_count_;
  std::vector<int> row_to_leaf_;  // reserved for score updater use
};

class SerialTreeLearner : public TreeLearner {
 public:
  explicit SerialTreeLearner(const Config* config) : config_(config) {}
  void Init(const Dataset* train_data, bool is_constant_hessian) override;
  void ResetTrainingData(const Dataset* train_data) override;
  void ResetConfig(const Config* config) override { config_ = config; }
  virtual Tree* Train(const score_t* gradients, const score_t* hessians,
                      bool is_first_tree) override;
  void SetBaggingData(const Dataset* subset, const data_size_t* used_indices,
                      data_size_t num_data) override;
  void AddPredictionToScore(const Tree* tree, double* out_score) override;
  void RenewTreeOutput(Tree* tree, const class ObjectiveFunction* obj,
                       std::function<double(const label_t*, int)> residual_getter,
                       data_size_t total_num_data, const data_size_t* bag_indices,
                       data_size_t bag_cnt, const double* train_score) override;

 protected:
  /*! Histogram for `leaf` into its slot; optionally by subtraction (parent - sibling). */
  void ComputeHistogram(int leaf, data_size_t cnt, const data_size_t* indices);
  /*! hook after a leaf histogram is built. Base: reconstruct the shared default bin
   *  of EFB-bundled features from the leaf totals (the EFB form of FixHistogram).
   *  Distributed learners reduce first, then call this base. */
  virtual void OnHistogramReady(int leaf) { MaterializeDefaultBins(leaf); }
  /*! Reconstruct implicit bins from leaf totals: the shared default bin of
   *  EFB-bundled features AND the default bin of sparse columns (the migbm
   *  analogue of the reference's FixHistogram). */
  void MaterializeDefaultBins(int leaf) {
    if (!train_data_->has_bundles() && !train_data_->has_sparse()) return;
    hist_t* hist = HistSlot(leaf_to_slot_[leaf]);
    const LeafContext& ctx = leaf_ctx_[leaf];
    const int nf = train_data_->num_features();
    for (int f = 0; f < nf; ++f) {
      int def_bin = -1;
      if (train_data_->feature_bundled(f)) def_bin = 0;
      else def_bin = train_data_->feature_sparse_default_bin(f);
      if (def_bin < 0) continue;
      hist_t* fh = hist + 2 * train_data_->hist_offset(f);
      double g = 0, h = 0;
      const int nb = train_data_->FeatureNumBin(f);
      for (int b = 0; b < nb; ++b) {
        if (b == def_bin) continue;
        g += fh[2 * b];
        h += fh[2 * b + 1];
      }
      fh[2 * def_bin] = ctx.sum_gradient - g;
      fh[2 * def_bin + 1] = ctx.sum_hessian - h;
    }
  }
  void MaterializeBundledBin0(int leaf) { MaterializeDefaultBins(leaf); }  // legacy name
  /*! hook to globalize root stats (data-parallel: allreduce). */
  virtual void ReduceRootStats(double* sum_g, double* sum_h, data_size_t* cnt) {
    (void)sum_g; (void)sum_h; (void)cnt;
  }
  /*! hook to globalize child row counts after a split (data-parallel: allreduce). */
  virtual void GlobalChildCounts(data_size_t* left_cnt, data_size_t* right_cnt) {
    (void)left_cnt; (void)right_cnt;
  }
  void SubtractHistogram(int dst_slot_leaf, int parent_slot, int sibling_slot);
  /*! Scan all used features of leaf, fill best_split_per_leaf_[leaf]. */
  virtual void FindBestSplitForLeaf(int leaf, const LeafContext& ctx);
  hist_t* HistSlot(int slot) { return hist_store_.data() + static_cast<size_t>(slot) * 2 * train_data_->num_total_bin(); }
  /*! per-node feature sampling mask (feature_fraction / bynode / interaction constraints) */
  std::vector<int8_t> SampleFeatures(bool per_node);
  /*! linear-tree post-pass: ridge-weighted LS fit per leaf */
  void CalculateLinear(Tree* tree);
  /*! predicate for partition: does row go left under split s of inner feature f? */
  std::function<bool(data_size_t)> MakeGoLeft(const SplitInfo& s) const;

  const Config* config_;
  const Dataset* train_data_ = nullptr;
  bool is_constant_hessian_ = false;
  const score_t* gradients_ = nullptr;
  const score_t* hessians_ = nullptr;
  std::vector<score_t> ordered_grad_, ordered_hess_;
  std::vector<uint8_t> in_leaf_mask_;   // sparse-column hist membership scratch
  // empirical col-wise vs row-wise histogram selection (reference
  // TrainingShareStates behavior): the first two root histograms are timed, one
  // per mode, then the faster mode is locked in. -1 undecided, 0 col, 1 row.
  int hist_mode_ = -1;
  double hist_trial_time_[2] = {0.0, 0.0};
  int hist_trials_done_ = 0;
  // per-tree hessian classification (1 = constant -> count mode) and the
  // interleaved (g,h) pair array the varying-hessian row-wise loop reads
  int tree_const_hess_ = -1;
  std::vector<score_t> gh_;
  // LOCAL (pre-reduce) leaf gradient totals of the last ComputeHistogram call;
  // distributed learners use them to materialize default bins before reducing
  double local_leaf_sum_g_ = 0.0, local_leaf_sum_h_ = 0.0;
  std::vector<score_t> quant_grad_, quant_hess_;  // CPU quantized-training grids
  uint32_t quant_seed_ = 0x9E3779B9u;
  DataPartition partition_;
  std::vector<hist_t> hist_store_;          // num_leaves slots x 2*num_total_bin
  std::vector<int> leaf_to_slot_;
  // histogram_pool_size enforcement (reference HistogramPool): when the cap is
  // below one-slot-per-leaf, slots are LRU-shared; an evicted parent histogram
  // forces the larger child to be recomputed instead of subtracted
  int pool_slots_ = 0;
  std::vector<int> slot_owner_;     // slot -> leaf (-1 free)
  std::vector<int64_t> slot_used_;  // LRU stamps
  int64_t slot_clock_ = 0;
  int AcquireSlot(int leaf, int pin_a = -1, int pin_b = -1);
  /*! rebuild a leaf's histogram if the pool evicted it */
  void EnsureLeafHistogram(int leaf);
  // ---- intermediate monotone constraints (monotone_constraints_method !=
  // "basic"): after a split inside a monotone subtree, bounds of CONTIGUOUS
  // leaves elsewhere in the tree are tightened against the new outputs and
  // their best splits recomputed (reference IntermediateLeafConstraints,
  // monotone_constraints.hpp:516-858 — algorithm reimplemented fresh)
  bool mono_intermediate_ = false;
  std::vector<int> mono_node_parent_;       // internal node -> parent (-1 root)
  std::vector<uint8_t> mono_leaf_in_subtree_;
  void MonotoneIntermediateUpdate(const Tree* tree, int left_leaf, int right_leaf,
                                  const SplitInfo& s, int split_node,
                                  bool is_numerical);
  void MonoGoDown(const Tree* tree, int node, const std::vector<int>& up_feats,
                  const std::vector<uint32_t>& up_thresholds,
                  const std::vector<uint8_t>& up_was_right, bool update_max,
                  int split_feature, const SplitInfo& s, bool use_left,
                  bool use_right, uint32_t split_threshold,
                  std::vector<int>* leaves_to_update);
  // ---- advanced ("monotone precise") mode: per-(leaf,feature) per-bin output
  // bounds recomputed from the CURRENT tree at scan time (reference
  // AdvancedLeafConstraints GoUp/GoDownToFindConstrainingLeaves,
  // monotone_constraints.hpp:858-1178 — fresh dense-per-bin design)
  bool mono_advanced_ = false;
  const Tree* mono_tree_ = nullptr;         // current tree during Train
  void MonoAdvBoundsForFeature(int leaf, int feature_inner, int num_numeric_bin,
                               MonoAdvBounds* out) const;
  void MonoAdvGoDown(const Tree* tree, int node, int feature_inner, bool want_min,
                     int s, int e, const std::vector<int>& up_feats,
                     const std::vector<uint32_t>& up_thresholds,
                     const std::vector<uint8_t>& up_was_right,
                     MonoAdvBounds* out) const;
  std::vector<SplitInfo> best_split_per_leaf_;
  std::vector<LeafContext> leaf_ctx_;
  std::vector<int8_t> is_feature_used_;     // per-tree mask
  // bagging
  const data_size_t* bag_indices_ = nullptr;
  data_size_t bag_cnt_ = 0;
  Random feature_rng_{0};
  Random extra_rng_{0};
  int iter_counter_ = 0;
  bool build_both_children_ = false;  // voting-parallel: no histogram subtraction
  // forced splits (forcedsplits_filename JSON)
  std::unique_ptr<ForcedNode> forced_root_;
  std::vector<const ForcedNode*> forced_of_leaf_;
  bool MakeForcedSplit(int leaf, const LeafContext& ctx, const ForcedNode* node,
                       SplitInfo* out);
  // CEGB / interaction-constraint state
  std::vector<int8_t> cegb_feature_used_;
  std::vector<std::set<int>> leaf_branch_features_;
  std::vector<std::set<int>> interaction_groups_;
};

/*! set by the HIP learner's static registrar when the device module is linked in */
extern TreeLearner* (*g_create_hip_learner)(const Config*);

}  // namespace migbm

#endif  // MIGBM_TREE_LEARNER_H_
