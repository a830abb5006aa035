/*! migbm TreeSHAP — exact per-tree SHAP value computation for predict_contrib.
 *  Algorithm: Lundberg et al. "Consistent Individualized Feature Attribution for Tree
 *  Ensembles" (the reference implements the same algorithm in tree.h TreeSHAP). */
#include "migbm/tree.h"

#include <vector>

namespace migbm {

namespace {

struct PathElem {
  int feature_index;
  double zero_fraction;
  double one_fraction;
  double pweight;
};

void ExtendPath(PathElem* path, int depth, double zero_fraction, double one_fraction,
                int feature_index) {
  path[depth].feature_index = feature_index;
  path[depth].zero_fraction = zero_fraction;
  path[depth].one_fraction = one_fraction;
  path[depth].pweight = depth == 0 ? 1.0 : 0.0;
  for (int i = depth - 1; i >= 0; --i) {
    path[i + 1].pweight += one_fraction * path[i].pweight * (i + 1) / static_cast<double>(depth + 1);
    path[i].pweight = zero_fraction * path[i].pweight * (depth - i) / static_cast<double>(depth + 1);
  }
}

void UnwindPath(PathElem* path, int depth, int path_index) {
  const double one_fraction = path[path_index].one_fraction;
  const double zero_fraction = path[path_index].zero_fraction;
  double next_one_portion = path[depth].pweight;
  for (int i = depth - 1; i >= 0; --i) {
    if (one_fraction != 0) {
      const double tmp = path[i].pweight;
      path[i].pweight = next_one_portion * (depth + 1) / ((i + 1) * one_fraction);
      next_one_portion = tmp - path[i].pweight * zero_fraction * (depth - i) /
                                   static_cast<double>(depth + 1);
    } else {
      path[i].pweight = (path[i].pweight * (depth + 1)) /
                        (zero_fraction * (depth - i));
    }
  }
  for (int i = path_index; i < depth; ++i) {
    path[i].feature_index = path[i + 1].feature_index;
    path[i].zero_fraction = path[i + 1].zero_fraction;
    path[i].one_fraction = path[i + 1].one_fraction;
  }
}

double UnwoundPathSum(const PathElem* path, int depth, int path_index) {
  const double one_fraction = path[path_index].one_fraction;
  const double zero_fraction = path[path_index].zero_fraction;
  double next_one_portion = path[depth].pweight;
  double total = 0;
  for (int i = depth - 1; i >= 0; --i) {
    if (one_fraction != 0) {
      const double tmp = next_one_portion * (depth + 1) / ((i + 1) * one_fraction);
      total += tmp;
      next_one_portion = path[i].pweight - tmp * zero_fraction * (depth - i) /
                                               static_cast<double>(depth + 1);
    } else if (zero_fraction != 0) {
      total += (path[i].pweight / zero_fraction) / ((depth - i) /
                                                    static_cast<double>(depth + 1));
    }
  }
  return total;
}

void TreeSHAPRec(const Tree* tree, const double* x, double* phi, int node, int depth,
                 PathElem* parent_path, double parent_zero_fraction,
                 double parent_one_fraction, int parent_feature_index) {
  // copy parent path (fresh storage segment past the parent's depth+1 entries)
  PathElem* path = parent_path + depth + 1;
  for (int i = 0; i < depth; ++i) path[i] = parent_path[i];
  ExtendPath(path, depth, parent_zero_fraction, parent_one_fraction, parent_feature_index);

  if (node < 0) {
    // leaf
    const int leaf = ~node;
    for (int i = 1; i <= depth; ++i) {
      const double w = UnwoundPathSum(path, depth, i);
      phi[path[i].feature_index] += w * (path[i].one_fraction - path[i].zero_fraction) *
                                    tree->LeafOutput(leaf);
    }
    return;
  }
  // internal node
  const int hot = tree->Decision(x, node);
  const int cold = hot == tree->left_child(node) ? tree->right_child(node)
                                                 : tree->left_child(node);
  auto node_count = [&](int n) -> double {
    return n < 0 ? std::max(1, tree->leaf_count(~n))
                 : std::max(1.0, tree->InternalCountSafe(n));
  };
  const double w_total = node_count(node);
  const double hot_zero_fraction = node_count(hot) / w_total;
  const double cold_zero_fraction = node_count(cold) / w_total;
  double incoming_zero_fraction = 1.0;
  double incoming_one_fraction = 1.0;
  // if this feature was seen before, undo and combine
  const int split_f = tree->split_feature(node);
  int path_index = 0;
  for (; path_index <= depth; ++path_index)
    if (path[path_index].feature_index == split_f) break;
  if (path_index != depth + 1) {
    incoming_zero_fraction = path[path_index].zero_fraction;
    incoming_one_fraction = path[path_index].one_fraction;
    UnwindPath(path, depth, path_index);
    --depth;
  }
  TreeSHAPRec(tree, x, phi, hot, depth + 1, path,
              hot_zero_fraction * incoming_zero_fraction, incoming_one_fraction, split_f);
  TreeSHAPRec(tree, x, phi, cold, depth + 1, path,
              cold_zero_fraction * incoming_zero_fraction, 0.0, split_f);
}

double ExpectedValue(const Tree* tree, int node) {
  if (node < 0) return tree->LeafOutput(~node);
  double lw = node < 0 ? 1 : 1;
  (void)lw;
  auto cnt = [&](int n) -> double {
    return n < 0 ? std::max(1, tree->leaf_count(~n)) : std::max(1.0, tree->InternalCountSafe(n));
  };
  double l = cnt(tree->left_child(node)), r = cnt(tree->right_child(node));
  return (l * ExpectedValue(tree, tree->left_child(node)) +
          r * ExpectedValue(tree, tree->right_child(node))) / (l + r);
}

}  // namespace

void TreeSHAP(const Tree* tree, const double* features, double* phi, int num_features) {
  if (tree->num_leaves() <= 1) {
    phi[num_features] += tree->LeafOutput(0);
    return;
  }
  phi[num_features] += ExpectedValue(tree, 0);
  const int max_depth = tree->num_leaves() + 1;
  std::vector<PathElem> path_store(static_cast<size_t>(max_depth + 2) * (max_depth + 2));
  TreeSHAPRec(tree, features, phi, 0, 0, path_store.data(), 1.0, 1.0, -1);
}

}  // namespace migbm
