"""Plotting utilities (parity target: reference python-package/lightgbm/plotting.py).
matplotlib/graphviz are optional; functions raise informative errors when missing."""
import numpy as np

from .basic import Booster
from .compat import GRAPHVIZ_INSTALLED, MATPLOTLIB_INSTALLED

__all__ = ["plot_importance", "plot_metric", "plot_tree", "create_tree_digraph"]


def _check_mpl():
    if not MATPLOTLIB_INSTALLED:
        raise ImportError("matplotlib is required for plotting (pip install matplotlib)")
    import matplotlib.pyplot as plt
    return plt


def plot_importance(booster, ax=None, height=0.2, xlim=None, ylim=None,
                    title="Feature importance", xlabel="Feature importance",
                    ylabel="Features", importance_type="auto", max_num_features=None,
                    ignore_zero=True, figsize=None, dpi=None, grid=True,
                    precision=3, **kwargs):
    plt = _check_mpl()
    if isinstance(booster, Booster):
        b = booster
    elif hasattr(booster, "booster_"):
        b = booster.booster_
    else:
        raise TypeError("booster must be a Booster or fitted LGBMModel")
    if importance_type == "auto":
        importance_type = "split"
    importance = b.feature_importance(importance_type=importance_type)
    names = b.feature_name()
    tuples = sorted(zip(names, importance), key=lambda x: x[1])
    if ignore_zero:
        tuples = [t for t in tuples if t[1] > 0]
    if max_num_features is not None and max_num_features > 0:
        tuples = tuples[-max_num_features:]
    if not tuples:
        raise ValueError("No features with non-zero importance")
    labels, values = zip(*tuples)
    if ax is None:
        _, ax = plt.subplots(1, 1, figsize=figsize, dpi=dpi)
    ylocs = np.arange(len(values))
    ax.barh(ylocs, values, align="center", height=height, **kwargs)
    for x, y in zip(values, ylocs):
        ax.text(x + 1, y, f"{x:.{precision}g}" if importance_type == "gain" else str(int(x)),
                va="center")
    ax.set_yticks(ylocs)
    ax.set_yticklabels(labels)
    if xlim is not None:
        ax.set_xlim(xlim)
    if ylim is not None:
        ax.set_ylim(ylim)
    ax.set_title(title)
    ax.set_xlabel(xlabel)
    ax.set_ylabel(ylabel)
    ax.grid(grid)
    return ax


def plot_metric(booster, metric=None, dataset_names=None, ax=None, xlim=None, ylim=None,
                title="Metric during training", xlabel="Iterations", ylabel="auto",
                figsize=None, dpi=None, grid=True):
    plt = _check_mpl()
    if isinstance(booster, dict):
        eval_results = booster
    elif hasattr(booster, "evals_result_"):
        eval_results = booster.evals_result_
    else:
        raise TypeError("booster must be an evals_result dict or fitted LGBMModel")
    if not eval_results:
        raise ValueError("eval results are empty (pass record_evaluation to training)")
    if ax is None:
        _, ax = plt.subplots(1, 1, figsize=figsize, dpi=dpi)
    names = dataset_names or list(eval_results.keys())
    chosen_metric = metric
    for name in names:
        metrics = eval_results[name]
        if chosen_metric is None:
            chosen_metric = next(iter(metrics))
        vals = metrics[chosen_metric]
        ax.plot(range(1, len(vals) + 1), vals, label=name)
    ax.legend(loc="best")
    if xlim is not None:
        ax.set_xlim(xlim)
    if ylim is not None:
        ax.set_ylim(ylim)
    ax.set_title(title)
    ax.set_xlabel(xlabel)
    ax.set_ylabel(chosen_metric if ylabel == "auto" else ylabel)
    ax.grid(grid)
    return ax


def _tree_to_graphviz(tree_info, feature_names, precision=3):
    import graphviz
    g = graphviz.Digraph()

    def add(node, parent=None, decision=None):
        if "leaf_index" in node:
            name = f"leaf{node['leaf_index']}"
            label = f"leaf {node['leaf_index']}: {node['leaf_value']:.{precision}g}"
            g.node(name, label=label)
        else:
            name = f"split{node['split_index']}"
            fid = node["split_feature"]
            fname = feature_names[fid] if fid < len(feature_names) else f"f{fid}"
            op = node.get("decision_type", "<=")
            label = f"{fname} {op} {node['threshold']:.{precision}g}"
            g.node(name, label=label)
            add(node["left_child"], name, "yes")
            add(node["right_child"], name, "no")
        if parent is not None:
            g.edge(parent, name, label=decision)
        return name

    add(tree_info["tree_structure"])
    return g


def create_tree_digraph(booster, tree_index=0, show_info=None, precision=3, **kwargs):
    if not GRAPHVIZ_INSTALLED:
        raise ImportError("graphviz is required for tree plotting")
    if hasattr(booster, "booster_"):
        booster = booster.booster_
    model = booster.dump_model()
    if tree_index >= len(model["tree_info"]):
        raise IndexError(f"tree_index {tree_index} out of range")
    return _tree_to_graphviz(model["tree_info"][tree_index], model["feature_names"],
                             precision)


def plot_tree(booster, ax=None, tree_index=0, figsize=None, dpi=None, precision=3,
              **kwargs):
    plt = _check_mpl()
    graph = create_tree_digraph(booster, tree_index=tree_index, precision=precision)
    import io
    try:
        s = graph.pipe(format="png")
    except Exception as e:
        raise RuntimeError(f"graphviz rendering failed: {e}")
    import matplotlib.image as mpimg
    if ax is None:
        _, ax = plt.subplots(1, 1, figsize=figsize, dpi=dpi)
    img = mpimg.imread(io.BytesIO(s))
    ax.imshow(img)
    ax.axis("off")
    return ax


def plot_split_value_histogram(booster, feature, bins=None, ax=None, width_coef=0.8,
                               xlim=None, ylim=None, title="Split value histogram for "
                               "feature with @index/name@ @feature@",
                               xlabel="Feature split value", ylabel="Count", figsize=None,
                               dpi=None, grid=True, **kwargs):
    """Plot the histogram of split threshold values used for `feature`
    (parity: reference plotting.plot_split_value_histogram)."""
    if not MATPLOTLIB_INSTALLED:
        raise ImportError("matplotlib is required for plotting")
    import matplotlib.pyplot as plt
    hist, edges = booster.get_split_value_histogram(feature, bins=bins)
    if hist.sum() == 0:
        raise ValueError(f"Cannot plot split value histogram: feature {feature} "
                         "was not used in splitting")
    if ax is None:
        _, ax = plt.subplots(1, 1, figsize=figsize, dpi=dpi)
    centers = (edges[:-1] + edges[1:]) / 2
    ax.bar(centers, hist, width=width_coef * (edges[1] - edges[0]))
    if xlim is not None:
        ax.set_xlim(xlim)
    if ylim is not None:
        ax.set_ylim(ylim)
    if title:
        ax.set_title(title.replace("@index/name@", "name" if isinstance(feature, str)
                                   else "index").replace("@feature@", str(feature)))
    ax.set_xlabel(xlabel)
    ax.set_ylabel(ylabel)
    if grid:
        ax.grid(True)
    return ax
