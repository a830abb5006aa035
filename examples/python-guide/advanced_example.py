"""Advanced features walkthrough: weights, init score, custom objective/metric,
continued training, model inspection (capability parity:
reference examples/python-guide/advanced_example.py)."""
import json

import numpy as np

import lightgbm_amd as lgb

rng = np.random.RandomState(0)
X = rng.randn(6000, 6)
y = (X[:, 0] + 0.5 * X[:, 1] > 0).astype(float)
w = np.where(y > 0, 2.0, 1.0)

train = lgb.Dataset(X[:5000], label=y[:5000], weight=w[:5000],
                    free_raw_data=False)
valid = train.create_valid(X[5000:], label=y[5000:])

params = {"objective": "binary", "metric": "auc", "num_leaves": 31}
bst = lgb.train(params, train, num_boost_round=10, valid_sets=[valid])

# continue training from the in-memory model, decayed learning rates
bst = lgb.train(params, train, num_boost_round=10, init_model=bst,
                valid_sets=[valid],
                callbacks=[lgb.reset_parameter(learning_rate=lambda i: 0.05 * (0.99 ** i))])

# custom objective + metric
def loglikelihood(preds, ds):
    labels = ds.get_label()
    p = 1.0 / (1.0 + np.exp(-preds))
    return (p - labels).astype(np.float32), (p * (1.0 - p)).astype(np.float32)

def binary_error(preds, ds):
    labels = ds.get_label()
    return "error", float(np.mean((preds > 0.0) != labels)), False

bst = lgb.train({"num_leaves": 31, "objective": "none"}, train, num_boost_round=10,
                valid_sets=[valid], fobj=loglikelihood, feval=binary_error)

# model inspection
print("trees:", bst.num_trees())
print("importance:", list(bst.feature_importance()))
d = bst.dump_model()
print("first tree keys:", sorted(d["tree_info"][0].keys()))
json.dumps(d)  # JSON-serializable
