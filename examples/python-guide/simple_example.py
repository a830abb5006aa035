"""Minimal train/eval/predict round (capability parity:
reference examples/python-guide/simple_example.py)."""
import numpy as np

import lightgbm_amd as lgb

rng = np.random.RandomState(0)
X = rng.randn(5000, 10)
y = X[:, 0] + 0.5 * np.sin(3 * X[:, 1]) + 0.1 * rng.randn(5000)
X_train, y_train = X[:4000], y[:4000]
X_test, y_test = X[4000:], y[4000:]

train_data = lgb.Dataset(X_train, label=y_train)
valid_data = train_data.create_valid(X_test, label=y_test)

params = {"objective": "regression", "metric": ["l2", "l1"], "num_leaves": 31,
          "learning_rate": 0.05}
evals = {}
bst = lgb.train(params, train_data, num_boost_round=100,
                valid_sets=[valid_data],
                callbacks=[lgb.early_stopping(10), lgb.record_evaluation(evals)])

print("best iteration:", bst.best_iteration)
pred = bst.predict(X_test, num_iteration=bst.best_iteration)
print("test rmse: %.4f" % np.sqrt(np.mean((pred - y_test) ** 2)))
bst.save_model("model.txt", num_iteration=bst.best_iteration)
bst2 = lgb.Booster(model_file="model.txt")
assert np.allclose(bst2.predict(X_test), pred)
