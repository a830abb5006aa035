"""sklearn-API walkthrough (capability parity:
reference examples/python-guide/sklearn_example.py)."""
import numpy as np
from sklearn.model_selection import GridSearchCV

import lightgbm_amd as lgb

rng = np.random.RandomState(0)
X = rng.randn(4000, 8)
y = X[:, 0] * 2 + X[:, 1] ** 2 + 0.1 * rng.randn(4000)

gbm = lgb.LGBMRegressor(num_leaves=31, learning_rate=0.05, n_estimators=40)
gbm.fit(X[:3000], y[:3000], eval_set=[(X[3000:], y[3000:])], eval_metric="l1",
        early_stopping_rounds=5)
print("best iteration:", gbm.best_iteration_)
print("feature importances:", list(gbm.feature_importances_))

gs = GridSearchCV(lgb.LGBMRegressor(verbosity=-1),
                  {"learning_rate": [0.01, 0.1], "n_estimators": [20, 40]}, cv=3)
gs.fit(X, y)
print("best params:", gs.best_params_)
