"""Optional-dependency shims (parity target: reference python-package/lightgbm/compat.py)."""

__all__ = [
    "PANDAS_INSTALLED", "pd_DataFrame", "pd_Series",
    "SCIPY_INSTALLED", "scipy_sparse",
    "MATPLOTLIB_INSTALLED", "GRAPHVIZ_INSTALLED",
    "SKLEARN_INSTALLED",
]

try:
    import pandas as pd
    pd_DataFrame = pd.DataFrame
    pd_Series = pd.Series
    PANDAS_INSTALLED = True
except ImportError:
    PANDAS_INSTALLED = False

    class pd_DataFrame:  # noqa: N801
        pass

    class pd_Series:  # noqa: N801
        pass

try:
    import scipy.sparse as scipy_sparse
    SCIPY_INSTALLED = True
except ImportError:
    scipy_sparse = None
    SCIPY_INSTALLED = False

try:
    import matplotlib  # noqa: F401
    MATPLOTLIB_INSTALLED = True
except ImportError:
    MATPLOTLIB_INSTALLED = False

try:
    import graphviz  # noqa: F401
    GRAPHVIZ_INSTALLED = True
except ImportError:
    GRAPHVIZ_INSTALLED = False

try:
    import sklearn  # noqa: F401
    SKLEARN_INSTALLED = True
except ImportError:
    SKLEARN_INSTALLED = False
