"""Functional metric API (L3). Parity: torchmetrics ``functional/__init__.py``."""
from metrics_amd.functional.classification import *  # noqa: F401,F403
from metrics_amd.functional.regression import *  # noqa: F401,F403
from metrics_amd.functional import classification, regression  # noqa: F401
