"""Functional shape metrics. Parity: reference functional/shape/procrustes.py."""
from metrics_amd.functional.shape.procrustes import procrustes_disparity

__all__ = ["procrustes_disparity"]
