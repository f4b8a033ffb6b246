"""Modular Concordance correlation. Parity: torchmetrics ``regression/concordance.py``."""
from __future__ import annotations

from typing import Any, Optional

from torch import Tensor

from metrics_amd.regression.pearson import PearsonCorrCoef
from metrics_amd.functional.regression.concordance import _concordance_corrcoef_compute
from metrics_amd.functional.regression.pearson import _final_aggregation


class ConcordanceCorrCoef(PearsonCorrCoef):
    """Concordance correlation coefficient (stateful)."""

    def compute(self) -> Tensor:
        """CCC from the running moments."""
        if (self.num_outputs == 1 and self.mean_x.numel() > 1) or (self.num_outputs > 1 and self.mean_x.ndim > 1):
            mean_x, mean_y, var_x, var_y, corr_xy, n_total = _final_aggregation(
                self.mean_x, self.mean_y, self.var_x, self.var_y, self.corr_xy, self.n_total
            )
        else:
            mean_x, mean_y = self.mean_x, self.mean_y
            var_x, var_y, corr_xy, n_total = self.var_x, self.var_y, self.corr_xy, self.n_total
        return _concordance_corrcoef_compute(mean_x, mean_y, var_x.clone(), var_y.clone(), corr_xy.clone(), n_total)
