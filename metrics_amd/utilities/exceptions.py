"""Exception types.

Parity: torchmetrics ``utilities/exceptions.py`` (TorchMetricsUserError /
TorchMetricsUserWarning).
"""


class MetricsUserError(Exception):
    """Raised when the metrics API is used incorrectly."""


class MetricsUserWarning(UserWarning):
    """Warning category for metrics-API misuse that is recoverable."""


# Aliases matching the reference names, so user code catching the reference
# exception names can switch without edits.
TorchMetricsUserError = MetricsUserError
TorchMetricsUserWarning = MetricsUserWarning
