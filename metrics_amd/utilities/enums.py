"""String-enum helpers used across the metric argument surface.

Parity: torchmetrics ``utilities/enums.py``.
"""
from __future__ import annotations

from enum import Enum
from typing import Optional


class EnumStr(str, Enum):
    """Case/format-insensitive string enum with a helpful error message."""

    @staticmethod
    def _name() -> str:
        return "Task"

    @classmethod
    def from_str(cls, value: str, source: str = "key") -> "EnumStr":
        try:
            return cls[value.replace("-", "_").upper()]
        except KeyError as err:
            valid = [m.lower() for m in cls._member_names_]
            raise ValueError(
                f"Invalid {cls._name()}: expected one of {valid}, but got {value}."
            ) from err

    def __str__(self) -> str:
        return self.value.lower()

    def __eq__(self, other: object) -> bool:
        if isinstance(other, str):
            return self.value.lower() == other.replace("-", "_").lower()
        return Enum.__eq__(self, other)

    def __hash__(self) -> int:
        return hash(self.value.lower())


class DataType(EnumStr):
    @staticmethod
    def _name() -> str:
        return "Data type"

    BINARY = "binary"
    MULTILABEL = "multi-label"
    MULTICLASS = "multi-class"
    MULTIDIM_MULTICLASS = "multi-dim multi-class"


class AverageMethod(EnumStr):
    @staticmethod
    def _name() -> str:
        return "Average method"

    MICRO = "micro"
    MACRO = "macro"
    WEIGHTED = "weighted"
    NONE = None  # type: ignore[assignment]
    SAMPLES = "samples"


class MDMCAverageMethod(EnumStr):
    GLOBAL = "global"
    SAMPLEWISE = "samplewise"


class ClassificationTask(EnumStr):
    """The three classification task flavors used by the task-dispatch wrappers."""

    @staticmethod
    def _name() -> str:
        return "Classification"

    BINARY = "binary"
    MULTICLASS = "multiclass"
    MULTILABEL = "multilabel"


class ClassificationTaskNoBinary(EnumStr):
    @staticmethod
    def _name() -> str:
        return "Classification"

    MULTICLASS = "multiclass"
    MULTILABEL = "multilabel"


class ClassificationTaskNoMultilabel(EnumStr):
    @staticmethod
    def _name() -> str:
        return "Classification"

    BINARY = "binary"
    MULTICLASS = "multiclass"


def _resolve_task(task: str, enum_cls: type = ClassificationTask) -> EnumStr:
    return enum_cls.from_str(str(task))


__all__ = [
    "EnumStr",
    "DataType",
    "AverageMethod",
    "MDMCAverageMethod",
    "ClassificationTask",
    "ClassificationTaskNoBinary",
    "ClassificationTaskNoMultilabel",
]
