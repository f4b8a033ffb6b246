from enum import Enum


class StrEnum(str, Enum):
    @classmethod
    def from_str(cls, value, source="key"):
        try:
            return cls[value.replace("-", "_").upper()]
        except KeyError:
            for m in cls:
                if m.value.lower() == value.lower():
                    return m
            raise
    def __str__(self):
        return self.value.lower()
