# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Feature/Entity metadata + per-value validators
(reference mlrun/features.py): ``Validator`` base with type checking,
``MinMaxValidator`` (value range), ``MinMaxLenValidator`` (length
range), ``RegexValidator`` (fullmatch), and the ``validator_kinds``
registry used to round-trip validators through feature-set dicts.

``Feature``/``Entity`` themselves live in
``feature_store.feature_set``; this module re-exports them and adds
the validator machinery so ``mlrun.features`` imports work unchanged."""

import math
import re

from .data_types import ValueType
from .feature_store.feature_set import Entity, Feature  # noqa: F401 re-export
from .model import ModelObj


def _clip_str(value, max_size: int = 40) -> str:
    text = str(value)
    return text[:max_size] + "..." if len(text) > max_size else text


class _RangeCheck:
    """Numeric-range type check for one ValueType."""

    def __init__(self, min_value, max_value):
        self.min = min_value
        self.max = max_value

    def check(self, value_type, value):
        try:
            if value < self.min:
                return False, {"message": "Value is smaller than min range",
                               "type": value_type, "min range": self.min,
                               "value": _clip_str(value)}
            if value > self.max:
                return False, {"message": "Value is greater than max range",
                               "type": value_type, "max range": self.max,
                               "value": _clip_str(value)}
        except Exception as err:
            return False, {"message": str(err), "type": value_type}
        return True, {}


class _ConvertCheck:
    """Type check by attempted conversion (float/bytes)."""

    def __init__(self, func):
        self.func = func

    def check(self, value_type, value):
        try:
            self.func(value)
        except Exception as err:
            return False, {"message": str(err), "type": value_type}
        return True, {}


# per-ValueType type validators (reference features.py:306
# ``type_validator``); bool/str/datetime need no check
type_validator = {
    ValueType.INT8: _RangeCheck(-(2 ** 7), 2 ** 7 - 1),
    ValueType.INT16: _RangeCheck(-(2 ** 15), 2 ** 15 - 1),
    ValueType.INT32: _RangeCheck(-(2 ** 31), 2 ** 31 - 1),
    ValueType.INT64: _RangeCheck(-(2 ** 63), 2 ** 63 - 1),
    ValueType.UINT8: _RangeCheck(0, 2 ** 8 - 1),
    ValueType.UINT16: _RangeCheck(0, 2 ** 16 - 1),
    ValueType.UINT32: _RangeCheck(0, 2 ** 32 - 1),
    ValueType.UINT64: _RangeCheck(0, 2 ** 64 - 1),
    ValueType.FLOAT: _ConvertCheck(float),
    ValueType.DOUBLE: _ConvertCheck(float),
    ValueType.BYTES: _ConvertCheck(bytes),
}
# INT128/UINT128 (reference uses math.pow bounds)
type_validator["int128"] = _RangeCheck(-math.pow(2, 127),
                                       math.pow(2, 127) - 1)
type_validator["uint128"] = _RangeCheck(0, math.pow(2, 128))


class Validator(ModelObj):
    """Base per-value validator (reference features.py:228): optional
    value_type check, subclasses add constraints.  ``check`` returns
    ``(ok, info_dict)``."""

    kind = ""
    _dict_fields = ["kind", "check_type", "severity"]

    def __init__(self, check_type: bool = None, severity: str = None):
        self._feature = None
        self.check_type = check_type
        self.severity = severity

    def set_feature(self, feature):
        self._feature = feature

    def check(self, value):
        if self.check_type and self._feature is not None:
            value_type = getattr(self._feature, "value_type", None)
            checker = type_validator.get(value_type)
            if checker is not None:
                return checker.check(value_type, value)
        return True, {}


class MinMaxValidator(Validator):
    """Valid when min <= value <= max (reference features.py:263)."""

    kind = "minmax"
    _dict_fields = Validator._dict_fields + ["min", "max"]

    def __init__(self, check_type: bool = None, severity: str = None,
                 min=None, max=None):
        super().__init__(check_type, severity)
        self.min = min
        self.max = max

    def check(self, value):
        ok, info = super().check(value)
        if not ok:
            return ok, info
        try:
            if self.min is not None and value < self.min:
                return False, {"message": "value is smaller than min",
                               "min": self.min,
                               "value": _clip_str(value)}
            if self.max is not None and value > self.max:
                return False, {"message": "value is greater than max",
                               "max": self.max,
                               "value": _clip_str(value)}
        except Exception as err:
            return False, {"message": str(err), "type": self.kind}
        return ok, info


class MinMaxLenValidator(Validator):
    """Valid when min <= len(value) <= max (reference features.py:324)."""

    kind = "minmaxlen"
    _dict_fields = Validator._dict_fields + ["min", "max"]

    def __init__(self, check_type: bool = None, severity: str = None,
                 min=None, max=None):
        super().__init__(check_type, severity)
        self.min = min
        self.max = max

    def check(self, value):
        ok, info = super().check(value)
        if not ok:
            return ok, info
        try:
            if self.min is not None and len(value) < self.min:
                return False, {"message": "Length value is smaller than min",
                               "min": self.min,
                               "length value": len(value)}
            if self.max is not None and len(value) > self.max:
                return False, {"message": "Length value is greater than max",
                               "max": self.max,
                               "length value": len(value)}
        except Exception as err:
            return False, {"message": str(err), "type": self.kind}
        return ok, info


class RegexValidator(Validator):
    """Valid when the regex fullmatches str(value)
    (reference features.py:387)."""

    kind = "regex"
    _dict_fields = Validator._dict_fields + ["regex"]

    def __init__(self, check_type: bool = None, severity: str = None,
                 regex=None):
        super().__init__(check_type, severity)
        self.regex = regex
        self.regex_compile = re.compile(regex) if regex else None

    def check(self, value):
        ok, info = super().check(value)
        if not ok:
            return ok, info
        try:
            if self.regex_compile is not None and \
                    not self.regex_compile.fullmatch(str(value)):
                return False, {
                    "message": "Value is not valid with regular expression",
                    "regexp": self.regex, "value": _clip_str(value)}
        except Exception as err:
            return False, {"message": str(err), "type": self.kind}
        return ok, info

    @classmethod
    def from_dict(cls, struct=None, fields=None, deprecated_fields=None):
        obj = super().from_dict(struct=struct, fields=fields,
                                deprecated_fields=deprecated_fields)
        obj.regex_compile = re.compile(obj.regex) if obj.regex else None
        return obj


validator_kinds = {
    "": Validator,
    "minmax": MinMaxValidator,
    "minmaxlen": MinMaxLenValidator,
    "regex": RegexValidator,
}


def validator_from_dict(struct) -> Validator:
    """Rebuild a validator from its dict form (Feature.validator
    setter path, reference features.py:125)."""
    if isinstance(struct, Validator):
        return struct
    kind = (struct or {}).get("kind", "")
    cls = validator_kinds.get(kind)
    if cls is None:
        raise ValueError(f"unknown validator kind {kind!r}")
    return cls.from_dict(struct)


__all__ = ["Entity", "Feature", "Validator", "MinMaxValidator",
           "MinMaxLenValidator", "RegexValidator", "validator_kinds",
           "validator_from_dict", "type_validator"]
