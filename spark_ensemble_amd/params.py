"""Typed parameter system.

MI355X-native re-design of the Spark ML ``Param`` registry that the reference
library uses as its entire configuration surface (reference:
core/src/main/scala/org/apache/spark/ml/ensemble/ensembleParams.scala and the
Spark ``org.apache.spark.ml.param`` package it builds on).  We reproduce the
observable contract — typed params with validators, per-class defaults,
``copy(extra)`` cloning, JSON metadata encode/decode with nested estimators
excluded — without any of the JVM machinery.

Key semantics kept from the reference:
  * every param has a (parent uid, name, doc, validator) identity
    (Spark ``Param``; see ensembleParams.scala:32-49 for a typical use),
  * ``set``/``get``/``getOrDefault``/``isDefined`` accessor family,
  * ``explainParams`` -> human readable listing,
  * JSON round trip used by the MLlib-compatible persistence layout
    (DefaultParamsWriter/Reader equivalents live in persistence.py).
"""

from __future__ import annotations

import copy as _copy
import threading
import uuid
from typing import Any, Callable, Dict, Iterable, List, Optional


class Param:
    """A typed parameter with an owner, name, doc string and validator."""

    __slots__ = ("parent", "name", "doc", "validator", "type_converter")

    def __init__(
        self,
        parent: "Params",
        name: str,
        doc: str,
        validator: Optional[Callable[[Any], bool]] = None,
        type_converter: Optional[Callable[[Any], Any]] = None,
    ):
        self.parent = parent.uid if isinstance(parent, Params) else str(parent)
        self.name = name
        self.doc = doc
        self.validator = validator
        self.type_converter = type_converter

    def _validate(self, value: Any) -> Any:
        if self.type_converter is not None:
            value = self.type_converter(value)
        if self.validator is not None and not self.validator(value):
            raise ValueError(
                f"{self.parent} parameter {self.name} given invalid value {value!r}"
            )
        return value

    def __repr__(self):
        return f"{self.parent}__{self.name}"

    def __hash__(self):
        return hash((self.parent, self.name))

    def __eq__(self, other):
        return (
            isinstance(other, Param)
            and self.parent == other.parent
            and self.name == other.name
        )


class ParamValidators:
    """Validator factory, mirroring Spark ``ParamValidators``
    (used at e.g. reference HasSubBag.scala:44-49, GBMParams.scala:57-63)."""

    @staticmethod
    def gt(lower):
        return lambda v: v > lower

    @staticmethod
    def gtEq(lower):
        return lambda v: v >= lower

    @staticmethod
    def lt(upper):
        return lambda v: v < upper

    @staticmethod
    def ltEq(upper):
        return lambda v: v <= upper

    @staticmethod
    def inRange(lo, hi, lower_inclusive=True, upper_inclusive=True):
        def check(v):
            ok_lo = v >= lo if lower_inclusive else v > lo
            ok_hi = v <= hi if upper_inclusive else v < hi
            return ok_lo and ok_hi

        return check

    @staticmethod
    def inArray(allowed):
        allowed = list(allowed)
        return lambda v: v in allowed

    @staticmethod
    def always_true():
        return lambda v: True


def _to_float(v):
    return float(v)


def _to_int(v):
    if isinstance(v, float) and not v.is_integer():
        raise ValueError(f"expected integer, got {v}")
    return int(v)


def _to_bool(v):
    if not isinstance(v, bool):
        raise ValueError(f"expected bool, got {v!r}")
    return v


def _to_lower_str(v):
    return str(v).lower()


_uid_lock = threading.Lock()
_uid_counters: Dict[str, int] = {}


def _gen_uid(prefix: str) -> str:
    # Spark uses <prefix>_<12 hex chars>; keep the shape for metadata parity.
    with _uid_lock:
        return f"{prefix}_{uuid.uuid4().hex[:12]}"


class Params:
    """Base class carrying a param registry; analog of Spark ``Params``.

    Subclasses declare params in ``_declare_params`` (called once per
    instance) and may install class defaults with ``_setDefault``.
    """

    def __init__(self, uid: Optional[str] = None):
        self.uid = uid or _gen_uid(type(self).__name__)
        self._params: Dict[str, Param] = {}
        self._paramMap: Dict[str, Any] = {}  # user-set values
        self._defaultParamMap: Dict[str, Any] = {}  # class defaults
        self._declare_params()

    # -- declaration ------------------------------------------------------
    def _declare_params(self):  # pragma: no cover - overridden
        pass

    def _param(self, name, doc, validator=None, conv=None) -> Param:
        p = Param(self, name, doc, validator, conv)
        self._params[name] = p
        return p

    def _float_param(self, name, doc, validator=None):
        return self._param(name, doc, validator, _to_float)

    def _int_param(self, name, doc, validator=None):
        return self._param(name, doc, validator, _to_int)

    def _bool_param(self, name, doc):
        return self._param(name, doc, None, _to_bool)

    def _str_param(self, name, doc, validator=None, lower=True):
        return self._param(name, doc, validator, _to_lower_str if lower else str)

    def _setDefault(self, **kwargs):
        for k, v in kwargs.items():
            p = self._params[k]
            self._defaultParamMap[k] = p._validate(v) if v is not None else v

    # -- access -----------------------------------------------------------
    def hasParam(self, name: str) -> bool:
        return name in self._params

    def getParam(self, name: str) -> Param:
        return self._params[name]

    @property
    def params(self) -> List[Param]:
        return [self._params[k] for k in sorted(self._params)]

    def isSet(self, name: str) -> bool:
        return self._resolve(name) in self._paramMap

    def hasDefault(self, name: str) -> bool:
        return self._resolve(name) in self._defaultParamMap

    def isDefined(self, name: str) -> bool:
        return self.isSet(name) or self.hasDefault(name)

    def _resolve(self, param) -> str:
        name = param.name if isinstance(param, Param) else param
        if name not in self._params:
            raise AttributeError(f"{type(self).__name__} has no param {name!r}")
        return name

    def set(self, param, value) -> "Params":
        name = self._resolve(param)
        self._paramMap[name] = self._params[name]._validate(value)
        return self

    def clear(self, param) -> "Params":
        self._paramMap.pop(self._resolve(param), None)
        return self

    def get(self, param):
        return self._paramMap.get(self._resolve(param))

    def getOrDefault(self, param):
        name = self._resolve(param)
        if name in self._paramMap:
            return self._paramMap[name]
        if name in self._defaultParamMap:
            return self._defaultParamMap[name]
        raise KeyError(f"param {name} is not set and has no default")

    def getOrNone(self, param):
        name = self._resolve(param)
        if name in self._paramMap:
            return self._paramMap[name]
        return self._defaultParamMap.get(name)

    def explainParam(self, param) -> str:
        name = self._resolve(param)
        p = self._params[name]
        cur = self._paramMap.get(name, "undefined")
        dft = self._defaultParamMap.get(name, "undefined")
        return f"{p.name}: {p.doc} (default: {dft}, current: {cur})"

    def explainParams(self) -> str:
        return "\n".join(self.explainParam(n) for n in sorted(self._params))

    # -- cloning ----------------------------------------------------------
    def copy(self, extra: Optional[Dict] = None) -> "Params":
        """Deep-ish copy like Spark ``defaultCopy`` + ``copyValues``:
        a fresh instance of the same class with the same uid and param
        values; ``extra`` maps Param or name -> value overrides.  Nested
        estimator params (objects exposing ``.copy``) are copied too, which
        is what the reference's ``copy(extra)`` does when it clones base
        learners (BaggingRegressor.scala:111-115)."""
        that = type(self)()
        that.uid = self.uid
        that._defaultParamMap = dict(self._defaultParamMap)
        for k, v in self._paramMap.items():
            if hasattr(v, "copy") and isinstance(v, Params):
                v = v.copy()
            elif isinstance(v, (list, tuple)) and v and isinstance(v[0], Params):
                v = [x.copy() for x in v]
            that._paramMap[k] = v
        if extra:
            for k, v in extra.items():
                that.set(k, v)
        return that

    # -- persistence helpers ---------------------------------------------
    # Names of params that hold nested estimators and must be excluded from
    # flat JSON metadata (saved as sub-directories instead) — the analog of
    # the reference's saveImpl skipping baseLearner/baseLearners/stacker
    # (BaggingRegressor.scala:46-62).
    _NESTED_PARAM_NAMES = ("baseLearner", "baseLearners", "stacker")

    def _jsonParams(self) -> Dict[str, Any]:
        out = {}
        for k, v in self._paramMap.items():
            if k in self._NESTED_PARAM_NAMES:
                continue
            out[k] = v
        return out

    def _setFromJson(self, d: Dict[str, Any]):
        for k, v in d.items():
            if self.hasParam(k):
                self.set(k, v)
        return self


class ParamMap(dict):
    """A {param-name: value} override map (Spark ``ParamMap`` analog)."""

    pass
