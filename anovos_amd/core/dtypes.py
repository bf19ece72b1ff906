"""Type system of the engine.

Mirrors the reference's dtype segregation rules (shared/utils.py:48-91 in
the reference): Spark dtype strings map onto three attribute kinds —

- ``string``                                      -> categorical
- ``double/float/int/bigint/long/smallint/decimal`` -> numerical
- everything else (timestamp, date, array, ...)   -> other

Storage in the column store (MI355X-first choices):

- numerical   : float32 or float64 torch tensor, NaN == null (branch-free
                null handling in every kernel; integers are widened so a
                single kernel family covers all numeric columns)
- categorical : int32 dictionary codes + a host-side dictionary of strings,
                code -1 == null (regex/string work runs over the small
                dictionary, never the rows)
- timestamp   : int64 microseconds since epoch, NULL_TS == null
"""

from __future__ import annotations

NULL_CODE = -1  # categorical null code
NULL_TS = -(2**63)  # timestamp null sentinel (int64 min)

# Spark-style dtype names considered numerical (reference shared/utils.py:64-73)
_NUMERIC_DTYPES = {
    "double",
    "float",
    "int",
    "integer",
    "bigint",
    "long",
    "smallint",
    "short",
    "tinyint",
    "byte",
    "decimal",
}

_TS_DTYPES = {"timestamp", "date", "datetime"}


def kind_of_dtype(dtype: str) -> str:
    """Return the attribute kind ('numerical'|'categorical'|'other') for a
    Spark-style dtype string. Decimal types appear as 'decimal(p,s)'."""
    d = dtype.lower()
    if d == "string":
        return "categorical"
    if d in _NUMERIC_DTYPES or d.startswith("decimal"):
        return "numerical"
    return "other"


def is_timestamp_dtype(dtype: str) -> bool:
    return dtype.lower() in _TS_DTYPES
