"""Kubernetes resource.Quantity arithmetic (apimachinery analog, pure python).

Supports the suffixes the operator actually meets in pod resource lists:
binary (Ki..Ei), decimal (k..E), milli (m) and plain integers/decimals.
"""
from __future__ import annotations

from decimal import Decimal
from typing import Union

_BINARY = {"Ki": 1024, "Mi": 1024**2, "Gi": 1024**3, "Ti": 1024**4,
           "Pi": 1024**5, "Ei": 1024**6}
_DECIMAL = {"n": Decimal("1e-9"), "u": Decimal("1e-6"), "m": Decimal("0.001"),
            "k": 1000, "M": 1000**2, "G": 1000**3, "T": 1000**4,
            "P": 1000**5, "E": 1000**6}


def parse_quantity(value: Union[str, int, float, None]) -> Decimal:
    """Parse a K8s quantity string into a Decimal number of base units."""
    if value is None:
        return Decimal(0)
    if isinstance(value, (int, float)):
        return Decimal(str(value))
    s = str(value).strip()
    if not s:
        return Decimal(0)
    for suffix, mult in _BINARY.items():
        if s.endswith(suffix):
            return Decimal(s[: -len(suffix)]) * mult
    # exponent form like 1e3
    if ("e" in s or "E" in s) and not s[-1].isalpha():
        return Decimal(s)
    for suffix, mult in _DECIMAL.items():
        if s.endswith(suffix):
            return Decimal(s[: -len(suffix)]) * Decimal(mult)
    return Decimal(s)


def format_quantity(value: Decimal) -> str:
    """Render a Decimal back into a compact canonical quantity string."""
    if value == value.to_integral_value():
        ivalue = int(value)
        # prefer binary suffixes for byte-ish large values
        for suffix in ("Ei", "Pi", "Ti", "Gi", "Mi", "Ki"):
            mult = _BINARY[suffix]
            if ivalue and ivalue % mult == 0:
                return f"{ivalue // mult}{suffix}"
        return str(ivalue)
    milli = value * 1000
    if milli == milli.to_integral_value():
        return f"{int(milli)}m"
    return str(value)


def add_quantities(a: Union[str, int, None], b: Union[str, int, None]) -> str:
    return format_quantity(parse_quantity(a) + parse_quantity(b))
