"""printf-style numeric format-specifier handling (reference: acg/fmtspec.{c,h}).

The reference parses/validates a C printf format specifier given via
``--numfmt`` and uses it when writing vector/matrix values
(used at hip/acg-hip.c:722-725).  Here we validate the same grammar
(flags, width, precision, length modifier, conversion) and provide a
formatter usable from Python.
"""

from __future__ import annotations

import re

# %[flags][width][.precision][length]conversion  — conversions valid for doubles
_FMT_RE = re.compile(
    r"^%"
    r"(?P<flags>[-+ #0']*)"
    r"(?P<width>\d+|\*)?"
    r"(?P<precision>\.(?:\d+|\*))?"
    r"(?P<length>l|ll|h|hh|L|j|z|t)?"
    r"(?P<conversion>[eEfFgGaA])$"
)


class FmtSpec:
    """A validated printf format specifier for floating-point output."""

    def __init__(self, spec: str = "%.17g"):
        m = _FMT_RE.match(spec)
        if m is None:
            raise ValueError(f"invalid format specifier for double output: {spec!r}")
        if "*" in spec:
            raise ValueError("'*' width/precision not supported in --numfmt")
        self.spec = spec
        # Python's % formatting accepts the same core grammar minus the
        # C length modifier and the ' (thousands) flag.
        py = spec.replace("'", "")
        length = m.group("length")
        if length:
            py = py.replace(length + m.group("conversion"), m.group("conversion"))
        # Python has no %a hex-float conversion via %-formatting; map to float.hex.
        self._hex = m.group("conversion") in "aA"
        self._py = py
        # honour the ' (thousands-grouping) flag like the reference fmtspec:
        # %-formatting cannot group, so build an equivalent format() spec
        # ({:[flags][width],[.prec][conv]}) and use it when ' is present
        self._groupfmt = None
        if "'" in m.group("flags") and not self._hex:
            flags = m.group("flags")
            conv = m.group("conversion")
            f = ""
            if "-" in flags:
                f += "<"
            if "+" in flags:
                f += "+"
            elif " " in flags:
                f += " "
            if "#" in flags:
                f += "#"
            if "0" in flags and "-" not in flags:
                f += "0"
            if m.group("width"):
                f += m.group("width")
            f += ","
            if m.group("precision"):
                f += m.group("precision")
            f += conv
            self._groupfmt = f

    def format(self, value: float) -> str:
        if self._hex:
            s = float(value).hex()
            return s.upper() if self.spec[-1] == "A" else s
        if self._groupfmt is not None:
            return format(float(value), self._groupfmt)
        return self._py % value

    def __call__(self, value: float) -> str:
        return self.format(value)

    def __repr__(self):
        return f"FmtSpec({self.spec!r})"


def parse_numfmt(spec: str) -> FmtSpec:
    """Parse and validate a ``--numfmt`` argument (fmtspec_parse analog)."""
    return FmtSpec(spec)
