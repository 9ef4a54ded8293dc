"""Arithmetic-expression evaluator for config values.

The reference keeps every option value as a string and evaluates it as an
arithmetic expression (``2 ** 30``, ``1405 + (64 / 2)``, ``128 * 1e6``) via a
vendored Boost.Spirit grammar (reference: userspace/include/srtb/program_options.hpp:148-201,
userspace/3rdparty/exprgrammar.hpp).  This is a from-scratch reimplementation of
that capability: a small recursive-descent parser supporting ``+ - * / %``,
``**``, unary ``+/-``, parentheses, and C/Python float literals (``1e6``,
``.5``, ``0x1f`` is NOT supported by the reference grammar and not here).

No ``eval()`` — config files are untrusted input.
"""

from __future__ import annotations


class ExprError(ValueError):
    pass


import math as _math
import sys as _sys

# case-insensitive symbol tables, mirroring the reference grammar
# (3rdparty/exprgrammar/include/suzerain/exprgrammar.hpp:118-178)
_CONSTANTS = {
    "digits": float(_sys.float_info.mant_dig),
    "digits10": float(_sys.float_info.dig),
    "e": _math.e,
    "epsilon": _sys.float_info.epsilon,
    "pi": _math.pi,
}

_UFUNCS = {
    "abs": abs, "acos": _math.acos, "asin": _math.asin, "atan": _math.atan,
    "ceil": _math.ceil, "cos": _math.cos, "cosh": _math.cosh,
    "exp": _math.exp, "floor": _math.floor, "log": _math.log,
    "log10": _math.log10, "sin": _math.sin, "sinh": _math.sinh,
    "sqrt": _math.sqrt, "tan": _math.tan, "tanh": _math.tanh,
}

_BFUNCS = {
    "atan2": _math.atan2, "max": max, "min": min, "pow": pow,
}


class _Parser:
    def __init__(self, text: str):
        self.text = text
        self.pos = 0

    def _skip_ws(self) -> None:
        while self.pos < len(self.text) and self.text[self.pos] in " \t":
            self.pos += 1

    def _peek(self) -> str:
        self._skip_ws()
        return self.text[self.pos] if self.pos < len(self.text) else ""

    def _expect(self, ch: str) -> None:
        if self._peek() != ch:
            raise ExprError(f"expected {ch!r} at pos {self.pos} in {self.text!r}")
        self.pos += 1

    # grammar:
    #   expr    := term (('+'|'-') term)*
    #   term    := power (('*'|'/'|'%') power)*
    #   power   := unary ('**' power)?          (right-associative)
    #   unary   := ('+'|'-')* atom
    #   atom    := number | '(' expr ')'
    def parse(self) -> float:
        v = self.expr()
        self._skip_ws()
        if self.pos != len(self.text):
            raise ExprError(f"trailing characters at pos {self.pos} in {self.text!r}")
        return v

    def expr(self) -> float:
        v = self.term()
        while True:
            c = self._peek()
            if c == "+":
                self.pos += 1
                v = v + self.term()
            elif c == "-":
                self.pos += 1
                v = v - self.term()
            else:
                return v

    def term(self) -> float:
        v = self.power()
        while True:
            c = self._peek()
            if c == "*":
                # careful: '**' belongs to power
                if self.text[self.pos : self.pos + 2] == "**":
                    return v
                self.pos += 1
                v = v * self.power()
            elif c == "/":
                self.pos += 1
                v = v / self.power()
            elif c == "%":
                self.pos += 1
                v = v % self.power()
            else:
                return v

    def power(self) -> float:
        v = self.unary()
        self._skip_ws()
        if self.text[self.pos : self.pos + 2] == "**":
            self.pos += 2
            return v ** self.power()
        return v

    def unary(self) -> float:
        sign = 1.0
        while True:
            c = self._peek()
            if c == "-":
                sign = -sign
                self.pos += 1
            elif c == "+":
                self.pos += 1
            else:
                break
        return sign * self.atom()

    def atom(self) -> float:
        c = self._peek()
        if c == "(":
            self.pos += 1
            v = self.expr()
            self._expect(")")
            return v
        if c.isalpha() or c == "_":
            return self.symbol()
        return self.number()

    def symbol(self) -> float:
        self._skip_ws()
        start = self.pos
        t = self.text
        while self.pos < len(t) and (t[self.pos].isalnum() or t[self.pos] == "_"):
            self.pos += 1
        name = t[start : self.pos].lower()
        if self._peek() == "(":
            self.pos += 1
            a = self.expr()
            if name in _UFUNCS:
                self._expect(")")
                return float(_UFUNCS[name](a))
            if name in _BFUNCS:
                self._expect(",")
                b = self.expr()
                self._expect(")")
                return float(_BFUNCS[name](a, b))
            raise ExprError(f"unknown function {name!r} in {t!r}")
        if name in _CONSTANTS:
            return _CONSTANTS[name]
        raise ExprError(f"unknown symbol {name!r} in {t!r}")

    def number(self) -> float:
        self._skip_ws()
        start = self.pos
        t = self.text
        n = len(t)
        i = self.pos
        while i < n and t[i].isdigit():
            i += 1
        if i < n and t[i] == ".":
            i += 1
            while i < n and t[i].isdigit():
                i += 1
        if i < n and t[i] in "eE":
            j = i + 1
            if j < n and t[j] in "+-":
                j += 1
            if j < n and t[j].isdigit():
                i = j
                while i < n and t[i].isdigit():
                    i += 1
        if i == start:
            raise ExprError(f"expected number at pos {start} in {t!r}")
        self.pos = i
        return float(t[start:i])


def evaluate(text: str) -> float:
    """Evaluate an arithmetic config expression to a float."""
    return _Parser(text.strip()).parse()


def evaluate_int(text: str) -> int:
    """Evaluate and round to nearest integer (config values like ``2 ** 30``)."""
    v = evaluate(text)
    r = round(v)
    if abs(v - r) > 1e-6 * max(1.0, abs(v)):
        raise ExprError(f"expected integer value, got {v} from {text!r}")
    return int(r)
