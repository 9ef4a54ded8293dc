import math

import pytest

from srtb_amd.utils.expr import ExprError, evaluate, evaluate_int


def test_reference_config_expressions():
    # expressions appearing verbatim in the reference's shipped config files
    assert evaluate("2 ** 30") == 2**30
    assert evaluate("2 ** 11") == 2**11
    assert evaluate("1405 + (64 / 2)") == 1437.0
    assert evaluate("128 * 1e6") == 128e6
    assert evaluate("1000 * 1e6") == 1e9
    assert evaluate("-478.80") == -478.80
    assert evaluate("-64") == -64.0


def test_precedence_and_associativity():
    assert evaluate("2 + 3 * 4") == 14
    assert evaluate("(2 + 3) * 4") == 20
    assert evaluate("2 ** 3 ** 2") == 512  # right-assoc
    # reference grammar: unary minus is part of primary, so (-2)**2
    # (exprgrammar.hpp:213-223: factor = primary ('**' factor)*)
    assert evaluate("-2 ** 2") == 4
    assert evaluate("2 ** -2") == 0.25
    assert evaluate("10 - 4 - 3") == 3  # left-assoc
    assert evaluate("7 % 4") == 3
    assert evaluate("1/2/2") == 0.25


def test_floats_and_scientific():
    assert evaluate("1e6") == 1e6
    assert evaluate("1.5e-3") == 1.5e-3
    assert evaluate(".5") == 0.5
    assert math.isclose(evaluate("3.14159"), 3.14159)


def test_functions_and_constants():
    # reference grammar symbol tables (exprgrammar.hpp:118-178)
    assert math.isclose(evaluate("pi"), math.pi)
    assert math.isclose(evaluate("PI"), math.pi)  # case-insensitive
    assert math.isclose(evaluate("e"), math.e)
    assert math.isclose(evaluate("sqrt(2)"), math.sqrt(2))
    assert math.isclose(evaluate("cos(0)"), 1.0)
    assert math.isclose(evaluate("pow(2, 10)"), 1024)
    assert math.isclose(evaluate("max(3, 5)"), 5)
    assert math.isclose(evaluate("min(3, 5)"), 3)
    assert math.isclose(evaluate("atan2(1, 1)"), math.pi / 4)
    assert math.isclose(evaluate("abs(-3) + floor(2.7)"), 5)
    with pytest.raises(ExprError):
        evaluate("nosuchfn(1)")


def test_evaluate_int():
    assert evaluate_int("2 ** 30") == 1073741824
    assert evaluate_int(" 16 ") == 16
    with pytest.raises(ExprError):
        evaluate_int("1.5")


def test_errors():
    with pytest.raises(ExprError):
        evaluate("2 +")
    with pytest.raises(ExprError):
        evaluate("(1")
    with pytest.raises(ExprError):
        evaluate("foo")
    with pytest.raises(ExprError):
        evaluate("1 2")
