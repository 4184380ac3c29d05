"""Hand-computed fixtures for the testinspect static metrics.

The original study computed features 12-14 (Halstead Volume, Cyclomatic
Complexity, Maintainability) via radon 5.1.0, which is not installable
here.  Expected values below were derived BY HAND from radon's documented
rules (see the rule summary in collect/testinspect.py) on functions whose
operator/branch structure is unambiguous — these tests fail if the
implementation drifts from the documented formulas.
"""

import ast
import math

import pytest

from flake16_framework_amd.collect.testinspect import (
    _ast_depth, _cyclomatic, _halstead_volume, static_metrics,
)

SNIPPET_A = '''
def f(a, b):
    c = a + b
    if a and b or c:
        return c - 1
    return 0
'''

# Hand derivation for SNIPPET_A:
#   CC: base 1 + if 1 + BoolOp(or) 1 + BoolOp(and) 1            = 4
#   Halstead operators: Add, Or, And, Sub        -> N1=4, n1=4
#   operands: a,b (from +), c (from or; the nested BoolOp is not
#   a leaf operand), a,b (from and), c,1 (from -) -> N2=7,
#   distinct {a,b,c,'1'} -> n2=4
#   Volume = (4+7) * log2(8) = 33.0
#   LOC = 5 (def line .. return 0)
#   MI = (171 - 5.2 ln 33 - 0.23*4 - 16.2 ln 5) * 100/171

SNIPPET_B = '''
def g(xs):
    with open("x") as fd:
        pass
    total = 0
    for x in xs:
        total += x
    else:
        assert total >= 0
    ys = [x * 2 for x in xs if x > 1 if x < 9]
    while total:
        total -= 1
    return [y for y in ys]
'''

# Hand derivation for SNIPPET_B:
#   CC: base 1 + with 1 + for 1 + for-else 1 + assert 1
#       + comp1 (1 generator + 2 ifs) 3 + while 1 + comp2 1     = 10
#   Halstead operators: Add (+=), GtE, Mult, Gt, Lt, Sub (-=)
#       -> N1=6, n1=6
#   operands: total,x / total,0 / x,2 / x,1 / x,9 / total,1
#       -> N2=12, distinct {total,x,'0','2','1','9'} -> n2=6
#   Volume = 18 * log2(12)

SNIPPET_TRY = '''
def h(x):
    try:
        x = 1 / x
    except ZeroDivisionError:
        x = 0
    except ValueError:
        x = -1
    else:
        x = 2
    return x
'''
# CC: base 1 + 2 handlers + try-else 1 = 4
# (the unary minus on -1 folds into the Constant in CPython's parser
# only for literals inside UnaryOp? no: -1 parses as UnaryOp(USub,
# Constant(1)) -> one UnaryOp operator)


def _fn(src):
    return ast.parse(src).body[0]


class TestCyclomatic:
    def test_snippet_a(self):
        assert _cyclomatic(_fn(SNIPPET_A)) == 4

    def test_snippet_b(self):
        assert _cyclomatic(_fn(SNIPPET_B)) == 10

    def test_try_handlers_and_else(self):
        assert _cyclomatic(_fn(SNIPPET_TRY)) == 4

    def test_plain_function_is_one(self):
        assert _cyclomatic(_fn("def p():\n    return 1\n")) == 1


class TestHalstead:
    def test_snippet_a_volume(self):
        assert _halstead_volume(_fn(SNIPPET_A)) == pytest.approx(33.0)

    def test_snippet_b_volume(self):
        assert _halstead_volume(_fn(SNIPPET_B)) == \
            pytest.approx(18 * math.log2(12))

    def test_no_operators_zero(self):
        assert _halstead_volume(_fn("def p():\n    return foo(1)\n")) == 0.0


class TestMaintainability:
    def test_snippet_a_mi(self):
        # static_metrics -> (depth, assertions, ext_modules, hv, cc, loc, mi)
        m = static_metrics(_fn(SNIPPET_A), set())
        hv, cc, loc, mi = m[3], m[4], m[5], m[6]
        assert (hv, cc, loc) == (pytest.approx(33.0), 4, 5)
        expect = (171 - 5.2 * math.log(33.0) - 0.23 * 4
                  - 16.2 * math.log(5)) * 100 / 171
        assert mi == pytest.approx(expect)

    def test_clamped_to_0_100(self):
        m = static_metrics(_fn("def p():\n    return 1\n"), set())
        assert 0.0 <= m[6] <= 100.0


class TestOtherStatics:
    def test_depth_and_assertions(self):
        fn = _fn(SNIPPET_A)
        assert _ast_depth(fn) >= 3
        fn_b = _fn(SNIPPET_B)
        m = static_metrics(fn_b, {"open"})
        assert m[1] == 1          # one assert
        assert m[2] == 1          # 'open' used and importable
