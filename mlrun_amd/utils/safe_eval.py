# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Restricted AST expression evaluator for user-supplied condition /
url / body expressions.

The reference evaluates these strings with ``eval(expr,
{"__builtins__": {}})`` (e.g. hyper-param ``stop_condition``,
notification conditions, serving remote url/body expressions), which is
escapable via attribute traversal (``().__class__...``).  This module
walks the AST instead and only permits comparisons, boolean/arithmetic
ops, literals, name lookups from the provided mapping, non-dunder
attribute access / subscripts, f-strings, and calls to a small
builtin whitelist or non-dunder methods — so there is no route to
``__class__``/``__globals__``.
"""

import ast
import typing

_ALLOWED_BUILTINS: typing.Dict[str, typing.Any] = {
    "str": str, "int": int, "float": float, "bool": bool, "len": len,
    "min": min, "max": max, "abs": abs, "round": round, "sum": sum,
    "sorted": sorted, "any": any, "all": all,
}

_BOOL_OPS = {ast.And: all, ast.Or: any}

_BIN_OPS = {
    ast.Add: lambda a, b: a + b,
    ast.Sub: lambda a, b: a - b,
    ast.Mult: lambda a, b: a * b,
    ast.Div: lambda a, b: a / b,
    ast.FloorDiv: lambda a, b: a // b,
    ast.Mod: lambda a, b: a % b,
    ast.Pow: lambda a, b: a ** b,
}

_CMP_OPS = {
    ast.Eq: lambda a, b: a == b,
    ast.NotEq: lambda a, b: a != b,
    ast.Lt: lambda a, b: a < b,
    ast.LtE: lambda a, b: a <= b,
    ast.Gt: lambda a, b: a > b,
    ast.GtE: lambda a, b: a >= b,
    ast.In: lambda a, b: a in b,
    ast.NotIn: lambda a, b: a not in b,
    ast.Is: lambda a, b: a is b,
    ast.IsNot: lambda a, b: a is not b,
}


class UnsafeExpressionError(ValueError):
    pass


class _Evaluator:
    def __init__(self, names: typing.Mapping[str, typing.Any]):
        self.names = names

    def visit(self, node):
        method = getattr(self, f"_visit_{type(node).__name__}", None)
        if method is None:
            raise UnsafeExpressionError(
                f"expression element {type(node).__name__} not allowed")
        return method(node)

    def _visit_Expression(self, node):
        return self.visit(node.body)

    def _visit_Constant(self, node):
        return node.value

    def _visit_Name(self, node):
        if node.id in self.names:
            return self.names[node.id]
        if node.id in _ALLOWED_BUILTINS:
            return _ALLOWED_BUILTINS[node.id]
        raise UnsafeExpressionError(f"unknown name {node.id!r}")

    def _visit_Attribute(self, node):
        if node.attr.startswith("_"):
            raise UnsafeExpressionError(
                f"attribute {node.attr!r} not allowed")
        return getattr(self.visit(node.value), node.attr)

    def _visit_Subscript(self, node):
        return self.visit(node.value)[self.visit(node.slice)]

    def _visit_Index(self, node):  # py<3.9 compat node
        return self.visit(node.value)

    def _visit_Slice(self, node):
        return slice(
            self.visit(node.lower) if node.lower else None,
            self.visit(node.upper) if node.upper else None,
            self.visit(node.step) if node.step else None)

    def _visit_BoolOp(self, node):
        op = _BOOL_OPS[type(node.op)]
        return op(bool(self.visit(v)) for v in node.values)

    def _visit_UnaryOp(self, node):
        value = self.visit(node.operand)
        if isinstance(node.op, ast.Not):
            return not value
        if isinstance(node.op, ast.USub):
            return -value
        if isinstance(node.op, ast.UAdd):
            return +value
        raise UnsafeExpressionError("unary operator not allowed")

    def _visit_BinOp(self, node):
        op = _BIN_OPS.get(type(node.op))
        if op is None:
            raise UnsafeExpressionError("binary operator not allowed")
        return op(self.visit(node.left), self.visit(node.right))

    def _visit_Compare(self, node):
        left = self.visit(node.left)
        for op, comparator in zip(node.ops, node.comparators):
            fn = _CMP_OPS.get(type(op))
            if fn is None:
                raise UnsafeExpressionError("comparison not allowed")
            right = self.visit(comparator)
            if not fn(left, right):
                return False
            left = right
        return True

    def _visit_Call(self, node):
        if node.keywords and any(k.arg is None for k in node.keywords):
            raise UnsafeExpressionError("** call expansion not allowed")
        func = node.func
        if isinstance(func, ast.Attribute):
            if func.attr.startswith("_"):
                raise UnsafeExpressionError(
                    f"method {func.attr!r} not allowed")
            target = getattr(self.visit(func.value), func.attr)
        elif isinstance(func, ast.Name) and func.id in _ALLOWED_BUILTINS:
            target = _ALLOWED_BUILTINS[func.id]
        else:
            raise UnsafeExpressionError("call target not allowed")
        args = [self.visit(a) for a in node.args]
        kwargs = {k.arg: self.visit(k.value) for k in node.keywords}
        return target(*args, **kwargs)

    def _visit_IfExp(self, node):
        return (self.visit(node.body) if self.visit(node.test)
                else self.visit(node.orelse))

    def _visit_Tuple(self, node):
        return tuple(self.visit(e) for e in node.elts)

    def _visit_List(self, node):
        return [self.visit(e) for e in node.elts]

    def _visit_Dict(self, node):
        return {self.visit(k): self.visit(v)
                for k, v in zip(node.keys, node.values)}

    def _visit_Set(self, node):
        return {self.visit(e) for e in node.elts}

    def _visit_JoinedStr(self, node):
        return "".join(self.visit(v) for v in node.values)

    def _visit_FormattedValue(self, node):
        value = self.visit(node.value)
        spec = self.visit(node.format_spec) if node.format_spec else ""
        if node.conversion == 114:  # !r
            value = repr(value)
        elif node.conversion == 115:  # !s
            value = str(value)
        return format(value, spec)


def safe_eval(expression: str,
              names: typing.Mapping[str, typing.Any]) -> typing.Any:
    """Evaluate a restricted expression over the given name mapping."""
    try:
        tree = ast.parse(expression, mode="eval")
    except SyntaxError as exc:
        raise UnsafeExpressionError(
            f"invalid expression {expression!r}: {exc}") from exc
    return _Evaluator(names).visit(tree)
