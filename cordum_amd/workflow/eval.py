"""Workflow expression mini-language + template interpolation.

Oracle: core/workflow/eval.go:17-100 (literals, dot paths over {input, ctx,
steps, item}, length()/first(), comparisons, unary !) and engine.go:873-964
(evalTemplates/evalTemplateString: a lone `${expr}` keeps the value's type,
mixed strings stringify each expression; nil renders as empty).
"""
from __future__ import annotations

from typing import Any, Dict, List


class EvalError(Exception):
    pass


_OPS = ["==", "!=", ">=", "<=", ">", "<"]


def eval_expr(expr: str, ctx: Dict[str, Any]) -> Any:
    expr = expr.strip()
    if not expr:
        raise EvalError("empty expression")

    if expr.startswith("!"):
        return not truthy(eval_expr(expr[1:], ctx))

    for op in _OPS:
        idx = expr.find(op)
        if idx >= 0:
            left = eval_expr(expr[:idx], ctx)
            right = eval_expr(expr[idx + len(op):], ctx)
            return _compare(left, right, op)

    if expr.startswith("length(") and expr.endswith(")"):
        val = eval_expr(expr[len("length("):-1], ctx)
        if isinstance(val, (list, str, dict)):
            return len(val)
        return 0
    if expr.startswith("first(") and expr.endswith(")"):
        val = eval_expr(expr[len("first("):-1], ctx)
        if isinstance(val, list) and val:
            return val[0]
        return None

    if len(expr) >= 2 and expr[0] == "'" and expr[-1] == "'":
        return expr.strip("'")
    if len(expr) >= 2 and expr[0] == '"' and expr[-1] == '"':
        return expr.strip('"')
    if expr == "true":
        return True
    if expr == "false":
        return False
    try:
        return float(expr)
    except ValueError:
        pass
    return resolve_path(expr, ctx)


def resolve_path(path: str, ctx: Dict[str, Any]) -> Any:
    cur: Any = ctx
    for part in path.split("."):
        if not isinstance(cur, dict):
            return None
        cur = cur.get(part)
    return cur


def _to_float(v: Any) -> float:
    if isinstance(v, bool):
        return 0.0
    if isinstance(v, (int, float)):
        return float(v)
    if isinstance(v, str):
        try:
            return float(v)
        except ValueError:
            return 0.0
    return 0.0


def _compare(a: Any, b: Any, op: str) -> bool:
    if isinstance(a, (int, float)) and not isinstance(a, bool):
        return _cmp(float(a), _to_float(b), op)
    if isinstance(a, str) and isinstance(b, str):
        return _cmp(a, b, op)
    # fallback equality on stringified values (Go fmt.Sprint)
    if op == "==":
        return _sprint(a) == _sprint(b)
    if op == "!=":
        return _sprint(a) != _sprint(b)
    return False


def _sprint(v: Any) -> str:
    if isinstance(v, bool):
        return "true" if v else "false"
    if v is None:
        return "<nil>"
    if isinstance(v, float) and v.is_integer():
        return str(int(v))
    return str(v)


def _cmp(a, b, op: str) -> bool:
    return {
        "==": a == b,
        "!=": a != b,
        ">": a > b,
        "<": a < b,
        ">=": a >= b,
        "<=": a <= b,
    }[op]


def truthy(v: Any) -> bool:
    if v is None:
        return False
    if isinstance(v, bool):
        return v
    if isinstance(v, str):
        return v != ""
    if isinstance(v, (int, float)):
        return v != 0
    return True


def eval_condition(expr: str, scope: Dict[str, Any]) -> bool:
    return truthy(eval_expr(expr, scope))


# -- templates ---------------------------------------------------------------


def eval_templates(value: Any, scope: Dict[str, Any]) -> Any:
    if value is None:
        return None
    if isinstance(value, str):
        return eval_template_string(value, scope)
    if isinstance(value, dict):
        return {k: eval_templates(v, scope) for k, v in value.items()}
    if isinstance(value, list):
        return [eval_templates(v, scope) for v in value]
    return value


def eval_template_string(s: str, scope: Dict[str, Any]) -> Any:
    if "${" not in s:
        return s
    trimmed = s.strip()
    if (
        trimmed.startswith("${")
        and trimmed.endswith("}")
        and trimmed.count("${") == 1
        and trimmed.count("}") == 1
    ):
        return eval_expr(trimmed[2:-1].strip(), scope)
    out = []
    rest = s
    while True:
        start = rest.find("${")
        if start == -1:
            out.append(rest)
            break
        out.append(rest[:start])
        rest = rest[start + 2:]
        end = rest.find("}")
        if end == -1:
            raise EvalError("unterminated template expression")
        expr = rest[:end].strip()
        rest = rest[end + 1:]
        val = eval_expr(expr, scope)
        if val is not None:
            out.append(_sprint(val))
    return "".join(out)


def eval_for_each(expr: str, scope: Dict[str, Any]) -> List[Any]:
    expr = expr.strip()
    if expr.startswith("${") and expr.endswith("}"):
        # accept the template-wrapped form used in workflow YAML
        val = eval_template_string(expr, scope)
    else:
        val = eval_expr(expr, scope)
    if val is None:
        return []
    if isinstance(val, list):
        return val
    raise EvalError(f"for_each expression must return array, got {type(val).__name__}")
