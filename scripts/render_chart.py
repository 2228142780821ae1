#!/usr/bin/env python3
"""Minimal Helm-template renderer for deploy/chart/wva-amd.

There is no `helm` binary in the build/CI image, so the chart-render
test (tests/test_chart_render.py) and `make chart-render` use this
renderer instead. It implements exactly the template subset the chart
uses — dotted .Values paths, if/else/end, range over lists, and the
quote/indent/default filters — and REJECTS anything else, which keeps
the chart inside a dialect that real Helm also renders (the constructs
are a strict subset of Go template + sprig).

Usage:
  python scripts/render_chart.py [--chart deploy/chart/wva-amd] \
      [--set key.path=value ...]
prints all rendered documents separated by `---`.
"""
from __future__ import annotations

import argparse
import os
import re
import sys
from typing import Any, List, Optional, Tuple

TAG_RE = re.compile(r"\{\{(-?)\s*(.*?)\s*(-?)\}\}")


class TemplateError(ValueError):
    pass


# --- value resolution -------------------------------------------------------

def resolve(path: str, values: Any, dot: Any, scope: Any = None) -> Any:
    """Resolve `.Values.a.b`, `.field` (range scope) or `$var` to a value."""
    if path.startswith("$"):
        name, _, rest = path[1:].partition(".")
        cur = (scope or {}).get(name)
        for p in [x for x in rest.split(".") if x]:
            if isinstance(cur, dict):
                cur = cur.get(p)
            else:
                return None
        return cur
    if not path.startswith("."):
        raise TemplateError(f"unsupported expression: {path!r}")
    parts = [p for p in path[1:].split(".") if p]
    if parts and parts[0] == "Values":
        cur: Any = values
        parts = parts[1:]
    else:
        cur = dot
    for p in parts:
        if isinstance(cur, dict):
            cur = cur.get(p)
        else:
            return None
        if cur is None:
            return None
    return cur


def apply_filters(value: Any, filters: List[str], values: Any, dot: Any,
                  scope: Any = None) -> Any:
    for f in filters:
        f = f.strip()
        if m := re.fullmatch(r"nindent\s+(\d+)", f):
            pad = " " * int(m.group(1))
            text = str("" if value is None else value)
            value = "\n" + "\n".join(
                pad + line if line else line for line in text.splitlines()
            )
        elif f == "quote":
            value = '"' + str("" if value is None else value).replace('"', '\\"') + '"'
        elif m := re.fullmatch(r"indent\s+(\d+)", f):
            pad = " " * int(m.group(1))
            text = str("" if value is None else value)
            value = "\n".join(
                pad + line if line else line for line in text.splitlines()
            )
        elif m := re.fullmatch(r"default\s+(.+)", f):
            if value in (None, "", [], {}):
                value = eval_atom(m.group(1), values, dot, scope)
        else:
            raise TemplateError(f"unsupported filter: {f!r}")
    return value


def eval_atom(expr: str, values: Any, dot: Any, scope: Any = None) -> Any:
    expr = expr.strip()
    if expr.startswith('"') and expr.endswith('"'):
        return expr[1:-1]
    if re.fullmatch(r"-?\d+", expr):
        return int(expr)
    if expr.startswith("toYaml "):
        import yaml

        val = eval_atom(expr[len("toYaml "):], values, dot, scope)
        return yaml.safe_dump(val, default_flow_style=False).rstrip("\n")
    return resolve(expr, values, dot, scope)


def eval_expr(expr: str, values: Any, dot: Any, scope: Any = None) -> Any:
    parts = expr.split("|")
    value = eval_atom(parts[0], values, dot, scope)
    return apply_filters(value, parts[1:], values, dot, scope)


# --- parsing ----------------------------------------------------------------

class Node:
    pass


class Text(Node):
    def __init__(self, s: str):
        self.s = s


class Expr(Node):
    def __init__(self, expr: str):
        self.expr = expr


class If(Node):
    def __init__(self, cond: str):
        self.cond = cond
        self.body: List[Node] = []
        self.orelse: List[Node] = []


class Range(Node):
    def __init__(self, expr: str):
        # `range .list`  OR  `range $k, $v := .map`
        self.vars: List[str] = []
        if ":=" in expr:
            head, _, expr = expr.partition(":=")
            self.vars = [v.strip().lstrip("$") for v in head.split(",")]
        self.expr = expr.strip()
        self.body: List[Node] = []


def tokenize(src: str) -> List[Tuple[str, str]]:
    """→ [(kind, payload)]: kind in text|tag."""
    out: List[Tuple[str, str]] = []
    pos = 0
    for m in TAG_RE.finditer(src):
        text = src[pos:m.start()]
        if m.group(1) == "-":  # left trim: strip trailing ws + newline
            text = re.sub(r"\n?[ \t]*$", "", text)
        out.append(("text", text))
        out.append(("tag", m.group(2)))
        pos = m.end()
        if m.group(3) == "-":  # right trim
            rest = src[pos:]
            stripped = re.sub(r"^[ \t]*\n?", "", rest)
            pos = len(src) - len(stripped)
    out.append(("text", src[pos:]))
    return out


def parse(tokens: List[Tuple[str, str]]) -> List[Node]:
    root: List[Node] = []
    stack: List[Node] = []

    def bucket() -> List[Node]:
        if not stack:
            return root
        top = stack[-1]
        if isinstance(top, If):
            return top.orelse if getattr(top, "_in_else", False) else top.body
        assert isinstance(top, Range)
        return top.body

    for kind, payload in tokens:
        if kind == "text":
            if payload:
                bucket().append(Text(payload))
            continue
        if payload.startswith("if "):
            node = If(payload[3:].strip())
            bucket().append(node)
            stack.append(node)
        elif payload == "else":
            if not stack or not isinstance(stack[-1], If):
                raise TemplateError("else outside if")
            stack[-1]._in_else = True  # type: ignore[attr-defined]
        elif payload.startswith("range "):
            node = Range(payload[6:].strip())
            bucket().append(node)
            stack.append(node)
        elif payload == "end":
            if not stack:
                raise TemplateError("unbalanced end")
            stack.pop()
        elif payload.startswith("#") or payload.startswith("/*"):
            continue  # comment
        else:
            bucket().append(Expr(payload))
    if stack:
        raise TemplateError("unclosed if/range")
    return root


def render_nodes(nodes: List[Node], values: Any, dot: Any,
                 scope: Any = None) -> str:
    out: List[str] = []
    for node in nodes:
        if isinstance(node, Text):
            out.append(node.s)
        elif isinstance(node, Expr):
            val = eval_expr(node.expr, values, dot, scope)
            out.append(str("" if val is None else val))
        elif isinstance(node, If):
            cond = eval_expr(node.cond, values, dot, scope)
            branch = node.body if cond not in (
                None, False, "", 0, [], {}, "false"
            ) else node.orelse
            out.append(render_nodes(branch, values, dot, scope))
        elif isinstance(node, Range):
            items = eval_expr(node.expr, values, dot, scope) or []
            if isinstance(items, dict):
                if len(node.vars) != 2:
                    raise TemplateError(
                        f"range over map needs $k, $v :=: {node.expr!r}"
                    )
                for key in sorted(items):
                    sub = dict(scope or {})
                    sub[node.vars[0]] = key
                    sub[node.vars[1]] = items[key]
                    out.append(render_nodes(node.body, values, dot, sub))
            elif isinstance(items, list):
                for item in items:
                    out.append(render_nodes(node.body, values, item, scope))
            else:
                raise TemplateError(f"range over non-iterable: {node.expr!r}")
    return "".join(out)


def render_template(src: str, values: Any) -> str:
    return render_nodes(parse(tokenize(src)), values, None)


# --- chart driver -----------------------------------------------------------

def deep_set(d: dict, dotted: str, value: Any) -> None:
    parts = dotted.split(".")
    for p in parts[:-1]:
        d = d.setdefault(p, {})
    d[parts[-1]] = value


def render_chart(chart_dir: str, overrides: Optional[dict] = None) -> List[dict]:
    """Render every template with values.yaml (+overrides); returns the
    parsed YAML documents (empty docs dropped)."""
    import yaml

    with open(os.path.join(chart_dir, "values.yaml")) as f:
        values = yaml.safe_load(f) or {}
    for key, val in (overrides or {}).items():
        deep_set(values, key, val)

    docs: List[dict] = []
    tdir = os.path.join(chart_dir, "templates")
    for name in sorted(os.listdir(tdir)):
        if not name.endswith((".yaml", ".yml")):
            continue
        with open(os.path.join(tdir, name)) as f:
            rendered = render_template(f.read(), values)
        try:
            for doc in yaml.safe_load_all(rendered):
                if doc:
                    docs.append(doc)
        except yaml.YAMLError as e:
            # a value that breaks the document structure (stray quote,
            # newline) must fail with the renderer's own error type, so
            # callers distinguish "bad values" from a renderer crash
            raise TemplateError(
                f"{name}: rendered output is not valid YAML: {e}"
            ) from e
    return docs


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--chart", default="deploy/chart/wva-amd")
    ap.add_argument("--set", action="append", default=[],
                    help="key.path=value override")
    args = ap.parse_args()
    overrides = {}
    for s in args.set:
        k, _, v = s.partition("=")
        if v in ("true", "false"):
            v = v == "true"
        overrides[k] = v
    import yaml

    docs = render_chart(args.chart, overrides)
    print(yaml.safe_dump_all(docs, sort_keys=False))


if __name__ == "__main__":
    sys.exit(main())
