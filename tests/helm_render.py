"""Minimal Helm-template renderer for chart CI (no helm binary in the
image). Supports exactly the constructs our templates use:

  {{ .Values.a.b }}  {{ .Release.Name }}  {{ .Release.Namespace }}
  {{ .Chart.Name }}  {{ <ref> | indent N }}  {{ toYaml . | indent N }}
  {{- if <ref> }} ... {{- end }}    {{- with <ref> }} ... {{- end }}

Anything else raises so template drift is caught, not silently skipped.
"""

from __future__ import annotations

import re


class Ctx:
    def __init__(self, values, release_name="k3samd", namespace="k3samd",
                 chart_name="k3samd-device-plugin"):
        self.root = {
            "Values": values,
            "Release": {"Name": release_name, "Namespace": namespace},
            "Chart": {"Name": chart_name},
        }
        self.dot = self.root


def lookup(ctx: Ctx, ref: str):
    ref = ref.strip()
    if ref == ".":
        return ctx.dot
    assert ref.startswith("."), ref
    cur = ctx.root
    for part in ref[1:].split("."):
        if not part:
            continue
        if not isinstance(cur, dict) or part not in cur:
            return None
        cur = cur[part]
    return cur


def to_yaml(value, level=0) -> str:
    import yaml
    return yaml.safe_dump(value, default_flow_style=False).rstrip("\n")


def indent(text: str, n: int) -> str:
    pad = " " * n
    return "\n".join(pad + line if line else line
                     for line in str(text).splitlines())


TAG = re.compile(r"\{\{-?\s*(.*?)\s*-?\}\}")


def render(template: str, ctx: Ctx) -> str:
    # normalize "{{-" whitespace chomping: remove preceding newline+spaces
    template = re.sub(r"\n[ \t]*\{\{-", "\n{{-CHOMP}}{{", template)
    template = template.replace("{{-CHOMP}}{{", "\x00{{")

    out = []
    pos = 0
    stack = []  # (kind, emit_before, saved_dot)

    def emitting():
        return all(e for _, e, _ in stack)

    for m in TAG.finditer(template):
        if emitting():
            chunk = template[pos:m.start()]
            out.append(chunk.replace("\x00\n", "").replace("\x00", ""))
        pos = m.end()
        expr = m.group(1)
        if expr.startswith("if "):
            cond = expr[3:].split()
            if cond and cond[0] in ("and", "or"):
                vals = [bool(lookup(ctx, c)) for c in cond[1:]]
                val = all(vals) if cond[0] == "and" else any(vals)
            else:
                val = bool(lookup(ctx, expr[3:]))
            stack.append(("if", bool(val), ctx.dot))
        elif expr.startswith("with "):
            val = lookup(ctx, expr[5:])
            stack.append(("with", bool(val), ctx.dot))
            if val:
                ctx.dot = val
        elif expr == "end":
            kind, _, saved = stack.pop()
            ctx.dot = saved
        elif not emitting():
            continue
        else:
            out.append(eval_expr(expr, ctx))
    if emitting():
        out.append(template[pos:].replace("\x00\n", "").replace("\x00", ""))
    rendered = "".join(out)
    # drop chomp markers on skipped branches and collapse blank-only lines
    rendered = rendered.replace("\x00", "")
    return rendered


def eval_expr(expr: str, ctx: Ctx) -> str:
    parts = [p.strip() for p in expr.split("|")]
    head = parts[0]
    if head == "toYaml .":
        value = to_yaml(ctx.dot)
    elif head.startswith("toYaml "):
        value = to_yaml(lookup(ctx, head[7:]))
    else:
        v = lookup(ctx, head)
        if v is None:
            raise KeyError(f"unresolved template reference: {head}")
        value = v if isinstance(v, str) else (
            "true" if v is True else "false" if v is False else str(v))
    for pipe in parts[1:]:
        mm = re.fullmatch(r"indent (\d+)", pipe)
        if mm:
            value = indent(value, int(mm.group(1)))
            continue
        raise ValueError(f"unsupported pipe: {pipe}")
    return str(value)
