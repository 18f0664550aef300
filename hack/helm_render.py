#!/usr/bin/env python3
"""Minimal Helm-template renderer (dev/CI tool).

No helm binary exists in this image, so this renders
``charts/aws-global-accelerator-controller`` with a small Go-template
evaluator covering exactly the constructs the chart uses (enumerated by
``grep -oh '{{[^}]*}}' templates/*.yaml``):

  {{ .Path.To.Value }}     context lookup (Values/Release/Chart)
  {{ include "name" . }}   named templates from _helpers.tpl ({{- define }})
  {{- if EXPR }} / {{- with EXPR }} / {{- end }}   blocks; with rebinds dot
  {{- toYaml EXPR | nindent N }}                   YAML re-serialization
  {{-  / -}}                                       whitespace trimming

Unsupported constructs raise, so a chart edit that outgrows the subset
fails loudly instead of rendering garbage.  ``--validate`` renders with
default values AND a matrix of overrides, YAML-parses every document and
checks minimal k8s shape (apiVersion/kind/metadata.name) — wired into
``make manifests-validate`` and covered by tests/test_helm_chart.py.
"""

from __future__ import annotations

import argparse
import re
import sys
from pathlib import Path

import yaml

CHART_DIR = Path(__file__).resolve().parent.parent / "charts" / "aws-global-accelerator-controller"

_ACTION = re.compile(r"\{\{(-?)\s*(.*?)\s*(-?)\}\}", re.S)


class TemplateError(Exception):
    pass


def _split(template: str):
    """Yield ('text', s) and ('action', body) segments with {{- -}} trims
    applied (Go template semantics: '-' eats adjacent whitespace)."""
    segments = []
    pos = 0
    for m in _ACTION.finditer(template):
        text = template[pos : m.start()]
        if m.group(1) == "-":
            text = text.rstrip(" \t\n")
        segments.append(("text", text))
        segments.append(("action", m.group(2), m.group(3) == "-"))
        pos = m.end()
    segments.append(("text", template[pos:]))
    # apply right-trim of an action to the following text segment
    out = []
    trim_next = False
    for seg in segments:
        if seg[0] == "text":
            text = seg[1]
            if trim_next:
                text = text.lstrip(" \t\n")
            trim_next = False
            out.append(("text", text))
        else:
            out.append(("action", seg[1]))
            trim_next = seg[2]
    return out


def _lookup(path: str, dot):
    """Resolve .A.B.C against the current dot context."""
    if path == ".":
        return dot
    value = dot
    for part in path.lstrip(".").split("."):
        if not isinstance(value, dict) or part not in value:
            return None
        value = value[part]
    return value


def _truthy(v) -> bool:
    return bool(v) and v != {} and v != []


def _to_yaml(value) -> str:
    return yaml.safe_dump(value, default_flow_style=False).rstrip("\n")


def _nindent(s: str, n: int) -> str:
    pad = " " * n
    return "\n" + "\n".join(pad + line if line else line for line in s.splitlines())


def _eval_expr(expr: str, dot, defines):
    """Evaluate a non-block action expression to a string/value."""
    expr = expr.strip()
    if expr.startswith("include "):
        m = re.match(r'include\s+"([^"]+)"\s+(\.\S*|\.)$', expr)
        if not m:
            raise TemplateError(f"unsupported include: {expr!r}")
        name, ctx_path = m.groups()
        if name not in defines:
            raise TemplateError(f"include of undefined template {name!r}")
        return render_segments(defines[name], _lookup(ctx_path, dot), defines).strip()
    if "|" in expr:
        left, *filters = [p.strip() for p in expr.split("|")]
        value = _eval_expr(left, dot, defines)
        for f in filters:
            m = re.match(r"nindent\s+(\d+)$", f)
            if not m:
                raise TemplateError(f"unsupported filter: {f!r}")
            value = _nindent(str(value), int(m.group(1)))
        return value
    if expr.startswith("toYaml"):
        m = re.match(r"toYaml\s+(\.\S*|\.)$", expr)
        if not m:
            raise TemplateError(f"unsupported toYaml: {expr!r}")
        return _to_yaml(_lookup(m.group(1), dot))
    if expr.startswith("."):
        value = _lookup(expr, dot)
        return "" if value is None else value
    raise TemplateError(f"unsupported expression: {expr!r}")


def render_segments(segments, dot, defines) -> str:
    out, i = [], 0

    def block(start: int):
        """Find the matching end for the block opened at segments[start]."""
        depth = 1
        j = start + 1
        while j < len(segments):
            seg = segments[j]
            if seg[0] == "action":
                body = seg[1].strip()
                if body.startswith(("if ", "if(", "with ", "range ")) or body in ("if", "with"):
                    depth += 1
                elif body == "end":
                    depth -= 1
                    if depth == 0:
                        return j
            j += 1
        raise TemplateError("unbalanced block: missing {{ end }}")

    while i < len(segments):
        seg = segments[i]
        if seg[0] == "text":
            out.append(seg[1])
            i += 1
            continue
        body = seg[1].strip()
        if body.startswith("if "):
            endpos = block(i)
            cond = _lookup(body[3:].strip(), dot)
            if _truthy(cond):
                out.append(render_segments(segments[i + 1 : endpos], dot, defines))
            i = endpos + 1
        elif body.startswith("with "):
            endpos = block(i)
            value = _lookup(body[5:].strip(), dot)
            if _truthy(value):
                out.append(render_segments(segments[i + 1 : endpos], value, defines))
            i = endpos + 1
        elif body.startswith("range "):
            raise TemplateError("range is not in the supported subset")
        elif body == "end":
            raise TemplateError("unexpected {{ end }}")
        else:
            out.append(str(_eval_expr(body, dot, defines)))
            i += 1
    return "".join(out)


def parse_defines(tpl_source: str) -> dict:
    """Extract {{- define "name" -}}...{{- end }} bodies from _helpers.tpl."""
    defines = {}
    pattern = re.compile(
        r'\{\{-?\s*define\s+"([^"]+)"\s*-?\}\}(.*?)\{\{-?\s*end\s*-?\}\}', re.S
    )
    for m in pattern.finditer(tpl_source):
        defines[m.group(1)] = _split(m.group(2))
    return defines


def _deep_merge(base: dict, override: dict) -> dict:
    merged = dict(base)
    for k, v in override.items():
        if isinstance(v, dict) and isinstance(merged.get(k), dict):
            merged[k] = _deep_merge(merged[k], v)
        else:
            merged[k] = v
    return merged


def render_chart(chart_dir: Path = CHART_DIR, values_override: dict | None = None,
                 namespace: str = "kube-system") -> dict:
    """Render every template; returns {relative_name: rendered_text}."""
    values = yaml.safe_load((chart_dir / "values.yaml").read_text()) or {}
    if values_override:
        values = _deep_merge(values, values_override)
    chart_meta = yaml.safe_load((chart_dir / "Chart.yaml").read_text()) or {}
    defines = {}
    helpers = chart_dir / "templates" / "_helpers.tpl"
    if helpers.exists():
        defines = parse_defines(helpers.read_text())
    dot = {
        "Values": values,
        "Release": {"Namespace": namespace, "Name": chart_meta.get("name", "release")},
        "Chart": {"Name": chart_meta.get("name", ""), "Version": chart_meta.get("version", "")},
    }
    rendered = {}
    for path in sorted((chart_dir / "templates").glob("*.yaml")):
        rendered[path.name] = render_segments(_split(path.read_text()), dot, defines)
    return rendered


def validate_rendered(rendered: dict) -> list:
    """YAML-parse every rendered doc; check k8s object shape.
    Returns [(template, kind, name), ...] of all objects."""
    objects = []
    for name, text in rendered.items():
        for doc in yaml.safe_load_all(text):
            if doc is None:
                continue
            for field in ("apiVersion", "kind"):
                if field not in doc:
                    raise TemplateError(f"{name}: rendered doc missing {field}: {doc}")
            meta = doc.get("metadata") or {}
            if not meta.get("name"):
                raise TemplateError(f"{name}: rendered {doc['kind']} has no metadata.name")
            objects.append((name, doc["kind"], meta["name"]))
    return objects


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--validate", action="store_true",
                        help="render default + override matrix and YAML-validate")
    parser.add_argument("--set-webhook-disabled", action="store_true")
    args = parser.parse_args()

    if args.validate:
        matrix = [
            ({}, "defaults"),
            ({"webhook": {"enabled": False}}, "webhook disabled"),
            ({"controller": {"metricsPort": 9090},
              "resources": {"limits": {"cpu": "500m"}},
              "nodeSelector": {"role": "infra"},
              "tolerations": [{"key": "infra", "operator": "Exists"}]},
             "metrics+scheduling overrides"),
        ]
        total = 0
        for override, label in matrix:
            objects = validate_rendered(render_chart(values_override=override))
            print(f"ok: {label}: {len(objects)} objects")
            total += len(objects)
        print(f"helm chart renders clean: {total} objects across {len(matrix)} value sets")
        return

    override = {"webhook": {"enabled": False}} if args.set_webhook_disabled else None
    for name, text in render_chart(values_override=override).items():
        print(f"---\n# Source: templates/{name}\n{text}")


if __name__ == "__main__":
    try:
        main()
    except TemplateError as e:
        print(f"helm render error: {e}", file=sys.stderr)
        sys.exit(1)
