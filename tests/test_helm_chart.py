"""Helm chart render tests: the chart templates are
evaluated by hack/helm_render.py (a minimal Go-template evaluator covering
exactly the chart's construct subset — no helm binary in this image),
YAML-validated, value-matrix-checked and pinned against a golden file."""

import sys
from pathlib import Path

import pytest
import yaml

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO / "hack"))

import helm_render  # noqa: E402


def objects(rendered):
    out = {}
    for name, text in rendered.items():
        for doc in yaml.safe_load_all(text):
            if doc:
                out[(doc["kind"], doc["metadata"]["name"])] = doc
    return out


class TestDefaultRender:
    def test_validate_matrix_passes(self):
        for override in ({}, {"webhook": {"enabled": False}}):
            helm_render.validate_rendered(
                helm_render.render_chart(values_override=override)
            )

    def test_expected_objects(self):
        objs = objects(helm_render.render_chart())
        kinds = {k for k, _ in objs}
        assert "Deployment" in kinds
        assert "ClusterRole" in kinds
        assert "ClusterRoleBinding" in kinds
        assert "ServiceAccount" in kinds
        assert "ValidatingWebhookConfiguration" in kinds

    def test_values_propagate(self):
        objs = objects(helm_render.render_chart(values_override={
            "controller": {"replicas": 5, "workers": 3, "clusterName": "prod",
                           "metricsPort": 9090},
            "image": {"repository": "example.com/agac", "tag": "v1.2.3"},
        }, namespace="agac-system"))
        deploy = next(d for (k, _), d in objs.items() if k == "Deployment")
        assert deploy["spec"]["replicas"] == 5
        assert deploy["metadata"]["namespace"] == "agac-system"
        container = deploy["spec"]["template"]["spec"]["containers"][0]
        assert container["image"] == "example.com/agac:v1.2.3"
        assert "--workers=3" in container["args"]
        assert "--cluster-name=prod" in container["args"]
        assert "--metrics-port=9090" in container["args"]

    def test_metrics_port_zero_omits_flag(self):
        objs = objects(helm_render.render_chart())
        deploy = next(d for (k, _), d in objs.items() if k == "Deployment")
        args = deploy["spec"]["template"]["spec"]["containers"][0]["args"]
        assert not any(a.startswith("--metrics-port") for a in args)

    def test_webhook_disabled_removes_webhook_objects(self):
        objs = objects(helm_render.render_chart(
            values_override={"webhook": {"enabled": False}}
        ))
        kinds = {k for k, _ in objs}
        assert "ValidatingWebhookConfiguration" not in kinds
        assert "Deployment" in kinds  # controller still rendered

    def test_webhook_registers_create_and_update(self):
        """ADVICE r1 parity fix must hold in the chart too."""
        objs = objects(helm_render.render_chart())
        vwc = next(
            d for (k, _), d in objs.items()
            if k == "ValidatingWebhookConfiguration"
        )
        ops = vwc["webhooks"][0]["rules"][0]["operations"]
        assert ops == ["CREATE", "UPDATE"]

    def test_scheduling_overrides_render(self):
        objs = objects(helm_render.render_chart(values_override={
            "resources": {"limits": {"cpu": "500m", "memory": "128Mi"}},
            "nodeSelector": {"role": "infra"},
            "tolerations": [{"key": "infra", "operator": "Exists"}],
        }))
        deploy = next(d for (k, _), d in objs.items() if k == "Deployment")
        pod = deploy["spec"]["template"]["spec"]
        assert pod["containers"][0]["resources"]["limits"]["cpu"] == "500m"
        assert pod["nodeSelector"] == {"role": "infra"}
        assert pod["tolerations"] == [{"key": "infra", "operator": "Exists"}]

    def test_golden_file_matches(self):
        golden = (REPO / "tests" / "golden" / "helm_default_render.yaml").read_text()
        current = "".join(
            f"---\n# Source: templates/{name}\n{text}\n"
            for name, text in helm_render.render_chart().items()
        )
        assert current == golden, (
            "chart default render drifted from the golden file; if the "
            "change is intentional, regenerate with "
            "`python hack/helm_render.py > tests/golden/helm_default_render.yaml`"
        )


class TestEvaluator:
    """The evaluator itself must be trustworthy for the above to mean
    anything."""

    def render(self, tpl, dot, defines=None):
        return helm_render.render_segments(
            helm_render._split(tpl), dot, defines or {}
        )

    def test_substitution_and_trim(self):
        assert self.render("a: {{ .Values.x }}", {"Values": {"x": 7}}) == "a: 7"
        assert self.render("a\n{{- if .Values.on }}\nb\n{{- end }}\nc",
                           {"Values": {"on": True}}) == "a\nb\nc"
        assert self.render("a\n{{- if .Values.on }}\nb\n{{- end }}\nc",
                           {"Values": {"on": False}}) == "a\nc"

    def test_with_rebinds_dot(self):
        out = self.render(
            "{{- with .Values.sel }}\nsel:\n  {{ .role }}\n{{- end }}",
            {"Values": {"sel": {"role": "infra"}}},
        )
        assert "infra" in out
        assert self.render(
            "{{- with .Values.sel }}x{{- end }}", {"Values": {"sel": {}}}
        ) == ""

    def test_toyaml_nindent(self):
        out = self.render(
            "resources:{{- toYaml .Values.r | nindent 2 }}",
            {"Values": {"r": {"limits": {"cpu": "1"}}}},
        )
        assert yaml.safe_load(out) == {"resources": {"limits": {"cpu": "1"}}}

    def test_nested_blocks(self):
        tpl = "{{- if .Values.a }}A{{- if .Values.b }}B{{- end }}{{- end }}"
        assert self.render(tpl, {"Values": {"a": 1, "b": 1}}) == "AB"
        assert self.render(tpl, {"Values": {"a": 1, "b": 0}}) == "A"
        assert self.render(tpl, {"Values": {"a": 0, "b": 1}}) == ""

    def test_unsupported_constructs_raise(self):
        with pytest.raises(helm_render.TemplateError):
            self.render("{{ range .Values.list }}x{{ end }}", {"Values": {}})
        with pytest.raises(helm_render.TemplateError):
            self.render("{{ .Values.x | upper }}", {"Values": {"x": "a"}})
        with pytest.raises(helm_render.TemplateError):
            self.render("{{ if .Values.x }}unclosed", {"Values": {"x": 1}})

    def test_include_uses_helpers(self):
        defines = helm_render.parse_defines(
            '{{- define "t.name" -}}\n{{ .Chart.Name }}\n{{- end }}'
        )
        out = self.render('name: {{ include "t.name" . }}',
                          {"Chart": {"Name": "agac"}}, defines)
        assert out == "name: agac"
