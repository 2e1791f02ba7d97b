"""Deployment artifacts stay well-formed (reference: charts/mcp-context-forge).
No cluster in CI — structural checks only."""

import pathlib
import re

import yaml

CHART = pathlib.Path(__file__).resolve().parent.parent / "deploy/helm/mcp-context-forge-amd"


def test_chart_and_values_parse():
    chart = yaml.safe_load((CHART / "Chart.yaml").read_text())
    assert chart["name"] == "mcp-context-forge-amd" and chart["apiVersion"] == "v2"
    values = yaml.safe_load((CHART / "values.yaml").read_text())
    assert values["gpu"]["resource"] == "amd.com/gpu"
    assert values["edge"]["workers"] >= 1
    assert "FORGE_DATABASE_URL" in values["env"]


def test_templates_reference_defined_values():
    values = yaml.safe_load((CHART / "values.yaml").read_text())

    def has_path(d, path):
        for part in path.split("."):
            if not isinstance(d, dict) or part not in d:
                return False
            d = d[part]
        return True

    for tpl in (CHART / "templates").glob("*.yaml"):
        for ref in re.findall(r"\.Values\.([A-Za-z0-9_.]+)", tpl.read_text()):
            assert has_path(values, ref), f"{tpl.name}: .Values.{ref} not in values.yaml"


def test_deployment_template_shape():
    text = (CHART / "templates/deployment.yaml").read_text()
    assert "serve" in text and "--workers" in text
    assert "readinessProbe" in text and "livenessProbe" in text
    assert "secretKeyRef" in text  # admin password never inline
