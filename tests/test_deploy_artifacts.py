"""Deploy-artifact lints: helm charts (reference: helm-chart/{kuberay-operator,
kuberay-apiserver,ray-cluster}) and Grafana dashboards (reference:
config/grafana/*.json) ship well-formed and cover the expected surface."""
import glob
import json
import os
import re

import yaml

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HELM = os.path.join(HERE, "deploy", "helm")
GRAFANA = os.path.join(HERE, "deploy", "grafana")


class TestHelmCharts:
    def test_all_three_charts_present(self):
        assert sorted(os.listdir(HELM)) == [
            "kuberay-amd-apiserver", "kuberay-amd-operator", "ray-cluster"]

    def test_chart_yaml_and_values_parse(self):
        for chart in os.listdir(HELM):
            meta = yaml.safe_load(open(os.path.join(HELM, chart, "Chart.yaml")))
            assert meta["apiVersion"] == "v2" and meta["name"], chart
            vals = yaml.safe_load(open(os.path.join(HELM, chart, "values.yaml")))
            assert isinstance(vals, dict) and vals, chart
            assert os.listdir(os.path.join(HELM, chart, "templates")), chart

    def test_template_value_refs_exist_in_values(self):
        """Every `.Values.x.y` referenced by a template resolves in the
        chart's values.yaml (catches rename drift without a helm binary)."""
        ref = re.compile(r"\.Values\.([A-Za-z0-9_.]+)")
        for chart in os.listdir(HELM):
            vals = yaml.safe_load(open(os.path.join(HELM, chart, "values.yaml")))
            for tpl in glob.glob(os.path.join(HELM, chart, "templates", "*")):
                for dotted in ref.findall(open(tpl).read()):
                    node = vals
                    for part in dotted.split("."):
                        assert isinstance(node, dict) and part in node, \
                            f"{chart}/{os.path.basename(tpl)}: .Values.{dotted}"
                        node = node[part]

    def test_ray_cluster_chart_is_mi355x_native(self):
        text = open(os.path.join(HELM, "ray-cluster", "templates",
                                 "raycluster.yaml")).read()
        vals = open(os.path.join(HELM, "ray-cluster", "values.yaml")).read()
        assert "amd.com/gpu" in text
        assert "nvidia" not in (text + vals).lower()
        assert "rocm" in vals  # ROCm image tag default


class TestGrafanaDashboards:
    def _dashboards(self):
        return sorted(glob.glob(os.path.join(GRAFANA, "*.json")))

    def test_reference_dashboard_families_covered(self):
        names = {os.path.basename(p) for p in self._dashboards()}
        for family in ("operator", "serve", "data", "train", "llm"):
            assert any(family in n for n in names), family

    def test_dashboards_are_valid_grafana_json(self):
        for path in self._dashboards():
            d = json.load(open(path))
            assert d.get("title") and d.get("panels"), path
            for p in d["panels"]:
                assert p.get("title"), path
                for t in p.get("targets", []):
                    assert t.get("expr"), f"{path}: {p['title']}"

    def test_promql_is_balanced(self):
        """Cheap PromQL sanity: parens/braces balance in every expr."""
        for path in self._dashboards():
            for p in json.load(open(path))["panels"]:
                for t in p.get("targets", []):
                    e = t["expr"]
                    assert e.count("(") == e.count(")"), e
                    assert e.count("{") == e.count("}"), e
