"""Render every chart template with the default values (mini renderer,
tests/helm_render.py) and validate the resulting Kubernetes objects."""

from pathlib import Path

import pytest
import yaml

from helm_render import Ctx, render

CHART = Path(__file__).resolve().parent.parent / "deploy" / "helm" / \
    "k3samd-device-plugin"


def render_template(name, values_overrides=None):
    values = yaml.safe_load((CHART / "values.yaml").read_text())
    if values_overrides:
        def deep_merge(dst, src):
            for k, v in src.items():
                if isinstance(v, dict) and isinstance(dst.get(k), dict):
                    deep_merge(dst[k], v)
                else:
                    dst[k] = v
        deep_merge(values, values_overrides)
    text = (CHART / "templates" / name).read_text()
    docs = [d for d in
            yaml.safe_load_all(render(text, Ctx(values))) if d]
    return docs


def test_daemonset_renders():
    (ds,) = render_template("daemonset.yaml")
    assert ds["kind"] == "DaemonSet"
    tmpl = ds["spec"]["template"]["spec"]
    assert tmpl["nodeSelector"] == {"amd.com/gpu.present": "true"}
    c = tmpl["containers"][0]
    assert c["image"] == "ghcr.io/k3samd/k3samd:0.1.0"
    assert "--config=/etc/k3samd/config.yaml" in c["command"]
    assert "--metrics-addr=0.0.0.0:9400" in c["command"]  # metrics default on
    assert c["livenessProbe"]["httpGet"]["path"] == "/metrics"
    assert "--use-cdi" not in c["command"]  # cdi default off
    vols = {v["name"] for v in tmpl["volumes"]}
    assert {"device-plugins", "config", "sys"} <= vols
    assert "cdi" not in vols


def test_daemonset_with_cdi():
    (ds,) = render_template("daemonset.yaml",
                            {"cdi": {"enabled": True}})
    tmpl = ds["spec"]["template"]["spec"]
    assert "--use-cdi" in tmpl["containers"][0]["command"]
    assert tmpl["initContainers"][0]["name"] == "cdi-gen"
    assert "cdi" in {v["name"] for v in tmpl["volumes"]}


def test_labeller_renders_only_when_gfd():
    docs = render_template("labeller-daemonset.yaml")
    assert len(docs) == 1
    ds = docs[0]
    assert "nodeSelector" not in ds["spec"]["template"]["spec"]
    docs_off = render_template("labeller-daemonset.yaml",
                               {"gfd": {"enabled": False}})
    assert docs_off == []


def test_configmap_embeds_plugin_config():
    (cm,) = render_template("configmap.yaml")
    assert cm["kind"] == "ConfigMap"
    cfg = yaml.safe_load(cm["data"]["config.yaml"])
    assert cfg["version"] == "v1"
    assert cfg["sharing"]["timeSlicing"]["resources"][0]["replicas"] == 4


def test_servicemonitor_gated():
    assert render_template("servicemonitor.yaml") == []
    docs = render_template("servicemonitor.yaml",
                           {"metrics": {"serviceMonitor": {"enabled": True}}})
    kinds = [d["kind"] for d in docs]
    assert kinds == ["Service", "ServiceMonitor"]
    sm = docs[1]
    assert sm["spec"]["endpoints"][0]["port"] == "metrics"


def test_runtimeclass_renders():
    (rc,) = render_template("runtimeclass.yaml")
    assert rc["kind"] == "RuntimeClass"
    assert rc["handler"] == "amd"
    assert rc["metadata"]["name"] == "amd"


def test_configmap_health_block_flows_through():
    """The additive health: thresholds reach the mounted config intact
    (the plugin's PluginConfig parser consumes exactly this document)."""
    (cm,) = render_template("configmap.yaml")
    cfg = yaml.safe_load(cm["data"]["config.yaml"])
    h = cfg["health"]
    assert h["maxUncorrectableErrors"] == 0
    assert h["maxCorrectableErrors"] == 10000
    assert h["maxDeferredErrors"] == 0
    assert h["maxFatalEvents"] == 0
    assert h["maxBadPages"] == -1
    assert h["maxResets"] == 0


def test_values_matrix_renders_valid_objects():
    """Sweep the user-facing knobs; every template must render to valid
    objects (or nothing, for gated templates) in every combination."""
    combos = [
        {},
        {"gfd": {"enabled": False}},
        {"cdi": {"enabled": True}},
        {"metrics": {"enabled": False}},
        {"runtimeClassName": "amd-exp"},
        {"gfd": {"enabled": False}, "cdi": {"enabled": True},
         "metrics": {"enabled": False}},
    ]
    for overrides in combos:
        for tpl in ("daemonset.yaml", "configmap.yaml",
                    "labeller-daemonset.yaml", "runtimeclass.yaml",
                    "servicemonitor.yaml"):
            docs = render_template(tpl, overrides)
            for d in docs:
                assert d.get("kind"), (tpl, overrides)
                assert d.get("apiVersion"), (tpl, overrides)
                meta = d.get("metadata", {})
                assert meta.get("name"), (tpl, overrides)
