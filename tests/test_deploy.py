"""Deploy-layer checks: manifests parse, the chart's embedded device-plugin
config round-trips through the C++ parser, and the workload contract
(runtimeClassName + amd.com/gpu limit, README.md:164 analog) holds."""

import json
import subprocess
from pathlib import Path

import pytest
import yaml

REPO = Path(__file__).resolve().parent.parent
MANIFESTS = REPO / "deploy" / "manifests"
CHART = REPO / "deploy" / "helm" / "k3samd-device-plugin"
PLUGIN = REPO / "native" / "bin" / "k3samd-device-plugin"


def load_docs(path):
    return [d for d in yaml.safe_load_all(path.read_text()) if d]


def test_all_manifests_parse():
    files = sorted(MANIFESTS.glob("*.yaml"))
    assert len(files) >= 4
    for f in files:
        assert load_docs(f), f


@pytest.mark.parametrize("name,kind,gpus", [
    ("mi-stream.yaml", "Pod", "1"),
    ("rocm-smi.yaml", "Pod", "1"),
    ("mi-allreduce-8gpu.yaml", "Pod", "8"),
])
def test_pod_workload_contract(name, kind, gpus):
    doc = load_docs(MANIFESTS / name)[0]
    assert doc["kind"] == kind
    spec = doc["spec"]
    assert spec["runtimeClassName"] == "amd"
    assert spec["restartPolicy"] == "Never"
    limits = spec["containers"][0]["resources"]["limits"]
    assert limits["amd.com/gpu"] == gpus


def test_jellyfin_manifest():
    docs = load_docs(MANIFESTS / "jellyfin.yaml")
    dep = next(d for d in docs if d["kind"] == "Deployment")
    svc = next(d for d in docs if d["kind"] == "Service")
    pod = dep["spec"]["template"]["spec"]
    assert pod["runtimeClassName"] == "amd"
    limits = pod["containers"][0]["resources"]["limits"]
    assert limits["amd.com/gpu"] == "1"
    assert dep["spec"]["strategy"]["type"] == "Recreate"
    assert dep["spec"]["progressDeadlineSeconds"] == 600
    assert svc["spec"]["ports"][0]["port"] == 8096


def test_chart_values_surface():
    """The reference's values surface (values.yaml:1-18) must exist."""
    values = yaml.safe_load((CHART / "values.yaml").read_text())
    assert values["gfd"]["enabled"] is True
    assert values["runtimeClassName"] == "amd"
    cfg = yaml.safe_load(values["config"]["map"]["default"])
    assert cfg["version"] == "v1"
    assert cfg["flags"]["migStrategy"] == "none"
    ts = cfg["sharing"]["timeSlicing"]
    assert ts["renameByDefault"] is False
    assert ts["failRequestsGreaterThanOne"] is False
    assert ts["resources"][0] == {"name": "amd.com/gpu", "replicas": 4}


def test_chart_config_parses_in_cpp(tmp_path):
    """The embedded config must be accepted by the C++ plugin parser with
    the same semantics (replicas=4 => 4x fan-out)."""
    from sysfs_builder import build_tree
    values = yaml.safe_load((CHART / "values.yaml").read_text())
    cfg = tmp_path / "config.yaml"
    cfg.write_text(values["config"]["map"]["default"])
    root = build_tree(tmp_path / "sys", n_gpus=2)
    out = subprocess.run(
        [str(PLUGIN), "--config", str(cfg), "--oneshot"],
        env={"K3SAMD_SYSFS_ROOT": str(root)},
        capture_output=True, text=True, timeout=60)
    j = json.loads(out.stdout)
    assert j["allocatable"] == 8  # 2 GPUs x replicas 4


def test_chart_templates_exist():
    names = {p.name for p in (CHART / "templates").glob("*.yaml")}
    assert {"daemonset.yaml", "labeller-daemonset.yaml", "configmap.yaml",
            "runtimeclass.yaml"} <= names


def test_mi_burn_job_contract():
    """Node-acceptance Job: GPU contract knobs + no retries (a burn that
    failed must be inspected, not silently re-run)."""
    doc = load_docs(MANIFESTS / "mi-burn.yaml")[0]
    assert doc["kind"] == "Job"
    assert doc["spec"]["backoffLimit"] == 0
    pod = doc["spec"]["template"]["spec"]
    assert pod["runtimeClassName"] == "amd"
    assert pod["restartPolicy"] == "Never"
    c = pod["containers"][0]
    assert c["resources"]["limits"]["amd.com/gpu"] == "1"
    assert "--burn" in c["command"]


def test_generated_config_property(tmp_path):
    """Property: random valid config documents (PyYAML-dumped, so any
    quoting/indent style yaml.dump picks) parse in the C++ plugin with
    the same semantics PyYAML sees — replicas drive fan-out, rename
    drives the advertised name."""
    import random
    from sysfs_builder import build_tree
    rng = random.Random(99)
    root = build_tree(tmp_path / "sys", n_gpus=2)
    for trial in range(12):
        replicas = rng.choice([1, 2, 3, 4, 7, 16, 64, 256])
        rename = rng.choice([True, False])
        doc = {
            "version": "v1",
            "flags": {"migStrategy": "none"},
            "sharing": {"timeSlicing": {
                "renameByDefault": rename,
                "failRequestsGreaterThanOne": rng.choice([True, False]),
                "resources": [{"name": "amd.com/gpu",
                               "replicas": replicas}],
            }},
            "health": {
                "maxUncorrectableErrors": rng.choice([0, 5, -1]),
                "maxCorrectableErrors": rng.choice([100, 10000, -1]),
                "maxResets": rng.choice([0, -1]),
            },
        }
        cfg = tmp_path / f"cfg{trial}.yaml"
        cfg.write_text(yaml.dump(doc,
                                 default_flow_style=rng.choice([False]) ))
        out = subprocess.run(
            [str(PLUGIN), "--config", str(cfg), "--oneshot"],
            env={"K3SAMD_SYSFS_ROOT": str(root)},
            capture_output=True, text=True, timeout=60)
        assert out.returncode == 0, (doc, out.stderr)
        j = json.loads(out.stdout)
        assert j["allocatable"] == 2 * replicas, (doc, j)
        expect_name = "amd.com/gpu.shared" if (rename and replicas > 1) \
            else "amd.com/gpu"
        assert j["resource"] == expect_name, (doc, j)
