"""Chart sanity without helm: plain-YAML chart files parse, and every
`.Values.<path>` referenced by a template resolves in values.yaml — the
drift that otherwise only surfaces at install time."""
import os
import re

import yaml

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CHART = os.path.join(ROOT, "charts", "gpu-provisioner-amd")

_VALUES_RE = re.compile(r"\.Values\.([A-Za-z0-9_.]+)")


def test_plain_chart_yaml_parses():
    for rel in ("Chart.yaml", "values.yaml"):
        docs = list(yaml.safe_load_all(open(os.path.join(CHART, rel))))
        assert docs and docs[0], rel
    for fname in os.listdir(os.path.join(CHART, "crds")):
        docs = list(yaml.safe_load_all(open(os.path.join(CHART, "crds", fname))))
        assert docs and docs[0]["kind"] == "CustomResourceDefinition", fname


def _resolve(values: dict, path: str) -> bool:
    cur = values
    for seg in path.split("."):
        if isinstance(cur, dict) and seg in cur:
            cur = cur[seg]
        else:
            return False
    return True


def test_template_value_references_resolve():
    values = yaml.safe_load(open(os.path.join(CHART, "values.yaml")))
    missing = []
    tpl_dir = os.path.join(CHART, "templates")
    for fname in os.listdir(tpl_dir):
        src = open(os.path.join(tpl_dir, fname)).read()
        for m in _VALUES_RE.finditer(src):
            path = m.group(1)
            if not _resolve(values, path):
                missing.append(f"{fname}: .Values.{path}")
    assert not missing, "template references without values defaults:\n" + "\n".join(
        sorted(set(missing))
    )


def test_values_azure_settings_match_deployment_env():
    """The env the Deployment wires must cover everything auth.config
    requires (LOCATION, ARM_*, AZURE_*) — install-time drift guard."""
    deployment = open(os.path.join(CHART, "templates", "deployment.yaml")).read()
    for env in (
        "LOCATION",
        "ARM_SUBSCRIPTION_ID",
        "ARM_RESOURCE_GROUP",
        "AZURE_TENANT_ID",
        "AZURE_CLIENT_ID",
        "AZURE_CLUSTER_NAME",
        "DEPLOYMENT_MODE",
        "METRICS_PORT",
        "HEALTH_PROBE_PORT",
    ):
        assert f"name: {env}" in deployment, f"deployment missing env {env}"
