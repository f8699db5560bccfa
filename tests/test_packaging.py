"""Packaging consistency: CRDs/values/examples parse; CRD schema covers the
fields the controller reads/writes; the chart's env surface matches the
config the binary consumes; the catalog's SKUs appear in examples."""
import os
import re

import yaml

from gpu_provisioner_amd.apis import v1 as karpv1
from gpu_provisioner_amd.providers.instancetype.catalog import InstanceTypeProvider

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CHART = os.path.join(ROOT, "charts", "gpu-provisioner-amd")


def _load(path):
    with open(path) as f:
        return list(yaml.safe_load_all(f))


def test_nodeclaim_crd_schema_covers_controller_fields():
    [crd] = _load(os.path.join(CHART, "crds", "karpenter.sh_nodeclaims.yaml"))
    assert crd["metadata"]["name"] == "nodeclaims.karpenter.sh"
    assert crd["spec"]["scope"] == "Cluster"
    v1 = crd["spec"]["versions"][0]
    assert v1["name"] == "v1" and v1["served"] and v1["storage"]
    assert v1["subresources"] == {"status": {}}
    spec_props = v1["schema"]["openAPIV3Schema"]["properties"]["spec"]["properties"]
    # every spec field the controller reads must be in the schema
    for field in ("nodeClassRef", "requirements", "resources", "taints",
                  "startupTaints", "terminationGracePeriod"):
        assert field in spec_props, field
    status_props = v1["schema"]["openAPIV3Schema"]["properties"]["status"]["properties"]
    for field in ("providerID", "imageID", "nodeName", "capacity", "allocatable", "conditions"):
        assert field in status_props, field
    ops = spec_props["requirements"]["items"]["properties"]["operator"]["enum"]
    assert set(ops) == {"In", "NotIn", "Exists", "DoesNotExist", "Gt", "Lt"}


def test_kaitonodeclass_crd_parses():
    [crd] = _load(os.path.join(CHART, "crds", "kaito.sh_kaitonodeclasses.yaml"))
    assert crd["metadata"]["name"] == "kaitonodeclasses.kaito.sh"
    assert crd["spec"]["group"] == "kaito.sh"


def test_values_surface_matches_reference_contract():
    [values] = _load(os.path.join(CHART, "values.yaml"))
    az = values["settings"]["azure"]
    for key in ("location", "resourceGroup", "subscriptionID", "tenantID",
                "clusterName", "clientID"):
        assert key in az, key
    assert values["settings"]["deploymentMode"] == "self-hosted"
    assert values["replicas"] == 1
    assert values["resources"]["requests"]["cpu"] == "200m"
    assert values["resources"]["limits"]["cpu"] == "500m"
    assert values["amdDevicePlugin"]["resourceName"] == karpv1.AMD_GPU_RESOURCE
    assert values["podLabels"]["azure.workload.identity/use"] == "true"


def test_deployment_env_covers_azure_config():
    """Every env var build_azure_config() requires must be wired in the
    deployment template (as the reference's deployment.yaml:69-90 does)."""
    with open(os.path.join(CHART, "templates", "deployment.yaml")) as f:
        text = f.read()
    for var in ("LOCATION", "ARM_SUBSCRIPTION_ID", "ARM_RESOURCE_GROUP",
                "AZURE_TENANT_ID", "AZURE_CLIENT_ID", "AZURE_CLUSTER_NAME",
                "DEPLOYMENT_MODE", "METRICS_PORT", "HEALTH_PROBE_PORT",
                "KARPENTER_SERVICE"):
        assert var in text, f"deployment template missing env {var}"
    assert "/healthz" in text and "/readyz" in text


def test_examples_use_catalog_skus_and_valid_names():
    catalog = InstanceTypeProvider()
    for name in ("v1-nodeclaim-mi355x.yaml", "azure-linux-annotation-nodeclaim.yaml"):
        [nc] = _load(os.path.join(ROOT, "examples", name))
        assert nc["apiVersion"] == "karpenter.sh/v1"
        assert re.match(r"^[a-z][a-z0-9]{0,11}$", nc["metadata"]["name"]), nc["metadata"]["name"]
        for req in nc["spec"]["requirements"]:
            if req["key"] == karpv1.INSTANCE_TYPE_LABEL_KEY:
                for sku in req["values"]:
                    assert catalog.get(sku) is not None, f"unknown SKU {sku}"
                    assert catalog.is_gpu_sku(sku)


def test_workload_example_requests_amd_gpu():
    [job] = _load(os.path.join(ROOT, "examples", "workload", "gpu-amd.yaml"))
    container = job["spec"]["template"]["spec"]["containers"][0]
    assert container["resources"]["requests"][karpv1.AMD_GPU_RESOURCE] == "8"
    sel = job["spec"]["template"]["spec"]["nodeSelector"]
    assert sel[karpv1.AMD_GPU_PRODUCT_LABEL_KEY] == "AMD-Instinct-MI355X"


def test_rbac_covers_controller_verbs():
    with open(os.path.join(CHART, "templates", "rbac.yaml")) as f:
        text = f.read()
    for resource in ("nodeclaims", "nodes", "pods/eviction", "events",
                     "volumeattachments", "leases", "customresourcedefinitions"):
        assert resource in text, f"rbac missing {resource}"


def test_node_stack_daemonsets_target_mi355x_nodes():
    with open(os.path.join(CHART, "templates", "amd-gpu-node-stack.yaml")) as f:
        text = f.read()
    assert "amd.com/gpu.product: AMD-Instinct-MI355X" in text
    assert "karpenter.sh/nodepool: kaito" in text
    assert "/dev/kfd" in text  # ROCm device nodes
    assert "gpu_provisioner_amd.nodeagent" in text


def test_dockerfile_builds_gfx950_agent():
    with open(os.path.join(ROOT, "Dockerfile")) as f:
        text = f.read()
    assert "--offload-arch=gfx950" in text
    assert "libmi355x_nodeagent.so" in text
    assert "USER 65532" in text  # nonroot, like the reference's distroless
