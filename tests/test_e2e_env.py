"""The e2e environment itself: backend selection semantics and the
kubeconfig-driven live client construction (E2E_LIVE
must be real and must fail LOUDLY when misconfigured, never silently re-run
the simulator)."""
import base64

import pytest
import yaml

from gpu_provisioner_amd.kube.http import HTTPClient
from tests.e2e_env import DISCOVERY_LABEL, InProcessEnv, make_env


def test_default_backend_is_in_process(monkeypatch):
    monkeypatch.delenv("E2E_LIVE", raising=False)
    env = make_env()
    assert isinstance(env, InProcessEnv)
    assert not env.is_live


def test_live_mode_fails_loudly_without_kubeconfig(monkeypatch, tmp_path):
    monkeypatch.setenv("E2E_LIVE", "1")
    monkeypatch.setenv("KUBECONFIG", str(tmp_path / "does-not-exist"))
    with pytest.raises(FileNotFoundError):
        make_env()


def test_discovery_label_stamped_on_spec_objects():
    env = make_env()
    nc = env.nodeclaim("x", {"app": "a"})
    assert nc["metadata"]["labels"][DISCOVERY_LABEL] == env.run_id


def _write_kubeconfig(tmp_path, user: dict, cluster_extra: dict = None) -> str:
    cfg = {
        "apiVersion": "v1",
        "kind": "Config",
        "current-context": "e2e",
        "contexts": [{"name": "e2e", "context": {"cluster": "c1", "user": "u1"}}],
        "clusters": [
            {"name": "c1", "cluster": {"server": "https://10.0.0.1:6443", **(cluster_extra or {})}}
        ],
        "users": [{"name": "u1", "user": user}],
    }
    p = tmp_path / "kubeconfig"
    p.write_text(yaml.safe_dump(cfg))
    return str(p)


def test_from_kubeconfig_token_auth(tmp_path):
    path = _write_kubeconfig(
        tmp_path, {"token": "sekret"}, {"insecure-skip-tls-verify": True}
    )
    client = HTTPClient.from_kubeconfig(path)
    assert client.base_url == "https://10.0.0.1:6443"
    assert client.http.headers["Authorization"] == "Bearer sekret"


def test_from_kubeconfig_token_file_and_ca_data(tmp_path):
    import certifi

    tok = tmp_path / "tok"
    tok.write_text("filetoken\n")
    # a real PEM bundle (certifi's) stands in for the cluster CA — httpx
    # parses the verify path eagerly at client construction
    path = _write_kubeconfig(
        tmp_path,
        {"tokenFile": str(tok)},
        {"certificate-authority": certifi.where()},
    )
    client = HTTPClient.from_kubeconfig(path)
    assert client.http.headers["Authorization"] == "Bearer filetoken"


def test_from_kubeconfig_ca_data_inline(tmp_path):
    import certifi

    ca_b64 = base64.b64encode(open(certifi.where(), "rb").read()).decode()
    path = _write_kubeconfig(
        tmp_path, {"token": "t"}, {"certificate-authority-data": ca_b64}
    )
    client = HTTPClient.from_kubeconfig(path)
    assert client.http.headers["Authorization"] == "Bearer t"


def test_from_kubeconfig_rejects_exec_plugin(tmp_path):
    path = _write_kubeconfig(
        tmp_path, {"exec": {"command": "az"}}, {"insecure-skip-tls-verify": True}
    )
    with pytest.raises(ValueError, match="exec credential plugin"):
        HTTPClient.from_kubeconfig(path)


def test_from_kubeconfig_missing_context(tmp_path):
    p = tmp_path / "kc"
    p.write_text(yaml.safe_dump({"apiVersion": "v1", "clusters": [], "users": [], "contexts": []}))
    with pytest.raises(ValueError, match="current-context"):
        HTTPClient.from_kubeconfig(str(p))
