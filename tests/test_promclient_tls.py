"""HTTPS Prometheus transport tests: real TLS handshake against a local
HTTPS endpoint with a self-signed CA, bearer-token injection (direct and
from file), and the mandatory-HTTPS validation paths.

The e2e analog of the reference's TLS handshake checks
(test/e2e/e2e_test.go:563-628) without a cluster."""

import subprocess
import threading
import time

import pytest

from wva_amd.controller.interfaces import PrometheusConfig
from wva_amd.controller.promclient import (
    HTTPPromAPI,
    PromQueryError,
    create_ssl_context,
    validate_tls_config,
)
from wva_amd.controller.utils import validate_prometheus_api, Backoff


@pytest.fixture(scope="module")
def tls_material(tmp_path_factory):
    """Self-signed server certificate for 127.0.0.1."""
    d = tmp_path_factory.mktemp("tls")
    crt, key = d / "tls.crt", d / "tls.key"
    subprocess.run(
        [
            "openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
            "-keyout", str(key), "-out", str(crt), "-days", "2",
            "-subj", "/CN=127.0.0.1",
            "-addext", "subjectAltName=IP:127.0.0.1",
        ],
        check=True,
        capture_output=True,
    )
    return {"crt": str(crt), "key": str(key), "dir": d}


class PromStub:
    """Minimal HTTPS Prometheus /api/v1/query endpoint with bearer check."""

    def __init__(self, tls, token=None):
        from fastapi import FastAPI, Request
        from fastapi.responses import JSONResponse
        import uvicorn

        app = FastAPI()
        self.queries = []

        @app.get("/api/v1/query")
        async def query(request: Request):
            if token is not None:
                if request.headers.get("authorization") != f"Bearer {token}":
                    return JSONResponse({"status": "error", "error": "unauthorized"}, status_code=401)
            q = request.query_params.get("query", "")
            self.queries.append(q)
            return {
                "status": "success",
                "data": {
                    "resultType": "vector",
                    "result": [
                        {"metric": {"__name__": q.split("{")[0]}, "value": [time.time(), "1"]}
                    ],
                },
            }

        self._server = uvicorn.Server(
            uvicorn.Config(
                app,
                host="127.0.0.1",
                port=0,
                log_level="error",
                ssl_certfile=tls["crt"],
                ssl_keyfile=tls["key"],
            )
        )
        self._thread = threading.Thread(target=self._server.run, daemon=True)

    def __enter__(self):
        self._thread.start()
        for _ in range(200):
            if self._server.started:
                break
            time.sleep(0.05)
        assert self._server.started
        port = self._server.servers[0].sockets[0].getsockname()[1]
        self.base_url = f"https://127.0.0.1:{port}"
        return self

    def __exit__(self, *exc):
        self._server.should_exit = True
        self._thread.join(timeout=5.0)


class TestHTTPSTransport:
    def test_query_with_ca_verification(self, tls_material):
        with PromStub(tls_material) as stub:
            api = HTTPPromAPI(
                PrometheusConfig(base_url=stub.base_url, ca_cert_path=tls_material["crt"])
            )
            out = api.query("up")
            assert out[0].value == 1.0
            assert out[0].timestamp > 0
            # the startup probe path works against a live endpoint
            validate_prometheus_api(api, Backoff(duration=0.1, factor=2.0, jitter=0.0, steps=2))

    def test_untrusted_cert_rejected(self, tls_material):
        with PromStub(tls_material) as stub:
            api = HTTPPromAPI(PrometheusConfig(base_url=stub.base_url))  # system CAs only
            with pytest.raises(PromQueryError):
                api.query("up")

    def test_insecure_skip_verify(self, tls_material):
        with PromStub(tls_material) as stub:
            api = HTTPPromAPI(
                PrometheusConfig(base_url=stub.base_url, insecure_skip_verify=True)
            )
            assert api.query("up")[0].value == 1.0

    def test_bearer_token_direct(self, tls_material):
        with PromStub(tls_material, token="s3cret") as stub:
            ok = HTTPPromAPI(
                PrometheusConfig(
                    base_url=stub.base_url,
                    ca_cert_path=tls_material["crt"],
                    bearer_token="s3cret",
                )
            )
            assert ok.query("up")[0].value == 1.0
            bad = HTTPPromAPI(
                PrometheusConfig(base_url=stub.base_url, ca_cert_path=tls_material["crt"])
            )
            with pytest.raises(PromQueryError):
                bad.query("up")

    def test_bearer_token_from_file(self, tls_material, tmp_path):
        token_file = tmp_path / "token"
        token_file.write_text("fil3token\n")
        with PromStub(tls_material, token="fil3token") as stub:
            api = HTTPPromAPI(
                PrometheusConfig(
                    base_url=stub.base_url,
                    ca_cert_path=tls_material["crt"],
                    token_path=str(token_file),
                )
            )
            assert api.query("up")[0].value == 1.0


class TestTLSValidation:
    def test_https_mandatory(self):
        with pytest.raises(ValueError):
            validate_tls_config(PrometheusConfig(base_url="http://x:9090"))
        validate_tls_config(PrometheusConfig(base_url="https://x:9090"))

    def test_missing_cert_files(self):
        with pytest.raises(ValueError):
            validate_tls_config(
                PrometheusConfig(base_url="https://x", ca_cert_path="/nope/ca.crt")
            )
        # skipped entirely when verification is off
        validate_tls_config(
            PrometheusConfig(
                base_url="https://x", ca_cert_path="/nope/ca.crt", insecure_skip_verify=True
            )
        )

    def test_min_tls_version(self, tls_material):
        import ssl

        ctx = create_ssl_context(PrometheusConfig(base_url="https://x", ca_cert_path=tls_material["crt"]))
        assert ctx.minimum_version >= ssl.TLSVersion.TLSv1_2

    def test_env_parsing(self, monkeypatch):
        from wva_amd.controller.promclient import parse_prometheus_config_from_env

        monkeypatch.setenv("PROMETHEUS_BASE_URL", "https://env:9090")
        monkeypatch.setenv("PROMETHEUS_TLS_INSECURE_SKIP_VERIFY", "true")
        monkeypatch.setenv("PROMETHEUS_BEARER_TOKEN", "tok")
        config = parse_prometheus_config_from_env()
        assert config.base_url == "https://env:9090"
        assert config.insecure_skip_verify is True
        assert config.bearer_token == "tok"

    def test_env_parsing_openshift_matrix(self, monkeypatch):
        # tls_test.go:86-117 — the full OpenShift thanos-querier env shape:
        # every field lands, empty client cert/key stay empty
        from wva_amd.controller.promclient import parse_prometheus_config_from_env

        monkeypatch.setenv(
            "PROMETHEUS_BASE_URL",
            "https://thanos-querier.openshift-monitoring.svc.cluster.local:9091",
        )
        monkeypatch.setenv("PROMETHEUS_TLS_INSECURE_SKIP_VERIFY", "false")
        monkeypatch.setenv("PROMETHEUS_CA_CERT_PATH", "/etc/openshift-ca/ca.crt")
        monkeypatch.setenv("PROMETHEUS_CLIENT_CERT_PATH", "")
        monkeypatch.setenv("PROMETHEUS_CLIENT_KEY_PATH", "")
        monkeypatch.setenv(
            "PROMETHEUS_SERVER_NAME", "thanos-querier.openshift-monitoring.svc"
        )
        monkeypatch.setenv(
            "PROMETHEUS_TOKEN_PATH",
            "/var/run/secrets/kubernetes.io/serviceaccount/token",
        )
        config = parse_prometheus_config_from_env()
        assert config.base_url.endswith(":9091")
        assert config.insecure_skip_verify is False
        assert config.ca_cert_path == "/etc/openshift-ca/ca.crt"
        assert config.client_cert_path == ""
        assert config.client_key_path == ""
        assert config.server_name == "thanos-querier.openshift-monitoring.svc"
        assert config.token_path == "/var/run/secrets/kubernetes.io/serviceaccount/token"
