"""Prometheus API clients: HTTPS (mTLS/bearer) and in-memory mock.

Parity with the reference's transport stack
(/root/reference/internal/utils/{tls,prometheus_transport}.go): mandatory
https:// scheme, minimum TLS 1.2, optional CA pool / client certificates /
SNI server name, bearer token directly or from a mounted file.

The query surface is the subset of promv1.API the controller uses: instant
vector queries returning (labels, value, timestamp) samples.
"""

from __future__ import annotations

import os
import ssl
import time
from dataclasses import dataclass, field
from typing import Dict, List, Protocol

from .interfaces import PrometheusConfig


@dataclass
class Sample:
    value: float
    timestamp: float = 0.0  # unix seconds; 0 -> "now"
    labels: Dict[str, str] = field(default_factory=dict)


class PromAPI(Protocol):
    def query(self, query: str) -> List[Sample]: ...


class PromQueryError(RuntimeError):
    pass


# ---------------------------------------------------------------- TLS config
def validate_tls_config(config: PrometheusConfig) -> None:
    """HTTPS is mandatory; certificate files must exist unless verification
    is explicitly skipped (tls.go:63-97)."""
    from urllib.parse import urlparse

    u = urlparse(config.base_url)
    if u.scheme != "https":
        raise ValueError(
            f"HTTPS is required - URL must use https:// scheme: {config.base_url}"
        )
    if config.insecure_skip_verify:
        return
    for path, what in (
        (config.ca_cert_path, "CA certificate"),
        (config.client_cert_path, "client certificate"),
        (config.client_key_path, "client key"),
    ):
        if path and not os.path.exists(path):
            raise ValueError(f"{what} file not found: {path}")


def create_ssl_context(config: PrometheusConfig) -> ssl.SSLContext:
    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_CLIENT)
    ctx.minimum_version = ssl.TLSVersion.TLSv1_2
    if config.insecure_skip_verify:
        ctx.check_hostname = False
        ctx.verify_mode = ssl.CERT_NONE
    elif config.ca_cert_path:
        ctx.load_verify_locations(cafile=config.ca_cert_path)
    else:
        ctx.load_default_certs()
    if config.client_cert_path and config.client_key_path:
        ctx.load_cert_chain(config.client_cert_path, config.client_key_path)
    return ctx


def parse_prometheus_config_from_env() -> PrometheusConfig:
    return PrometheusConfig(
        base_url=os.environ.get("PROMETHEUS_BASE_URL", ""),
        insecure_skip_verify=os.environ.get("PROMETHEUS_TLS_INSECURE_SKIP_VERIFY") == "true",
        ca_cert_path=os.environ.get("PROMETHEUS_CA_CERT_PATH", ""),
        client_cert_path=os.environ.get("PROMETHEUS_CLIENT_CERT_PATH", ""),
        client_key_path=os.environ.get("PROMETHEUS_CLIENT_KEY_PATH", ""),
        server_name=os.environ.get("PROMETHEUS_SERVER_NAME", ""),
        bearer_token=os.environ.get("PROMETHEUS_BEARER_TOKEN", ""),
        token_path=os.environ.get("PROMETHEUS_TOKEN_PATH", ""),
    )


class HTTPPromAPI:
    """HTTPS Prometheus client over httpx (instant queries)."""

    def __init__(self, config: PrometheusConfig) -> None:
        import httpx

        validate_tls_config(config)
        token = config.bearer_token
        if not token and config.token_path:
            with open(config.token_path) as f:
                token = f.read().strip()
        headers = {"Authorization": f"Bearer {token}"} if token else {}
        self._client = httpx.Client(
            base_url=config.base_url,
            headers=headers,
            verify=create_ssl_context(config),
            timeout=10.0,
        )

    def query(self, query: str) -> List[Sample]:
        try:
            resp = self._client.get("/api/v1/query", params={"query": query})
            resp.raise_for_status()
            body = resp.json()
        except Exception as e:
            raise PromQueryError(f"failed to query Prometheus: {e}") from e
        if body.get("status") != "success":
            raise PromQueryError(f"Prometheus query failed: {body}")
        data = body.get("data", {})
        if data.get("resultType") != "vector":
            return []
        out = []
        for item in data.get("result", []):
            ts, val = item.get("value", [0, "nan"])
            out.append(
                Sample(value=float(val), timestamp=float(ts), labels=item.get("metric", {}))
            )
        return out


class MockPromAPI:
    """In-memory promv1.API analog for tests.

    Parity with /root/reference/test/utils/unitutils.go:137-243: configured
    results per exact query string; configured errors; unknown queries
    default to a single fresh zero sample so availability validation
    passes.
    """

    def __init__(self) -> None:
        self.query_results: Dict[str, List[Sample]] = {}
        self.query_errors: Dict[str, Exception] = {}
        self.queries_seen: List[str] = []

    def set_result(self, query: str, value: float, *, age_seconds: float = 0.0) -> None:
        self.query_results[query] = [
            Sample(value=value, timestamp=time.time() - age_seconds)
        ]

    def set_error(self, query: str, err: Exception) -> None:
        self.query_errors[query] = err

    def query(self, query: str) -> List[Sample]:
        self.queries_seen.append(query)
        if query in self.query_errors:
            raise PromQueryError(str(self.query_errors[query]))
        if query in self.query_results:
            return self.query_results[query]
        return [Sample(value=0.0, timestamp=time.time())]
