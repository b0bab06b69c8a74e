#!/usr/bin/env bash
# Generate a self-signed CA + server certificate for dev/e2e Prometheus
# TLS (the controller mandates HTTPS).  Counterpart of the reference's
# hack/tls-certs helper.
#   hack/gen-tls-certs.sh [OUT_DIR] [CN] [SAN]
set -euo pipefail

OUT=${1:-/tmp/wva-tls}
CN=${2:-prometheus.monitoring.svc}
SAN=${3:-DNS:${CN},IP:127.0.0.1}

mkdir -p "$OUT"
openssl req -x509 -newkey rsa:2048 -nodes -days 365 \
  -keyout "$OUT/ca.key" -out "$OUT/ca.crt" -subj "/CN=wva-dev-ca"
openssl req -newkey rsa:2048 -nodes \
  -keyout "$OUT/tls.key" -out "$OUT/tls.csr" -subj "/CN=${CN}"
openssl x509 -req -in "$OUT/tls.csr" -days 365 \
  -CA "$OUT/ca.crt" -CAkey "$OUT/ca.key" -CAcreateserial \
  -extfile <(printf "subjectAltName=%s" "$SAN") -out "$OUT/tls.crt"
rm -f "$OUT/tls.csr" "$OUT/ca.srl"

echo "wrote $OUT/{ca.crt,ca.key,tls.crt,tls.key}"
echo "controller env: PROMETHEUS_CA_CERT_PATH=$OUT/ca.crt"
