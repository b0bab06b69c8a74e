#!/usr/bin/env bash
# Tear down the emulated-MI355X Kind cluster created by setup.sh.
set -euo pipefail
kind delete cluster --name "${CLUSTER:-wva-amd}"
