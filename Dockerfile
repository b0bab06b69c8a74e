# Controller image (reference analog: the static-Go-binary Dockerfile).
# The control plane is pure CPython plus the torch-free native sizing
# module (_queue_native_cpu: pybind11/numpy/OpenMP build of the same host
# solver as the full gfx950 extension) — no libtorch, no ROCm in the
# image.  wva_amd.ops prefers torch-native > torch-free native > Python,
# so the deployed controller sizes fleets at native speed.
FROM python:3.10-slim AS build

RUN apt-get update && apt-get install -y --no-install-recommends g++ \
    && rm -rf /var/lib/apt/lists/* \
    && pip install --no-cache-dir pybind11==3.0.4 setuptools numpy==2.2.6

WORKDIR /src
COPY setup.py ./
COPY wva_amd/ wva_amd/
# torch absent here -> setup.py builds only _queue_native_cpu
RUN python setup.py build_ext --inplace

FROM python:3.10-slim

# libgomp1: _queue_native_cpu.so is built -fopenmp and links libgomp.so.1;
# without it the import fails and sizing silently degrades to pure Python
# (advisor r01 medium).
RUN apt-get update && apt-get install -y --no-install-recommends libgomp1 \
    && rm -rf /var/lib/apt/lists/*

COPY requirements-image.txt /tmp/requirements-image.txt
RUN pip install --no-cache-dir -r /tmp/requirements-image.txt

RUN useradd --uid 65532 --no-create-home nonroot
WORKDIR /app
COPY --from=build /src/wva_amd/ wva_amd/

# Image smoke check: the torch-free native sizing path must be importable
# in this exact runtime (catches a missing libgomp1 / ABI drift at build
# time rather than as a silent runtime fallback).
RUN python -c "from wva_amd import ops; assert ops.native_cpu_available(), ops._native_cpu_err"

USER 65532:65532
EXPOSE 8443 8081
# --kube-backend defaults to 'auto': in-cluster when KUBERNETES_SERVICE_HOST
# is present (any pod), the in-memory dev backend otherwise.
ENTRYPOINT ["python", "-m", "wva_amd"]
