# Controller image (reference analog: the static-Go-binary Dockerfile).
# The control plane is pure CPython plus the torch-free native sizing
# module (_queue_native_cpu: pybind11/numpy/OpenMP build of the same host
# solver as the full gfx950 extension) — no libtorch, no ROCm in the
# image.  wva_amd.ops prefers torch-native > torch-free native > Python,
# so the deployed controller sizes fleets at native speed.
FROM python:3.10-slim AS build

RUN apt-get update && apt-get install -y --no-install-recommends g++ \
    && rm -rf /var/lib/apt/lists/* \
    && pip install --no-cache-dir pybind11 setuptools numpy

WORKDIR /src
COPY setup.py ./
COPY wva_amd/ wva_amd/
# torch absent here -> setup.py builds only _queue_native_cpu
RUN python setup.py build_ext --inplace

FROM python:3.10-slim

RUN pip install --no-cache-dir \
        pydantic fastapi uvicorn httpx prometheus-client pyyaml numpy

RUN useradd --uid 65532 --no-create-home nonroot
WORKDIR /app
COPY --from=build /src/wva_amd/ wva_amd/

USER 65532:65532
EXPOSE 8443 8081
ENTRYPOINT ["python", "-m", "wva_amd"]
