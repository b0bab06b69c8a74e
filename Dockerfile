# Controller image (reference analog: the static-Go-binary Dockerfile).
# The control plane is pure CPython — the gfx950 queue-solver kernel is
# used by the offline profiling/benchmark tooling, and wva_amd.ops falls
# back to the pure-Python sizing path when the native extension is absent
# — so the runtime image is a slim Python base, not a ROCm base.
FROM python:3.10-slim

RUN pip install --no-cache-dir \
        pydantic fastapi uvicorn httpx prometheus-client pyyaml numpy

RUN useradd --uid 65532 --no-create-home nonroot
WORKDIR /app
COPY wva_amd/ wva_amd/

USER 65532:65532
EXPOSE 8443 8081
ENTRYPOINT ["python", "-m", "wva_amd"]
