# Operator image: ROCm base (hipcc for the gfx950 probe extension) +
# the Python operator.  Two-stage: build the HIP extension, then ship a
# slim runtime layer (reference Dockerfile is distroless Go; the amdgpu
# node path needs the ROCm runtime libraries).
FROM rocm/dev-ubuntu-22.04:7.2 AS build
WORKDIR /src
COPY cro_amd/ cro_amd/
RUN /opt/rocm/bin/hipcc --offload-arch=gfx950 -O3 -fPIC -shared \
      cro_amd/hip/probe.hip -o cro_amd/hip/libcroprobe.so && \
    /opt/rocm/bin/hipcc --offload-arch=gfx950 -O2 \
      cro_amd/agent/croagent.cpp -Lcro_amd/hip -lcroprobe \
      -Wl,-rpath,'$ORIGIN/../hip' -o cro_amd/agent/croagent

FROM rocm/rocm-runtime-ubuntu-22.04:7.2
RUN useradd -u 65532 -r nonroot
WORKDIR /app
COPY --from=build /src/cro_amd/ cro_amd/
COPY bench.py ./
RUN pip install --no-cache-dir pydantic httpx fastapi uvicorn prometheus_client pyyaml
USER 65532:65532
ENTRYPOINT ["python3", "-m", "cro_amd.cmd.main"]
