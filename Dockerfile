# cea_amd device-plugin / node-stack image.
# Parity: /root/reference/Dockerfile (the Go plugin image) — here the image
# carries the Python control plane + the in-tree native artifacts built for
# gfx950 (libceaamd_smi.so, libceaamd_gpu.so, all_reduce_perf).
FROM rocm/dev-ubuntu-22.04:7.2
WORKDIR /opt/cea-amd

RUN apt-get update && apt-get install -y --no-install-recommends \
      python3 python3-pip && \
    rm -rf /var/lib/apt/lists/* && \
    pip3 install --no-cache-dir grpcio protobuf prometheus_client \
      pyyaml requests fastapi uvicorn

COPY Makefile ./
COPY csrc/ csrc/
COPY cea_amd/ cea_amd/
COPY cmd/ cmd/
COPY bench.py ./

RUN make all && ln -s /opt/cea-amd/cea_amd/bin /opt/cea-amd/bin

ENV PYTHONPATH=/opt/cea-amd
ENTRYPOINT ["python3", "/opt/cea-amd/cmd/amd_gpu.py"]
CMD ["--enable-health-monitoring", "--enable-container-gpu-metrics"]
