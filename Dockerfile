# arks-amd/runtime: one image for the engine pod, the operator, the gateway,
# the PD router and the model downloader (the reference splits these across
# dockerfiles/Dockerfile{,.gateway,.scripts}; here they are all Python
# entrypoints of the same package, selected by command).
#
#   engine:    python3 -m arks_amd.server --port 8080 --model /models/... \
#                  --served-model-name NAME --tensor-parallel-size 8
#   operator:  python3 -m arks_amd.controlplane
#   gateway:   python3 -m arks_amd.gateway --port 8080
#   router:    python3 -m arks_amd.router --pd-disaggregation ...
#   download:  python3 -m arks_amd.loader.download
#
# Base: ROCm 7.x PyTorch image with gfx950 support.
FROM rocm/pytorch:latest

WORKDIR /opt/arks
COPY arks_amd/ arks_amd/
COPY setup.py* pyproject.toml* ./

# Build the gfx950 HIP extension in-tree (no network needed at runtime).
ENV PYTORCH_ROCM_ARCH=gfx950
RUN python3 -m arks_amd.ops.build

ENV PYTHONPATH=/opt/arks
ENV HSA_ENABLE_IPC_MODE_LEGACY=0
EXPOSE 8080
CMD ["python3", "-m", "arks_amd.server", "--help"]
