# WVA-AMD controller image (ROCm base for optional on-node calibration).
# The control plane itself is CPU-only; the ops extension is built for
# gfx950 at image build time so calibration jobs can run on MI355X nodes.
FROM rocm/pytorch:rocm7.0_ubuntu22.04_py3.10_pytorch_2.10

WORKDIR /app
COPY wva_amd/ wva_amd/
COPY deploy/ deploy/
COPY pyproject.toml README.md ./

RUN PYTORCH_ROCM_ARCH=gfx950 python -m wva_amd.ops.build

ENTRYPOINT ["python", "-m", "wva_amd"]
