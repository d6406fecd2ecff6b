# Engine + operator image (ROCm 7, gfx950).
FROM rocm/pytorch:rocm7.0_ubuntu22.04_py3.10_pytorch_2.10
WORKDIR /workspace
COPY . /workspace
ENV PYTORCH_ROCM_ARCH=gfx950
RUN python setup.py build_ext --inplace
# operator:  python -m kubeai_amd.controlplane.manager --config /config/config.yaml
# engine:    python -m kubeai_amd.engine.server --model <dir> --port 8000
ENTRYPOINT ["python"]
