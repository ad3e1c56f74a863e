# MI355X (gfx950) training/inference image: ROCm 7.x + PyTorch-ROCm.
FROM rocm/pytorch:latest
WORKDIR /workspace/perceiver-mi355x
COPY . .
ENV PYTORCH_ROCM_ARCH=gfx950
RUN pip install -e ".[text,vision,dev]" && python -m perceiver_amd.ops.build
CMD ["python", "-m", "pytest", "tests/", "-q", "-m", "not gpu"]
