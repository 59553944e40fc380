# MI355X autoscaler container image (reference row #11 equivalent:
# the reference shipped python:3 + pip install + CMD python main.py).
# Base: ROCm runtime with PyTorch-ROCm, matching the dev environment.
FROM rocm/pytorch:rocm7.0_ubuntu22.04_py3.10_pytorch_2.10

WORKDIR /opt/mi355x-autoscaler
COPY mi355x_autoscaler/ mi355x_autoscaler/
COPY setup.py bench.py ./

# build the gfx950 HIP extension in-tree (no JIT cache at runtime)
ENV PYTORCH_ROCM_ARCH=gfx950
RUN python setup.py build_ext --inplace

# one GPU is enough for the controller; sharded mode is launched via
# torchrun (see deploy/autoscaler.yaml)
ENTRYPOINT ["python", "-m", "mi355x_autoscaler"]
CMD ["--backend", "gpu", "--sleep", "10", "-v"]
