# gofr-mi355x serving image (reference: Dockerfile:1-13, adapted to a
# ROCm runtime base — the data plane needs the HIP runtime + gfx950).
FROM rocm/pytorch:rocm7.2_ubuntu22.04_py3.10_pytorch AS build
WORKDIR /src
COPY . .
ENV PYTORCH_ROCM_ARCH=gfx950
RUN python setup.py build_ext --inplace && \
    python -c "import __graft_entry__ as g; g.build()"

FROM rocm/pytorch:rocm7.2_ubuntu22.04_py3.10_pytorch
WORKDIR /app
COPY --from=build /src /app
ENV HSA_ENABLE_IPC_MODE_LEGACY=0
EXPOSE 8000 9000
CMD ["python", "examples/http-server/main.py"]
