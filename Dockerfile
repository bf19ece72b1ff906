# anovos_amd demo image — MI355X (gfx950) native. The reference's demo
# image bundled Spark + JVM; here the substrate is ROCm + PyTorch-ROCm.
# Run with the AMD GPU device nodes mapped:
#   docker build . -t anovos-amd-demo
#   docker run --device=/dev/kfd --device=/dev/dri anovos-amd-demo
FROM rocm/pytorch:latest

WORKDIR /anovos_amd
COPY anovos_amd ./anovos_amd
COPY config ./config
COPY tools ./tools
COPY bin ./bin
COPY run_anovos_demo.sh setup.py* ./

ENV PYTORCH_ROCM_ARCH=gfx950
# build the HIP extension in-tree for gfx950 (cross-compiles without a GPU)
RUN python -c "from anovos_amd.ops.hip.build import build; build(verbose=False)"

CMD ["bash", "run_anovos_demo.sh"]
