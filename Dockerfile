# Build image for the MI355X-native KV-cache framework.
# Base: ROCm 7.x + PyTorch-ROCm (the engine-side wheel additionally needs
# the gfx950 extensions built in-tree).
FROM rocm/pytorch:rocm7.2_ubuntu22.04_py3.10_pytorch
WORKDIR /app
COPY llm_d_kv_cache_amd ./llm_d_kv_cache_amd
COPY api ./api
COPY examples ./examples
COPY tools ./tools
ENV PYTORCH_ROCM_ARCH=gfx950
RUN python -m llm_d_kv_cache_amd build
# control-plane entrypoint by default; override for engine-side use
# (python -m llm_d_kv_cache_amd {serve,score,evict,build})
CMD ["python", "-m", "llm_d_kv_cache_amd", "serve"]
