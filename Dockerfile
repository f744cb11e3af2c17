# production-stack-amd engine/router image (ROCm 7.x base with PyTorch-ROCm)
FROM rocm/pytorch:rocm7.0_ubuntu22.04_py3.10_pytorch_2.10
WORKDIR /workspace
COPY . /workspace
ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0
RUN python3 setup.py build_ext --inplace && \
    make -C operator && \
    pip install fastapi uvicorn aiohttp httpx msgpack xxhash \
        prometheus-client psutil safetensors
# engine:  python3 -m production_stack_amd.engine.server <model>
# router:  python3 -m production_stack_amd.router.app --help
# kv ctrl: python3 -m production_stack_amd.kvpool.controller
# operator: /workspace/operator/psoperator
EXPOSE 8000 8001 9000 9400 14001
CMD ["python3", "-m", "production_stack_amd.engine.server", "llama-3-8b"]
