# torchbeast_amd build/runtime environment (ref capability: the reference's
# Dockerfile builds a CUDA/conda stack; this one targets ROCm/MI355X).
#
# Requires the host to expose the GPUs: run with
#   docker run --device=/dev/kfd --device=/dev/dri --group-add video ...

FROM rocm/pytorch:latest

WORKDIR /workspace/torchbeast_amd
COPY . .

ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0

RUN python setup.py build_ext --inplace && \
    python -m pytest tests -q -m "not gpu"

# Benchmark-style training run (synthetic frames; no gym needed).
CMD ["python", "-m", "torchbeast_amd.polybeast_learner", \
     "--env", "synthetic:4x84x84:6", \
     "--num_actors", "512", \
     "--batch_size", "32", \
     "--unroll_length", "80", \
     "--total_steps", "200000000"]
