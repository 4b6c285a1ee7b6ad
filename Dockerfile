# ROCm deployment image for the edge-node simulation (reference parity:
# Dockerfile:1-14, but on a ROCm base so the same image runs on MI355X).
FROM rocm/pytorch:latest

WORKDIR /app
COPY horizonml_amd ./horizonml_amd
COPY setup.py train.py data_parallel_train.py layer_model_parallel_train.py \
     tensor_parallel_train.py hybrid_parallel_train.py main.py ./

# Build the gfx950 extension in-image (cross-compiles without a GPU).
RUN PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace || true

RUN pip install --no-cache-dir pandas psutil tqdm matplotlib || true

CMD ["python", "train.py"]
