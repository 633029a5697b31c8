# MI355X (gfx950) runtime image — replaces the reference's CUDA 11.4 image
# (/root/reference/Dockerfile: nvidia/cuda:11.4.1-cudnn8 + jax[cuda11]).
FROM rocm/pytorch:rocm7.2_ubuntu22.04_py3.10_pytorch_release_2.10

ENV PYTORCH_ROCM_ARCH=gfx950 \
    HSA_ENABLE_IPC_MODE_LEGACY=0

WORKDIR /workspace/3dim-mi355x
COPY . .

RUN pip install --no-deps -e . && \
    python setup.py build_ext --inplace

# training:  python train.py --folder /data/cars_train_val --model full
# 8-GPU DP:  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
#                --master-addr 127.0.0.1 train.py --model full --sidelength 128
CMD ["python", "train.py", "--help"]
