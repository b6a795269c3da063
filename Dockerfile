# gordo-amd base image: builder + server + client in one image
# (the reference's unified gordo-base image, Dockerfile:1-105).
# Expects a ROCm base with PyTorch-ROCm for MI355X.
FROM rocm/pytorch:rocm7.0_ubuntu22.04_py3.10_pytorch_2.10
ENV PYTORCH_ROCM_ARCH=gfx950

WORKDIR /code
COPY . /code

RUN python setup.py build_ext --inplace && pip install --no-deps -e .

# k8s entrypoints used by the Argo workflow template
COPY build.sh /usr/bin/build
RUN chmod a+x /usr/bin/build

CMD ["gordo", "--help"]
