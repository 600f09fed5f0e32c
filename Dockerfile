# kolibrie_amd — MI355X-native SPARQL/RDF stream-reasoning engine.
# (ref: reference Dockerfile uses Ubuntu 22.04 + Rust; the MI355X build
#  ships on the ROCm base image with PyTorch-ROCm and hipcc for gfx950.)
FROM rocm/pytorch:latest

WORKDIR /opt/kolibrie_amd
COPY . .

ENV PYTORCH_ROCM_ARCH=gfx950
RUN python3 setup.py build_ext --inplace \
 && python3 -m pytest tests -q -m "not gpu"

EXPOSE 8080
# web-UI profile (ref docker-compose.yml web-ui vs dev profiles)
CMD ["python3", "-m", "kolibrie_amd.frontends.http_server", \
     "--host", "0.0.0.0", "--port", "8080", "--device", "cuda:0"]
