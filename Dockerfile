# gpu-docker-api-amd daemon image (ROCm base: hipcc + amdsmi + librccl).
# The daemon itself needs no GPU at build time — hipcc cross-compiles gfx950.
FROM rocm/dev-ubuntu-22.04:7.0
WORKDIR /app
COPY gpu_docker_api_amd ./gpu_docker_api_amd
COPY csrc ./csrc
COPY Makefile bench.py ./
RUN python3 -m pip install --no-cache-dir fastapi uvicorn aiohttp httpx pydantic pybind11 \
 && python3 -m gpu_docker_api_amd.ops.build
EXPOSE 2378
# docker runtime driver needs /var/run/docker.sock mounted; proc runtime does not
ENTRYPOINT ["python3", "-m", "gpu_docker_api_amd"]
CMD ["--runtime", "docker", "--inventory", "auto", "--addr", "0.0.0.0:2378"]
