# Two-stage build (mirrors the reference's golang->slim structure,
# Dockerfile:1-17, retargeted to ROCm): build the native extensions against
# the ROCm toolchain, then run on the same base (the HIP runtime is needed
# at runtime for the agent's probe; the scheduler itself is CPU-only).
FROM rocm/pytorch:latest AS build
WORKDIR /src
COPY . .
RUN python build_native.py

FROM rocm/pytorch:latest
WORKDIR /app
COPY --from=build /src /app
EXPOSE 39999
ENTRYPOINT ["python", "-m", "elastic_gpu_scheduler_amd.cmd.main"]
