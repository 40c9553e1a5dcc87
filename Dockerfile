# MI355X gpushare device plugin image.
# Reference used a two-stage golang build (Dockerfile:1-12); here the build
# stage compiles the native extensions (amdsmi shim, devlist codec, gfx950
# canary) against the ROCm toolchain and the runtime stage carries only the
# package + ROCm runtime libs it dlopens (libamd_smi, libamdhip64).
FROM rocm/dev-ubuntu-22.04:7.2 AS build
WORKDIR /src
COPY . .
ENV PYTORCH_ROCM_ARCH=gfx950
RUN python3 -m pip install --no-cache-dir pybind11 && \
    python3 -m gpushare_amd.native.build && \
    python3 -m pip install --no-cache-dir .

FROM rocm/rocm-terminal:7.2
RUN python3 -m pip install --no-cache-dir grpcio protobuf pyyaml prometheus_client
COPY --from=build /usr/local/lib/python3*/dist-packages /usr/local/lib/python3/dist-packages
COPY --from=build /usr/local/bin/amdgpushare-device-plugin \
                  /usr/local/bin/gpushare-scheduler-extender \
                  /usr/local/bin/gpushare-top \
                  /usr/local/bin/kubectl-inspect-gpushare /usr/local/bin/
ENV HSA_ENABLE_IPC_MODE_LEGACY=0
ENTRYPOINT ["amdgpushare-device-plugin"]
