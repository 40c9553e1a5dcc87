# MI355X gpushare device plugin image.
#
# Reference used a two-stage golang build (Dockerfile:1-12); here the build
# stage compiles the native pieces (amdsmi shim, devlist codec, gfx950
# canary, memguard preload lib) against the ROCm toolchain and the runtime
# stage carries only the package + the ROCm runtime libs it dlopens
# (libamd_smi, libamdhip64).
#
# Everything installs into the self-contained prefix /opt/gpushare:
#   /opt/gpushare/pkgs      python packages (pip --target; PYTHONPATH)
#   /opt/gpushare/pkgs/bin  console scripts (PATH)
# so the runtime stage never depends on the distro's dist-packages layout
# (round-1 bug: COPY to an unversioned /usr/local/lib/python3/dist-packages
# that is not on sys.path).
#
# Smoke: `docker run --rm <img> amdgpushare-device-plugin --selftest` runs
# a full register → ListAndWatch → Allocate pass on mock devices with an
# in-process stub kubelet — no GPU, no cluster — and is wired as the
# HEALTHCHECK below. tests/test_daemon_e2e.py::test_dockerfile_contract
# statically cross-checks this file against setup.py on every CI run.

FROM rocm/dev-ubuntu-22.04:7.2 AS build
WORKDIR /src
COPY . .
ENV PYTORCH_ROCM_ARCH=gfx950
RUN python3 -m pip install --no-cache-dir pybind11 && \
    python3 -m gpushare_amd.native.build && \
    python3 -m pip install --no-cache-dir --target /opt/gpushare/pkgs . && \
    python3 -m pip install --no-cache-dir --target /opt/gpushare/pkgs \
        grpcio protobuf pyyaml prometheus_client
# fail the BUILD if any expected console script is missing
RUN for s in amdgpushare-device-plugin gpushare-scheduler-extender \
             gpushare-top kubectl-inspect-gpushare gpushare-podgetter; do \
        test -x /opt/gpushare/pkgs/bin/$s || { echo "missing $s"; exit 1; }; \
    done && \
    PYTHONPATH=/opt/gpushare/pkgs PATH=/opt/gpushare/pkgs/bin:$PATH \
        amdgpushare-device-plugin --selftest

FROM rocm/rocm-terminal:7.2
COPY --from=build /opt/gpushare /opt/gpushare
ENV PYTHONPATH=/opt/gpushare/pkgs \
    PATH=/opt/gpushare/pkgs/bin:${PATH} \
    HSA_ENABLE_IPC_MODE_LEGACY=0
HEALTHCHECK --interval=60s --timeout=30s \
    CMD amdgpushare-device-plugin --selftest || exit 1
ENTRYPOINT ["amdgpushare-device-plugin"]
