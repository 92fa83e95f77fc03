# selkies-amd: MI355X-native remote desktop streaming framework.
#
# Reference parity: the reference Dockerfile builds its web bundle and
# wheel in stages (SURVEY.md §2.6). Ours is simpler by design — the web
# client is dependency-free static JS (no node build) and the native
# engine is one `make` (hipcc, gfx950) — so one ROCm stage does it all.
#
#   docker build -t selkies-amd .
#   docker run --device=/dev/kfd --device=/dev/dri --ipc=host \
#       -p 8080:8080 selkies-amd
#
# Base: any ROCm >= 7.2 image with hipcc + runtime for gfx950.
ARG ROCM_IMAGE=rocm/dev-ubuntu-22.04:7.2
FROM ${ROCM_IMAGE} AS build

RUN apt-get update && apt-get install -y --no-install-recommends \
        python3-dev python3-pip libx11-dev libxext-dev libxfixes-dev \
        libxtst-dev make g++ && \
    rm -rf /var/lib/apt/lists/*
RUN pip3 install --no-cache-dir pybind11 setuptools wheel

WORKDIR /src
COPY . .
# native engine (HIP kernels for gfx950 + C++ runtime), built in-tree so
# the .so ships inside the wheel via package-data
RUN make -C native -j"$(nproc)" PYTORCH_ROCM_ARCH=gfx950
# joystick interposer + fake-udev shims (LD_PRELOAD addons)
RUN make -C addons/js-interposer && make -C addons/fake-udev
RUN pip3 wheel . --no-build-isolation --no-deps -w /wheels

FROM ${ROCM_IMAGE}
RUN apt-get update && apt-get install -y --no-install-recommends \
        python3-pip libx11-6 libxext6 libxfixes3 libxtst6 \
        x11-xserver-utils && \
    rm -rf /var/lib/apt/lists/*
COPY --from=build /wheels /wheels
RUN pip3 install --no-cache-dir /wheels/*.whl && rm -rf /wheels
COPY --from=build /src/addons/js-interposer/*.so \
                  /src/addons/fake-udev/*.so /opt/selkies/lib/

ENV SELKIES_PORT=8080
EXPOSE 8080
ENTRYPOINT ["selkies"]
