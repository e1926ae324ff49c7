# MI355X agent image — ROCm replaces the reference's 2-stage CUDA 12.1 build
# (reference Dockerfile: nvidia/cuda base, pyenv python, torch cu121, TRT).
# Here: ROCm 7.x base with PyTorch-ROCm; the HIP extension is compiled for
# gfx950 at build time (cross-compiles without a GPU).

FROM rocm/pytorch:latest AS build

WORKDIR /app
COPY setup.py ./
COPY ai_rtc_agent_amd ./ai_rtc_agent_amd
RUN PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

FROM rocm/pytorch:latest

# VCN hardware codec userspace (VA-API): enables the experimental hardware
# encode session (AIRTC_VCN_EXPERIMENTAL=1; docs/webrtc.md). The software
# H.264 codec needs nothing beyond the base image (OpenSSL ships in it for
# the DTLS-SRTP endpoint).
RUN apt-get update && apt-get install -y --no-install-recommends \
        libva2 libva-drm2 mesa-va-drivers && \
    rm -rf /var/lib/apt/lists/* || true

WORKDIR /app
COPY --from=build /app/ai_rtc_agent_amd ./ai_rtc_agent_amd
COPY agent_main.py bench.py build.py download.py ./

# cache layout parity (reference Dockerfile:49-52)
ENV HF_HOME=/models/hf \
    HF_HUB_CACHE=/models/hf/hub \
    ENGINES_CACHE=/models/engines \
    CIVITAI_CACHE=/models/civitai

# hardware codec toggles (reference Dockerfile:54-56 NVENC/NVDEC).
# NOTE: the VCN session additionally requires the explicit
# AIRTC_VCN_EXPERIMENTAL=1 opt-in until hardware-validated; the standard
# software H.264 codec is the default either way (media/codec.py).
ENV VCN_ENC=true \
    VCN_DEC=true

# RCCL/IPC over the host driver (required for multi-process GPU work)
ENV HSA_ENABLE_IPC_MODE_LEGACY=0

EXPOSE 8888
EXPOSE 40000-40100/udp

CMD ["python", "agent_main.py", "--port", "8888", "--udp-ports", "40000-40100"]
