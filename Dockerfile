# amd-dra-driver image: all components (kubelet plugins, controller, webhook,
# compute-domain daemon supervisor + native fabricd, HIP probes for gfx950).
# Base must provide ROCm >= 7.0 userspace + PyTorch-ROCm.
FROM rocm/pytorch:latest

WORKDIR /opt/amd-dra-driver

COPY k8s_dra_driver_gpu_amd/ k8s_dra_driver_gpu_amd/
COPY native/ native/
COPY deployments/ deployments/
COPY bench.py __graft_entry__.py ./

# build the CDNA4 probe library (gfx950) and the C++ fabric daemon
# (fabricd links OpenSSL for the optional mTLS peer-mesh mode)
RUN apt-get update && apt-get install -y --no-install-recommends libssl-dev \
    && rm -rf /var/lib/apt/lists/* \
    && python -m k8s_dra_driver_gpu_amd.ops.build && make -C native -j4

ENV PYTHONPATH=/opt/amd-dra-driver \
    FABRICD_PATH=/opt/amd-dra-driver/native/bin/fabricd \
    FABRICCTL_PATH=/opt/amd-dra-driver/native/bin/fabricctl

# default: the GPU kubelet plugin; other components override the command
# (see deployments/helm/amd-dra-driver/templates/*.yaml)
CMD ["python", "-m", "k8s_dra_driver_gpu_amd.cmd.gpu_kubelet_plugin"]
