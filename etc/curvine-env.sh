#!/bin/bash
# Environment defaults for curvine-amd daemons (etc/curvine-env.sh analog).
# Source this before launching cv-server / cv-fuse.

export CURVINE_HOME="${CURVINE_HOME:-$(cd "$(dirname "${BASH_SOURCE[0]}")"/..; pwd)}"

# master / worker / client bind hostnames (env overlay over the TOML)
export CURVINE_MASTER_HOSTNAME="${CURVINE_MASTER_HOSTNAME:-127.0.0.1}"
export CURVINE_WORKER_HOSTNAME="${CURVINE_WORKER_HOSTNAME:-127.0.0.1}"
export CURVINE_CLIENT_HOSTNAME="${CURVINE_CLIENT_HOSTNAME:-127.0.0.1}"

export CURVINE_CONF="${CURVINE_CONF:-$CURVINE_HOME/etc/curvine-cluster.toml}"

# ROCm runtime: dmabuf IPC is the only mode the MI355X pool's host
# driver supports — required for RCCL / cross-process device memory
export HSA_ENABLE_IPC_MODE_LEGACY=0

# gfx950 target for any JIT extension builds
export PYTORCH_ROCM_ARCH="${PYTORCH_ROCM_ARCH:-gfx950}"
