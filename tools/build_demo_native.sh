#!/bin/bash
# Build the native C++ demo against torch + HIP (gfx950).
set -e
cd "$(dirname "$0")/.."
TORCH=$(python -c "import torch, os; print(os.path.dirname(torch.__file__))")
PYINC=$(python -c "import sysconfig; print(sysconfig.get_paths()['include'])")
PYLIB=$(python -c "import sysconfig; print(sysconfig.get_config_var('LIBDIR'))")
hipcc --offload-arch=gfx950 -O2 -std=c++17 \
  -D__HIP_PLATFORM_AMD__=1 -DUSE_ROCM=1 -DHIPBLAS_V2 -DCUDA_HAS_FP16=1 \
  -D__HIP_NO_HALF_OPERATORS__=1 -D__HIP_NO_HALF_CONVERSIONS__=1 \
  -DHIP_ENABLE_WARP_SYNC_BUILTINS=1 -DTORCH_API_INCLUDE_EXTENSION_H \
  -DTORCH_EXTENSION_NAME=demo_native \
  -I"$TORCH/include" -I"$TORCH/include/torch/csrc/api/include" -I"$PYINC" \
  tools/demo_native.cpp ddstore_amd/csrc/ddstore_kernels.hip \
  -L"$TORCH/lib" -L/opt/rocm/lib -L"$PYLIB" -lc10 -ltorch -ltorch_cpu -ltorch_python \
  -lc10_hip -ltorch_hip -lamdhip64 -lroctx64 -lpython3.10 \
  -Wl,-rpath,"$TORCH/lib" -Wl,-rpath,/opt/rocm/lib \
  -o tools/demo_native
echo "built tools/demo_native"
