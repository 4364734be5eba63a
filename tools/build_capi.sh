#!/bin/bash
# Build libdlaf_c.so (C ABI over dlaf_amd via embedded Python).
set -e
cd "$(dirname "$0")/.."
PYINC=$(python3 -c "import sysconfig; print(sysconfig.get_paths()['include'])")
PYLIBDIR=$(python3 -c "import sysconfig; print(sysconfig.get_config_var('LIBDIR'))")
PYLIB=$(python3 -c "import sysconfig; print(sysconfig.get_config_var('LDLIBRARY').replace('lib','',1).replace('.so','').replace('.a',''))")
PB11=$(python3 -c "import pybind11; print(pybind11.get_include())")
g++ -O2 -shared -fPIC -std=c++17 csrc/capi/dlaf_c.cpp \
    -I"$PYINC" -I"$PB11" -Iinclude \
    -L"$PYLIBDIR" -l"$PYLIB" -ldl \
    -o libdlaf_c.so
echo "built libdlaf_c.so"
