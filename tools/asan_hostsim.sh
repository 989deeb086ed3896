#!/bin/bash
# Rebuild the single-lane CPU kernel build with AddressSanitizer and run the
# differential suites under it — memory-safety net for the kernel logic
# (sanitizer tier).  GPU-side the same logic runs bounds-checked
# against arena caps.
set -e
cd "$(dirname "$0")/.."
export GGRMCP_HOSTSIM_FLAGS="-fsanitize=address -fno-omit-frame-pointer -g"
python -c "from ggrmcp_amd.ops import build; build.build_hostsim(force=True)"
ASAN_LIB=$(gcc -print-file-name=libasan.so)
LD_PRELOAD=$ASAN_LIB ASAN_OPTIONS=detect_leaks=0 \
  python -m pytest tests/test_hostsim.py tests/test_hostsim_fuzz.py tests/test_double_exactness.py -q -p no:cacheprovider "$@"
# restore the normal build
unset GGRMCP_HOSTSIM_FLAGS
python -c "from ggrmcp_amd.ops import build; build.build_hostsim(force=True)"

# ---- tier 2: native transport + HTTP frontend under ASAN -------------------
# (pure C++ .so shared between CPU tests and GPU serving; covers the h2
# batch client incl. the per-call response cap, and the epoll reactors
# incl. drain/CORS/header-filter/reuse-port paths)
CXX_ASAN="-O1 -g -std=c++17 -fPIC -shared -fvisibility=hidden \
  -fsanitize=address -fno-omit-frame-pointer \
  -I$(python -c 'import pybind11; print(pybind11.get_include())') \
  -I/usr/include/python3.10 -Iggrmcp_amd/ops/csrc"
TMPD=$(mktemp -d)
cp ggrmcp_amd/ops/_h2grpc.so ggrmcp_amd/ops/_frontend.so "$TMPD/"
restore() { cp "$TMPD/_h2grpc.so" "$TMPD/_frontend.so" ggrmcp_amd/ops/; rm -rf "$TMPD"; }
trap restore EXIT
g++ ggrmcp_amd/ops/csrc/h2grpc.cpp -x c++ $CXX_ASAN \
  -I"${GGRMCP_NGHTTP2_INCLUDE:-/opt/conda/include}" -l:libnghttp2.so.14 \
  -pthread -o ggrmcp_amd/ops/_h2grpc.so
g++ ggrmcp_amd/ops/csrc/frontend.cpp -x c++ $CXX_ASAN -pthread \
  -o ggrmcp_amd/ops/_frontend.so
LD_PRELOAD=$ASAN_LIB ASAN_OPTIONS=detect_leaks=0 \
  python -m pytest tests/test_native_transport.py tests/test_native_frontend.py \
  tests/test_session_table.py tests/test_decode_fallback.py \
  -q -p no:cacheprovider "$@"
