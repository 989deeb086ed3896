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
  python -m pytest tests/test_hostsim.py tests/test_hostsim_fuzz.py -q -p no:cacheprovider "$@"
# restore the normal build
unset GGRMCP_HOSTSIM_FLAGS
python -c "from ggrmcp_amd.ops import build; build.build_hostsim(force=True)"
