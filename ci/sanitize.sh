#!/bin/bash
# ASAN sanitizer lane over the host-native code (the MI355X analog of the
# reference's compute-sanitizer-wrapped JVM profile, pom.xml:239-283).
# Rebuilds src/host with AddressSanitizer into .build/asan/_host_asan.so
# and runs the host-native test tier against it.
set -e
cd "$(dirname "$0")/.."
python build_native.py --asan-host
ASAN_LIB=$(g++ -print-file-name=libasan.so)
ASAN_OPTIONS=detect_leaks=0:abort_on_error=1 \
LD_PRELOAD="$ASAN_LIB" \
SRJ_HOST_SO="$PWD/.build/asan/_host_asan.so" \
python -m pytest tests/test_resource_adaptor.py tests/test_kudo.py \
    tests/test_parquet.py tests/test_tz.py -q -m "not gpu" "$@"
