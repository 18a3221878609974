#!/bin/bash
# ThreadSanitizer sweep of the C++ control plane (the
# reference relies on Rust ownership; the native C++ here gets TSAN).
#
# Builds a TSAN-instrumented hypha_amd/_core.so into a scratch dir, swaps
# it in, runs the threading-heavy suites under libtsan, restores the real
# module, and summarizes reports. glibc's _dl_deallocate_tls report is a
# known false positive with detached threads in dlopen'd modules.
#
# Usage: bash tools/tsan_check.sh [pytest-args...]
set -u
cd "$(dirname "$0")/.."
ARGS=${@:-tests/test_network.py tests/test_tls.py -q}

SCRATCH=$(mktemp -d)
trap 'if [ -f "$SCRATCH/_core_real.so" ]; then cp "$SCRATCH/_core_real.so" hypha_amd/_core.so; fi; echo "[tsan_check] real module restored"' EXIT

echo "[tsan_check] building instrumented _core.so ..."
g++ -O1 -g -std=c++17 -fPIC -shared -pthread -fsanitize=thread \
  -DTORCH_EXTENSION_NAME=_core \
  -Icpp/include \
  -I"$(python3 -c 'import pybind11; print(pybind11.get_include())')" \
  -I"$(python3 -c 'import sysconfig; print(sysconfig.get_paths()["include"])')" \
  cpp/src/net.cpp cpp/bindings/core_bindings.cpp -lssl -lcrypto \
  -o "$SCRATCH/_core.so" || exit 1

LIBTSAN=$(ldconfig -p | awk '/libtsan\.so/{print $NF; exit}')
[ -z "$LIBTSAN" ] && { echo "libtsan not found"; exit 1; }

cp hypha_amd/_core.so "$SCRATCH/_core_real.so"
cp "$SCRATCH/_core.so" hypha_amd/_core.so

echo "[tsan_check] running: pytest $ARGS (expect ~20x slowdown)"
TSAN_OPTIONS="log_path=$SCRATCH/tsan_report" \
  LD_PRELOAD="$LIBTSAN" python3 -m pytest $ARGS
rc=$?

echo "[tsan_check] report summaries:"
grep -h "SUMMARY: ThreadSanitizer" "$SCRATCH"/tsan_report.* 2>/dev/null | sort | uniq -c \
  || echo "  (none - clean)"
exit $rc
