#!/bin/bash
# Sanitizer sweep of the C++ control plane (the reference
# relies on Rust ownership; the native C++ here gets TSAN/ASAN).
#
# Builds an instrumented hypha_amd/_core.so into a scratch dir, swaps it
# in, runs the threading-heavy suites under the sanitizer runtime,
# restores the real module, and summarizes reports.
#
# Usage:
#   bash tools/tsan_check.sh [--asan] [pytest-args...]
#
# Modes:
#   default (TSAN): ~20x slowdown. Found and fixed (round 2): four
#     teardown fd races (close-while-blocked-reader), a gw_host_
#     failover string race, and bounded-drain lifetime hazards.
#   --asan: ~2x slowdown, catches use-after-free/overflow without TSAN's
#     mutex-identity weakness. The full network+TLS suites run clean.
#
# KNOWN FALSE POSITIVES (do not chase):
#   * glibc `_dl_deallocate_tls` races — detached threads in dlopen'd
#     modules.
#   * TSAN "double lock of a mutex ... already destroyed" /
#     shared_ptr races around Gateway::relay_* — gcc-11 libtsan keeps a
#     stale sync object when a heap block holding a trivially-initialized
#     std::mutex is freed and the address is reused by a NEW object
#     (std::mutex never calls pthread_mutex_init, so TSAN cannot see the
#     re-initialization). The same paths are ASAN-clean, confirming no
#     real lifetime bug: run with --asan to discriminate.
set -u
cd "$(dirname "$0")/.."

MODE=tsan
if [ "${1:-}" = "--asan" ]; then MODE=asan; shift; fi
ARGS=${@:-tests/test_network.py tests/test_tls.py -q}

SCRATCH=$(mktemp -d)
restore() {
  if [ -f "$SCRATCH/_core_real.so" ]; then
    cp "$SCRATCH/_core_real.so" hypha_amd/_core.so
    echo "[tsan_check] real module restored"
  fi
}
trap restore EXIT INT TERM

echo "[tsan_check] building $MODE-instrumented _core.so ..."
g++ -O1 -g -std=c++17 -fPIC -shared -pthread -fsanitize=$([ $MODE = tsan ] && echo thread || echo address) \
  -DTORCH_EXTENSION_NAME=_core \
  -Icpp/include \
  -I"$(python3 -c 'import pybind11; print(pybind11.get_include())')" \
  -I"$(python3 -c 'import sysconfig; print(sysconfig.get_paths()["include"])')" \
  cpp/src/net.cpp cpp/bindings/core_bindings.cpp -lssl -lcrypto \
  -o "$SCRATCH/_core.so" || exit 1

if [ $MODE = tsan ]; then
  LIBSAN=$(ldconfig -p | awk '/libtsan\.so/{print $NF; exit}')
  PRELOAD="$LIBSAN"
  OPTVAR=TSAN_OPTIONS
else
  LIBSAN=$(ldconfig -p | awk '/libasan\.so/{print $NF; exit}')
  # libstdc++ must be preloaded too or ASAN's __cxa_throw interception
  # fails ("real___cxa_throw != 0" CHECK) on the first C++ exception
  PRELOAD="$LIBSAN $(ldconfig -p | awk '/libstdc\+\+\.so.6 /{print $NF; exit}')"
  OPTVAR=ASAN_OPTIONS
fi
[ -z "$LIBSAN" ] && { echo "sanitizer runtime not found"; exit 1; }

cp hypha_amd/_core.so "$SCRATCH/_core_real.so"
cp "$SCRATCH/_core.so" hypha_amd/_core.so

echo "[tsan_check] running: pytest $ARGS ($MODE)"
env "$OPTVAR=detect_leaks=0 log_path=$SCRATCH/report" \
  LD_PRELOAD="$PRELOAD" python3 -m pytest $ARGS
rc=$?

echo "[tsan_check] report summaries:"
grep -h -e "SUMMARY: ThreadSanitizer" -e "ERROR: AddressSanitizer" \
  "$SCRATCH"/report.* 2>/dev/null | sort | uniq -c \
  || echo "  (none - clean)"
exit $rc
