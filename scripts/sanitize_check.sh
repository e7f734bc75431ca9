#!/bin/bash
# Sanitizer job for the first-party C++ runtime (bus / store / wfdb):
# rebuilds the pybind extensions with AddressSanitizer (or ThreadSanitizer
# with SAN=thread) and runs the concurrency-heavy test files under the
# sanitizer runtime. The reference has no race detection at all (SURVEY.md
# §5); this is the CI job our lock-free/mmap code requires.
#
# Usage: scripts/sanitize_check.sh [thread|address]
set -euo pipefail
cd "$(dirname "$0")/.."

SAN="${1:-address}"
GCCDIR=$(dirname "$(gcc -print-file-name=libasan.so)")
case "$SAN" in
  address) LIB="$GCCDIR/libasan.so"; FLAG=-fsanitize=address ;;
  thread)  LIB="$GCCDIR/libtsan.so"; FLAG=-fsanitize=thread ;;
  *) echo "unknown sanitizer $SAN"; exit 2 ;;
esac
# libstdc++ must be preloaded too or the sanitizer's __cxa_throw
# interceptor cannot resolve the real symbol (C++ exceptions from the
# modules would abort the run)
LIB="$LIB $(g++ -print-file-name=libstdc++.so.6)"

PYINC=$(python3 -c "import sysconfig; print(sysconfig.get_paths()['include'])")
PBINC=$(python3 -c "import pybind11; print(pybind11.get_include())")
EXT=$(python3 -c "import sysconfig; print(sysconfig.get_config_var('EXT_SUFFIX'))")

SANDIR=build/sanitize
mkdir -p "$SANDIR"
for mod in bus store wfdb; do
  g++ -O1 -g -std=c++17 -shared -fPIC "$FLAG" -fno-omit-frame-pointer \
      -I"$PBINC" -I"$PYINC" "tskd_amd/csrc/$mod.cpp" \
      -o "$SANDIR/_tskd_${mod}${EXT}" -lpthread
done

# Point the package at the sanitized builds via a shadow package dir.
rm -rf "$SANDIR/tskd_amd"
mkdir -p "$SANDIR/tskd_amd"
for f in tskd_amd/*.py tskd_amd/*/; do cp -r "$f" "$SANDIR/tskd_amd/"; done
cp "$SANDIR"/_tskd_*"$EXT" "$SANDIR/tskd_amd/"

echo "== running bus/store/wfdb tests under ${SAN} sanitizer =="
KFILT=""
if [ "$SAN" = thread ]; then
  # TSan is ~20x slower and does not model cross-PROCESS mmap'd mutexes;
  # scope it to the in-process concurrency tests (ASan covers the rest).
  # Retention tests excluded: destructive trim racing readers is the
  # DOCUMENTED semantics (reader discards + skips) — TSan reports the
  # intentional unsynchronized zeroing.
  KFILT='-k not grow_beyond and not batch_and_growth and not two_producer and not multiprocess and not cohort and not trim and not gap'
fi
env LD_PRELOAD="$LIB" \
    ASAN_OPTIONS=detect_leaks=0:abort_on_error=1 \
    TSAN_OPTIONS=halt_on_error=1 \
    PYTHONPATH="$SANDIR" \
    python3 -m pytest tests/test_bus.py tests/test_store_wfdb.py -q -x ${KFILT:+"$KFILT"}
echo "== ${SAN} sanitizer clean =="
