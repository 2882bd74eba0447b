#!/bin/bash
# ASan + TSan builds of the native control plane under client-thread
# stress (SURVEY §5 race-detection CI).  The driver includes
# control_plane.cc directly; pybind headers resolve via python3 but no
# interpreter is started (the Python-fallback path stays unarmed).
set -e
cd "$(dirname "$0")/../.."
OUT="${1:-build/sanitize}"
mkdir -p "$OUT"
PYINC=$(python3 -c "import sysconfig; print(sysconfig.get_paths()['include'])")
PYBINC=$(python3 -c "import pybind11; print(pybind11.get_include())")
PYLIB=$(python3-config --ldflags --embed 2>/dev/null || python3-config --ldflags)
# ROCm's clang: its TSan runtime intercepts pthread_cond_clockwait
# (libstdc++'s condition_variable::wait_for); gcc-11's libtsan does
# not, which fabricates "double lock" / race reports around every
# barrier wait.
CXX=${CXX:-/opt/rocm/lib/llvm/bin/clang++}
for SAN in address thread; do
  BIN="$OUT/control_stress_$SAN"
  $CXX -std=c++17 -g -O1 -fsanitize=$SAN -fno-omit-frame-pointer \
    -I"$PYINC" -I"$PYBINC" tools/sanitize/control_stress.cc \
    -o "$BIN" $PYLIB -lpthread
  echo "== $SAN =="
  if [ "$SAN" = thread ]; then
    TSAN_OPTIONS="halt_on_error=1" "$BIN"
  else
    ASAN_OPTIONS="detect_leaks=0" "$BIN"
  fi
done
echo "sanitize OK"
