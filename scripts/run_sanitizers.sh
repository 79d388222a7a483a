#!/usr/bin/env bash
# Repeatable ASAN + TSAN pass over the native server (VERDICT r1 next #6:
# "make sanitizers repeatable ... keep the artifact").
#
# Rebuilds _fastserver.so and _h2tool.so with the requested sanitizer,
# runs the full native-server test matrix (conformance + stress + fuzz +
# wire-fuzz) with the sanitizer runtime preloaded into CPython, saves the
# log under profiles/sanitizers/, and restores clean builds afterwards.
#
# Usage: scripts/run_sanitizers.sh [asan|tsan|all]
set -u

cd "$(dirname "$0")/.."
REPO="$PWD"
OUTDIR="$REPO/profiles/sanitizers"
mkdir -p "$OUTDIR"
NATIVE="$REPO/k8s_device_plugin_amd/native"

TESTS="tests/test_fastserver.py tests/test_fastserver_stress.py \
tests/test_fastserver_fuzz.py tests/test_fastserver_wirefuzz.py \
tests/test_grpcgo_conformance.py"

PYINC=$(python3 - <<'EOF'
import sysconfig, pybind11
print(f"-I{sysconfig.get_paths()['include']} -I{pybind11.get_include()}")
EOF
)

build_with() {
    local flags="$1"
    g++ -O1 -g -std=c++17 $flags -shared -fPIC \
        "$NATIVE/fastserver.cpp" -o "$NATIVE/_fastserver.so" -ldl -pthread \
        $PYINC || return 1
    g++ -O1 -g -std=c++17 $flags -shared -fPIC \
        "$NATIVE/h2tool.cpp" -o "$NATIVE/_h2tool.so" -ldl $PYINC || return 1
}

run_one() {
    local name="$1" flags="$2" preload="$3" opts_var="$4" opts_val="$5"
    local log="$OUTDIR/${name}_fastserver.log"
    echo "=== $name: rebuilding native server with $flags ==="
    if ! build_with "$flags"; then
        echo "$name build FAILED" | tee "$log"
        return 1
    fi
    echo "=== $name: running native-server matrix ==="
    {
        echo "# $name pass over the native server test matrix"
        echo "# date: $(date -u +%Y-%m-%dT%H:%M:%SZ)"
        echo "# compiler: $(g++ --version | head -1)"
        echo "# flags: $flags"
        echo "# tests: $TESTS"
        echo
    } > "$log"
    LD_PRELOAD="$preload" \
        PYTHONMALLOC=malloc \
        AMDXDP_SANITIZER="$name" \
        env "$opts_var=$opts_val" \
        timeout 1200 python3 -m pytest $TESTS -q -p no:cacheprovider \
        >> "$log" 2>&1
    local rc=$?
    echo >> "$log"
    echo "# exit code: $rc" >> "$log"
    if [ $rc -ne 0 ]; then
        echo "$name FAILED (rc=$rc) — see $log"
        tail -40 "$log"
        return $rc
    fi
    # a sanitizer failure may abort the process yet pytest still exits 0
    # on partial runs; double-check the log for reports
    if grep -qE "ERROR: (Address|Thread)Sanitizer|WARNING: ThreadSanitizer" "$log"; then
        echo "$name: sanitizer reports found — see $log"
        grep -E "ERROR: (Address|Thread)Sanitizer|WARNING: ThreadSanitizer" "$log" | head
        return 2
    fi
    echo "$name PASS — log kept at $log"
}

MODE="${1:-all}"
FAIL=0

ASAN_SO=$(g++ -print-file-name=libasan.so)
TSAN_SO=$(g++ -print-file-name=libtsan.so)

if [ "$MODE" = "asan" ] || [ "$MODE" = "all" ]; then
    run_one asan "-fsanitize=address -fno-omit-frame-pointer" "$ASAN_SO" \
        ASAN_OPTIONS "detect_leaks=0:abort_on_error=1:strict_string_checks=1" \
        || FAIL=1
fi
if [ "$MODE" = "tsan" ] || [ "$MODE" = "all" ]; then
    # TSAN sees only our instrumented .so plus intercepted pthread/libc
    # calls; CPython itself is uninstrumented.  history_size raised for
    # long-lived server threads.
    run_one tsan "-fsanitize=thread -fno-omit-frame-pointer" "$TSAN_SO" \
        TSAN_OPTIONS "halt_on_error=0:exitcode=0:history_size=4:suppressions=$REPO/scripts/tsan.supp:log_path=stderr" \
        || FAIL=1
fi

echo "=== restoring clean native builds ==="
python3 -m k8s_device_plugin_amd.native.build --force >/dev/null

exit $FAIL
