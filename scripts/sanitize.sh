#!/bin/bash
# Sanitizer lane (SURVEY §5 race-detection item): builds and runs the
# engine stress binary under TSAN and ASAN. Exit 0 = both clean.
set -e
cd "$(dirname "$0")/.."
python build_ext.py --tsan
TSAN_OPTIONS="halt_on_error=1 second_deadlock_stack=1" timeout 600 ./bin/tsan_stress
python build_ext.py --asan
ASAN_OPTIONS="detect_leaks=0" timeout 600 ./bin/asan_stress
echo "sanitizer lane clean (tsan + asan)"
