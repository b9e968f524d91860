#!/usr/bin/env bash
# ThreadSanitizer pass over the shm transport protocols (SURVEY.md §5.2,
# VERDICT r1 item 9). Builds tests/tsan/ring_tsan.cc — the C++
# re-statement of parallel/queue.py's SPSC ring and parallel/weights.py's
# seqlock — and runs it under TSan. Non-zero exit = a race report or a
# protocol (checksum / torn-snapshot) violation.
set -euo pipefail
cd "$(dirname "$0")/.."
out="${TMPDIR:-/tmp}/ring_tsan"
g++ -std=c++17 -O1 -g -fsanitize=thread tests/tsan/ring_tsan.cc \
    -o "$out" -lpthread
TSAN_OPTIONS="halt_on_error=1" "$out"
