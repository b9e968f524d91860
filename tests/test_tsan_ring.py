"""ThreadSanitizer pass over the shm transport protocols.

Compiles tests/tsan/ring_tsan.cc — the C++ re-statement of the SPSC
trajectory ring (parallel/queue.py) and the seqlock weight publication
(parallel/weights.py) with the x86-TSO publication ordering mapped to
release/acquire — and runs it under TSan. A race report or a
checksum/torn-snapshot violation fails the test.
"""

import shutil
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parent.parent


def _tsan_available() -> bool:
    if shutil.which("g++") is None:
        return False
    probe = subprocess.run(
        ["g++", "-fsanitize=thread", "-x", "c++", "-", "-o", "/dev/null"],
        input=b"int main(){return 0;}", capture_output=True)
    return probe.returncode == 0


@pytest.mark.skipif(not _tsan_available(),
                    reason="g++ -fsanitize=thread unavailable")
def test_ring_and_seqlock_tsan_clean(tmp_path):
    exe = tmp_path / "ring_tsan"
    build = subprocess.run(
        ["g++", "-std=c++17", "-O1", "-g", "-fsanitize=thread",
         str(ROOT / "tests" / "tsan" / "ring_tsan.cc"), "-o", str(exe),
         "-lpthread"], capture_output=True, text=True)
    assert build.returncode == 0, build.stderr
    run = subprocess.run([str(exe)], capture_output=True, text=True,
                         env={"TSAN_OPTIONS": "halt_on_error=1",
                              "PATH": "/usr/bin:/bin"}, timeout=120)
    sys.stderr.write(run.stderr)
    assert run.returncode == 0, (run.stdout, run.stderr)
    assert "OK" in run.stdout
