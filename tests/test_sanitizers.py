"""ASan/UBSan coverage of the untrusted-input parsers (SURVEY §5's
sanitizer gap): compiles the torch-free fuzz harness with
-fsanitize=address,undefined and runs structured mutation fuzzing over
the C++ protobuf wire reader and the REST dense-JSON parser. Any OOB
read/write or UB aborts the binary."""
import shutil
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


@pytest.mark.timeout(300)
def test_parser_fuzz_under_asan(tmp_path):
    gxx = shutil.which("g++")
    if gxx is None:
        pytest.skip("no g++")
    binary = tmp_path / "fuzz_parsers"
    build = subprocess.run(
        [gxx, "-std=c++17", "-O1", "-g",
         "-fsanitize=address,undefined", "-fno-sanitize-recover=all",
         "-I", str(REPO / "tfservingcache_amd" / "engine" / "csrc"),
         str(REPO / "scripts" / "fuzz_parsers.cpp"), "-o", str(binary)],
        capture_output=True, text=True, timeout=180)
    assert build.returncode == 0, build.stderr[-3000:]
    run = subprocess.run([str(binary), "20000"], capture_output=True,
                         text=True, timeout=120)
    assert run.returncode == 0, (run.stdout + run.stderr)[-3000:]
    assert "iterations clean" in run.stdout
