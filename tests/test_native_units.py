"""Run the native C++ unit-test binary (TPS math, selection, admission,
HTTP/chunked/SSE parsing, JSON, crypto) from pytest."""
import subprocess
from pathlib import Path

REPO = Path(__file__).resolve().parents[1]


def test_native_unit_suite():
    binary = REPO / "gateway" / "bin" / "llmlb-test"
    if not binary.exists():
        subprocess.run(["make", "-C", str(REPO / "gateway"), "-j", "4"],
                       check=True, stdout=subprocess.DEVNULL)
    res = subprocess.run([str(binary)], capture_output=True, text=True,
                         timeout=60)
    assert res.returncode == 0, res.stderr + res.stdout
