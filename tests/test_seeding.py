"""stable_seed must be identical across processes (hash() is not)."""

import subprocess
import sys


def _derive(hashseed):
    code = (
        "from dts_amd.utils.seeding import stable_seed;"
        "print(stable_seed(7, 'judge', 3), stable_seed(1, 'user', 4, 'hi'))"
    )
    out = subprocess.run(
        [sys.executable, "-c", code],
        capture_output=True,
        text=True,
        env={"PYTHONHASHSEED": hashseed, "PATH": "/usr/bin:/bin"},
        cwd=".",
    )
    assert out.returncode == 0, out.stderr
    return out.stdout.strip()


def test_stable_across_hash_randomization():
    assert _derive("1") == _derive("2")
