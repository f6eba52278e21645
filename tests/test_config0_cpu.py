"""BASELINE.json configs[0] plumbing driver (tools/config0_driver.py): the
db_bench-style fillrandom+compact flow through the worker boundary on CPU,
with the newest-version view verified."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_config0_fillrandom_compact(tmp_path):
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "config0_driver.py"),
         "--keys", "60000", "--memtable-entries", "10000", "--verify",
         "--workdir", str(tmp_path / "w")],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["verified"] is True
    assert r["l0_runs"] == 6
    assert r["l2_entries"] > 0 and r["l2_files"] >= 1
