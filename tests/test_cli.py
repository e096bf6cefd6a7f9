"""CLI tests (python -m spark_tfrecord_amd)."""

import json
import os
import subprocess
import sys

import numpy as np

import spark_tfrecord_amd as stf


def run_cli(*args):
    return subprocess.run(
        [sys.executable, "-m", "spark_tfrecord_amd", "--engine", "cpu", *args],
        capture_output=True, text=True,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


class TestCli:
    def test_count_schema_head_validate_convert(self, tmp_sandbox):
        out = str(tmp_sandbox / "cli")
        stf.write_tfrecord({"a": np.arange(25, dtype=np.int64),
                            "b": [f"v{i}" for i in range(25)]},
                           out, engine="cpu")
        r = run_cli("count", out)
        assert r.returncode == 0 and r.stdout.strip() == "25"
        r = run_cli("schema", out)
        assert r.returncode == 0 and "a" in r.stdout and "b" in r.stdout
        r = run_cli("head", out, "-n", "2")
        lines = [json.loads(x) for x in r.stdout.strip().splitlines()]
        assert len(lines) == 2 and lines[0]["a"] == 0
        r = run_cli("validate", out)
        assert r.returncode == 0 and json.loads(r.stdout)["ok"]
        dst = str(tmp_sandbox / "cli_gz")
        r = run_cli("convert", out, dst, "--codec", "gzip")
        assert r.returncode == 0, r.stderr
        assert stf.count_tfrecord(dst, engine="cpu") == 25
        assert any(f.endswith(".gz") for f in os.listdir(dst))
