"""Fail-stop semantics (reference SURVEY §5.3): a dying worker brings the
master down with a non-zero exit code."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_master_fails_when_worker_dies(tmp_path):
    ri = tmp_path / 'resource_info'
    ri.write_text('127.0.0.1:\n')
    env = dict(os.environ)
    for var in ('WORLD_SIZE', 'RANK', 'LOCAL_RANK', 'PARALLAX_RUN_OPTION'):
        env.pop(var, None)
    env['PYTHONPATH'] = REPO
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, 'tests', 'crash_driver.py'),
         str(ri)],
        env=env, capture_output=True, text=True, timeout=180)
    assert proc.returncode != 0
    assert 'injected worker failure' in (proc.stdout + proc.stderr)
