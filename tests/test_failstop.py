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


def test_failstop_world2_kills_surviving_worker(tmp_path):
    """World 2: only worker 1 dies; the master must detect it, kill the
    still-running worker 0's process group and exit non-zero."""
    ri = tmp_path / 'resource_info'
    ri.write_text('127.0.0.1:0,1\n')
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
    assert 'stopping the job' in (proc.stdout + proc.stderr)
