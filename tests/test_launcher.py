"""Integration test of the MASTER launch path: the simple example driver
run as a subprocess — master parses resource_info, spawns one (CPU)
worker re-executing the driver, worker trains and exits."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
DRIVER = os.path.join(REPO, 'examples', 'simple', 'simple_driver.py')


def test_simple_driver_end_to_end(tmp_path):
    ri = tmp_path / 'resource_info'
    ri.write_text('127.0.0.1:\n')
    env = dict(os.environ)
    for var in ('WORLD_SIZE', 'RANK', 'LOCAL_RANK', 'PARALLAX_RUN_OPTION'):
        env.pop(var, None)
    env['PYTHONPATH'] = REPO
    proc = subprocess.run(
        [sys.executable, DRIVER, '--steps', '60',
         '--resource_info', str(ri)],
        env=env, capture_output=True, text=True, timeout=240)
    assert proc.returncode == 0, proc.stderr[-3000:]
    assert 'final loss:' in proc.stdout
    final = float(proc.stdout.strip().rsplit('final loss:', 1)[1])
    assert final < 1.0  # loss decreased from the random-init value
