"""End-to-end partition-search harness test (reference SURVEY §3.5): the
master restarts the job per trial partition count, collects the stat-window
iteration time from every worker over the BaseManager queue, walks p, fits
the cost model and relaunches the final job."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
DRIVER = os.path.join(REPO, 'tests', 'search_driver.py')


def test_partition_search_end_to_end(tmp_path):
    ri = tmp_path / 'resource_info'
    ri.write_text('127.0.0.1:\n')
    env = dict(os.environ)
    for var in ('WORLD_SIZE', 'RANK', 'LOCAL_RANK', 'PARALLAX_RUN_OPTION',
                'PARALLAX_PARTITIONS', 'PARALLAX_MIN_PARTITIONS'):
        env.pop(var, None)
    env['PYTHONPATH'] = REPO
    env['PARALLAX_COLLECT_STAT_START'] = '3'
    env['PARALLAX_COLLECT_STAT_END'] = '8'
    proc = subprocess.run(
        [sys.executable, DRIVER, str(ri)], env=env, capture_output=True,
        text=True, timeout=420)
    out = proc.stdout + proc.stderr
    assert proc.returncode == 0, out[-4000:]
    # the search ran at least two trial partition counts…
    assert 'running with 2 partitions' in out, out[-4000:]
    assert 'running with 4 partitions' in out, out[-4000:]
    # …picked an optimum and relaunched the final job to completion
    assert 'optimal partitions:' in out, out[-4000:]
    assert 'done p=' in out, out[-4000:]


def test_partition_search_world2(tmp_path):
    """Search with TWO spawned workers: the master collects one stat
    report per worker per trial and both workers finish the final job."""
    ri = tmp_path / 'resource_info'
    ri.write_text('127.0.0.1:0,1\n')
    env = dict(os.environ)
    for var in ('WORLD_SIZE', 'RANK', 'LOCAL_RANK', 'PARALLAX_RUN_OPTION',
                'PARALLAX_PARTITIONS', 'PARALLAX_MIN_PARTITIONS'):
        env.pop(var, None)
    env['PYTHONPATH'] = REPO
    env['PARALLAX_COLLECT_STAT_START'] = '3'
    env['PARALLAX_COLLECT_STAT_END'] = '8'
    proc = subprocess.run(
        [sys.executable, DRIVER, str(ri)], env=env, capture_output=True,
        text=True, timeout=420)
    out = proc.stdout + proc.stderr
    assert proc.returncode == 0, out[-4000:]
    assert 'optimal partitions:' in out, out[-4000:]
    assert out.count('done p=') == 2, out[-4000:]
