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


def _run_driver(path, extra, tmp_path, timeout=420):
    import subprocess
    import sys
    env = dict(os.environ)
    for k in ('WORLD_SIZE', 'RANK', 'LOCAL_RANK', 'MASTER_ADDR',
              'MASTER_PORT', 'PARALLAX_RESOURCE_INFO'):
        env.pop(k, None)
    ri = tmp_path / 'resource_info'
    ri.write_text('127.0.0.1:\n')
    env['PYTHONPATH'] = REPO
    out = subprocess.run(
        [sys.executable, path, '--resource_info', str(ri),
         '--max_steps', '2'] + extra,
        capture_output=True, text=True, timeout=timeout, env=env,
        cwd=REPO)
    assert out.returncode == 0, (path, out.stderr[-3000:])
    return out


def test_lm1b_driver_smoke(tmp_path):
    """Reference-parity driver smoke (lm1b_distributed_driver.py
    synthetic mode) through the real master->worker launch path."""
    _run_driver(os.path.join(REPO, 'examples', 'lm1b',
                             'lm1b_distributed_driver.py'),
                ['--batch_size', '8', '--num_steps', '4',
                 '--vocab_size', '1000', '--log_frequency', '1'],
                tmp_path)


def test_skip_thoughts_driver_smoke(tmp_path):
    _run_driver(os.path.join(REPO, 'examples', 'skip_thoughts',
                             'skip_thoughts_driver.py'),
                ['--batch_size', '4', '--seq_len', '6',
                 '--log_frequency', '1'], tmp_path)


def test_cnn_driver_smoke(tmp_path):
    _run_driver(os.path.join(REPO, 'examples', 'cnn',
                             'cnn_distributed_driver.py'),
                ['--model', 'trivial', '--batch_size', '4',
                 '--log_frequency', '1'], tmp_path)


def test_nmt_driver_smoke(tmp_path):
    _run_driver(os.path.join(REPO, 'examples', 'nmt',
                             'nmt_distributed_driver.py'),
                ['--batch_size', '4', '--src_len', '6', '--tgt_len', '6',
                 '--log_frequency', '1'], tmp_path)


def test_simple_driver_world2_spawn(tmp_path):
    """The REAL master spawner launching TWO workers (gloo on CPU):
    resource_info with two device slots -> two spawned processes through
    the env-var role protocol, loss converges, master exits 0."""
    import subprocess
    import sys
    env = dict(os.environ)
    for k in ('WORLD_SIZE', 'RANK', 'LOCAL_RANK', 'MASTER_ADDR',
              'MASTER_PORT', 'PARALLAX_RESOURCE_INFO'):
        env.pop(k, None)
    env['PYTHONPATH'] = REPO
    res = tmp_path / 'resource_info'
    res.write_text('127.0.0.1:0,1\n')
    out = subprocess.run(
        [sys.executable,
         os.path.join(REPO, 'examples', 'simple', 'simple_driver.py'),
         '--steps', '8', '--resource_info', str(res)],
        capture_output=True, text=True, timeout=420, env=env, cwd=REPO)
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
    assert 'world=2' in out.stdout + out.stderr
    assert 'final loss' in out.stdout
