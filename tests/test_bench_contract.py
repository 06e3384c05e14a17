"""bench.py driver-contract test: one JSON line with the exact keys,
honest dtype/config, N=1 default (the driver parses this line verbatim;
a malformed line would invalidate the round's BENCH evidence)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {'metric', 'value', 'unit', 'n_gpus', 'steps', 'warmup',
            'ms_per_step', 'higher_is_better', 'scaling', 'vs_baseline',
            'dtype', 'data', 'config'}


def _run_bench(*extra):
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, 'bench.py'),
         '--model', 'resnet50', '--batch-size', '4',
         '--steps', '2', '--warmup', '1', '--no-companion'] + list(extra),
        capture_output=True, text=True, timeout=600,
        env=dict(os.environ, PYTHONPATH=REPO))
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.strip().splitlines()
             if ln.startswith('{')]
    assert len(lines) == 1, out.stdout   # exactly ONE JSON line
    return json.loads(lines[0])


def test_bench_json_contract_cpu():
    r = _run_bench()
    assert REQUIRED <= set(r)
    assert r['metric'] == 'ResNet-50 images/sec'
    assert r['unit'] == 'images/s'
    assert r['n_gpus'] == 1 and r['steps'] == 2 and r['warmup'] == 1
    assert r['higher_is_better'] is True
    assert r['scaling'] == 'weak'
    assert r['data'] == 'synthetic'
    assert r['dtype'] == 'fp32'          # honest: no GPU -> no bf16
    assert r['config']['precision'] == 'fp32'
    assert r['config']['global_batch'] == 4
    assert r['value'] > 0 and r['ms_per_step'] > 0
    # value consistency: value == batch * steps / elapsed
    implied = 4 * 1000.0 / r['ms_per_step']
    assert abs(implied - r['value']) / r['value'] < 0.01
    assert r['vs_baseline'] == round(r['value'] / 7520.0, 4)


def test_bench_json_contract_n2_gloo():
    """The driver's SCALE invocation shape: torch.distributed.run with 2
    ranks (gloo on CPU) — exactly one JSON line from rank 0 with
    n_gpus=2, the MAX-over-ranks elapsed path exercised."""
    from parallax_amd.lib import get_free_port
    out = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', str(get_free_port()),
         os.path.join(REPO, 'bench.py'), '--gpus', '2',
         '--model', 'resnet50', '--batch-size', '2',
         '--steps', '2', '--warmup', '1', '--no-companion'],
        capture_output=True, text=True, timeout=600,
        env=dict(os.environ, PYTHONPATH=REPO))
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.strip().splitlines()
             if ln.startswith('{')]
    assert len(lines) == 1, out.stdout
    r = json.loads(lines[0])
    assert r['n_gpus'] == 2
    assert r['config']['global_batch'] == 4
    assert r['config']['parallelism'] == 'dp2 (AR)'
