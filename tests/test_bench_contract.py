"""bench.py driver-contract tests: the exact launch pattern the round-end
driver uses (torchrun, one rank per device, JSON line on rank 0), run here
with --device cpu over gloo."""
import json
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO, 'bench.py')


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _run(args, world=None, timeout=240):
    if world:
        cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
               f'--nproc-per-node={world}', '--master-addr', '127.0.0.1',
               '--master-port', str(_free_port()), BENCH, '--gpus', str(world)]
    else:
        cmd = [sys.executable, BENCH]
    r = subprocess.run(cmd + args, capture_output=True, text=True,
                       timeout=timeout, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:] + r.stdout[-500:]
    lines = [l for l in r.stdout.splitlines() if l.startswith('{')]
    assert len(lines) == 1, f"expected ONE JSON line, got {lines}"
    return json.loads(lines[0])


TINY = ['--rows', '4000', '--features', '8', '--n-blocks', '4',
        '--band', '16', '--device', 'cpu', '--no-cpu-baseline',
        '--steps', '2', '--warmup', '1']


def test_single_process_contract():
    d = _run(TINY)
    for key in ('metric', 'value', 'unit', 'n_gpus', 'steps', 'warmup',
                'ms_per_step', 'higher_is_better', 'scaling', 'vs_baseline',
                'dtype', 'data', 'config'):
        assert key in d, key
    assert d['n_gpus'] == 1 and d['data'] == 'synthetic'
    assert d['config']['workload'].startswith('cfg4_synth')


def test_torchrun_world2_contract():
    d = _run(TINY, world=2)
    assert d['n_gpus'] == 2
    assert d['value'] > 0


def test_torchrun_world2_multi_part():
    d = _run(TINY + ['--parts', '2'], world=2)
    assert d['config']['parts'] == 2
    assert d['value'] > 0
