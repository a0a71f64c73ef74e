"""The driver depends on bench.py's JSON contract (BASELINE.json metric,
one JSON line on stdout from rank 0). Guard it for every bench mode so a
schedule/solver change can't silently break the round-end measurement.

Runs tiny CPU configs (--cpu) of each mode and validates the JSON keys
and value sanity.
"""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {
    'metric', 'value', 'unit', 'n_gpus', 'steps', 'warmup', 'ms_per_step',
    'higher_is_better', 'scaling', 'vs_baseline', 'dtype', 'data', 'config',
}


def _run_bench(extra, timeout=600):
    cmd = [sys.executable, os.path.join(ROOT, 'bench.py'), '--cpu',
           '--steps', '1', '--warmup', '0'] + extra
    out = subprocess.run(cmd, capture_output=True, text=True, cwd=ROOT,
                         timeout=timeout)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    return json.loads(line)


def _check(d):
    assert REQUIRED_KEYS <= set(d), REQUIRED_KEYS - set(d)
    assert d['metric'] == 'visibilities/sec calibrated'
    assert d['value'] > 0 and d['ms_per_step'] > 0
    assert d['higher_is_better'] is True
    assert d['scaling'] == 'weak'
    assert d['data'] == 'synthetic'
    assert isinstance(d['config'], dict) and 'model' in d['config']


def test_bench_default_contract():
    d = _run_bench(['--stations', '8', '--dirs', '2', '--srcs', '2',
                    '--tilesz', '4', '--chan', '2'])
    _check(d)
    assert d['config']['stations'] == 8


def test_bench_rtr_contract():
    d = _run_bench(['--stations', '48', '--dirs', '2', '--srcs', '2',
                    '--tilesz', '2', '--chan', '2', '--solver', 'rtr',
                    '--no-consensus'])
    _check(d)


def test_bench_bandpass_contract():
    d = _run_bench(['--mode', 'bandpass', '--stations', '8', '--dirs', '2',
                    '--srcs', '2', '--tilesz', '2', '--chan', '16',
                    '--nsolbw', '4'])
    _check(d)


def test_bench_two_rank_launch():
    """Launch bench.py exactly the way the driver's scaling harness does
    (torch.distributed.run, one process per 'GPU', 127.0.0.1 rendezvous)
    with 2 CPU/gloo ranks: rank 0 must print the JSON contract line with
    the whole-job aggregate value."""
    env = dict(os.environ)
    env['SAGECAL_BENCH_BACKEND'] = 'gloo'
    cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
           '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
           '--master-port', '29547', os.path.join(ROOT, 'bench.py'),
           '--gpus', '2', '--cpu', '--steps', '1', '--warmup', '0',
           '--stations', '8', '--dirs', '2', '--srcs', '2',
           '--tilesz', '4', '--chan', '2']
    out = subprocess.run(cmd, capture_output=True, text=True, cwd=ROOT,
                         env=env, timeout=900)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.strip().splitlines()
            if l.startswith('{')][-1]
    d = json.loads(line)
    _check(d)
    assert d['config']['parallelism'].endswith('2')


def test_bench_eight_rank_launch():
    """The driver's 8-GPU scale run shape, on CPU/gloo: 8 ranks, one
    band each, fused consensus all-reduce; rank 0 prints the aggregate
    JSON line."""
    env = dict(os.environ)
    env['SAGECAL_BENCH_BACKEND'] = 'gloo'
    cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
           '--nproc-per-node', '8', '--master-addr', '127.0.0.1',
           '--master-port', '29549', os.path.join(ROOT, 'bench.py'),
           '--gpus', '8', '--cpu', '--steps', '1', '--warmup', '0',
           '--stations', '8', '--dirs', '2', '--srcs', '2',
           '--tilesz', '2', '--chan', '2']
    out = subprocess.run(cmd, capture_output=True, text=True, cwd=ROOT,
                         env=env, timeout=900)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.strip().splitlines()
            if l.startswith('{')][-1]
    d = json.loads(line)
    _check(d)
    assert d['config']['parallelism'].endswith('8')


def test_bench_interval_clamp(monkeypatch):
    """--intervals is clamped to the coherency residency budget for
    very large arrays (guard against OOM at naive 512+/16 settings)."""
    import bench as bench_mod
    import sagecal_amd.msdata as md

    class A:
        pass
    a = A()
    a.__dict__.update(stations=2048, dirs=20, srcs=5, tilesz=60,
                      chan=8, freq0=150e6, bandwidth=180e3,
                      intervals=16, shapelet_dirs=0)
    seen = {}

    def fake_init(self, **kw):
        seen['tilesz'] = kw.get('tilesz')
        raise SystemExit
    monkeypatch.setattr(md.SyntheticMS, '__init__', fake_init)
    try:
        bench_mod.build_problem(a, 'cpu', None)
    except SystemExit:
        pass
    assert a.intervals == 1                  # clamped from 16
    assert seen['tilesz'] == 60              # tilesz * clamped intervals
