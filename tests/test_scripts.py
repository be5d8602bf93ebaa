"""Offline tooling tests: log parsing, SWA averaging, plot generation."""

import os
import subprocess
import sys
import tempfile

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, 'scripts'))

SAMPLE_LOG = """started server
waiting training
100 200
epoch 0
win rate = Nan (0)
generation stats = Nan (0)
loss = p:-0.242 v:0.250 ent:1.698 total:-0.099
updated model(12)
epoch 1
win rate = 0.538 (7.0 / 13)
generation stats = 0.000 +- 0.975
loss = p:-0.150 v:0.230 ent:1.650 total:-0.080
updated model(25)
epoch 2
win rate (random) = 0.600 (9.0 / 15)
win rate (rulebase) = 0.400 (4.0 / 10)
generation stats = 0.100 +- 0.900
loss = p:-0.100 v:0.210 ent:1.600 total:-0.060
"""


def test_parse_log_roundtrip(tmp_path):
    from plot_common import parse_log
    p = tmp_path / 'log.txt'
    p.write_text(SAMPLE_LOG)
    data = parse_log(str(p))
    assert data['epochs'] == [0, 1, 2]
    assert data['steps'] == [12, 25]
    assert len(data['losses']) == 3
    assert data['losses'][1][1]['v'] == pytest.approx(0.23)
    assert len(data['win_rates']) == 3
    assert data['win_rates'][0][2] == pytest.approx(0.538)
    assert data['win_rates'][1][1] == 'random'
    assert len(data['gen_stats']) == 2


def test_plots_render(tmp_path):
    pytest.importorskip('matplotlib')
    log = tmp_path / 'log.txt'
    log.write_text(SAMPLE_LOG)
    for script, out in [('loss_plot.py', 'l.png'),
                        ('win_rate_plot.py', 'w.png'),
                        ('stats_plot.py', 's.png')]:
        outp = tmp_path / out
        res = subprocess.run(
            [sys.executable, os.path.join(REPO, 'scripts', script),
             str(log), str(outp)], capture_output=True, text=True, timeout=120)
        assert res.returncode == 0, res.stderr
        assert outp.exists()


def test_swa_averaging(tmp_path):
    """Equal-weight average of checkpoints."""
    from handyrl_amd.envs.tictactoe import SimpleConv2dModel
    m = SimpleConv2dModel()
    os.makedirs(tmp_path / 'models', exist_ok=True)
    states = []
    for i in (1, 2):
        for p in m.parameters():
            p.data.fill_(float(i))
        torch.save(m.state_dict(), tmp_path / 'models' / ('%d.pth' % i))
        states.append({k: v.clone() for k, v in m.state_dict().items()})

    res = subprocess.run(
        [sys.executable, os.path.join(REPO, 'scripts', 'aux_swa.py'), 'TicTacToe'],
        cwd=tmp_path, capture_output=True, text=True, timeout=180,
        env={**os.environ, 'PYTHONPATH': REPO})
    assert res.returncode == 0, res.stderr
    swa = torch.load(tmp_path / 'models' / 'swa.pth')
    for k, v in swa.items():
        if v.dtype.is_floating_point and 'running_var' not in k and 'num_batches' not in k:
            # parameters were filled with 1 then 2 -> average 1.5
            if k.endswith('weight') or k.endswith('bias'):
                assert torch.allclose(v, torch.full_like(v, 1.5)), k
