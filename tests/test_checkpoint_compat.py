"""Checkpoint (.pth state_dict) layout compatibility with the reference
networks: identical keys and shapes, and reference weights load into our
models with strict=True (BASELINE.md names the .pth layout a
compatibility requirement; reference train.py:441-454 save format)."""

import ast
import os
import sys

import torch

REFERENCE = '/root/reference'


def _load_ref_module(relpath, class_names):
    """Exec only the named top-level classes from a reference source file
    (read-only; avoids module-level imports we cannot satisfy, e.g.
    kaggle_environments)."""
    src = open(os.path.join(REFERENCE, relpath)).read()
    tree = ast.parse(src)
    keep = [n for n in tree.body
            if isinstance(n, ast.ClassDef) and n.name in class_names]
    mod = ast.Module(body=keep, type_ignores=[])
    ns = {'torch': torch, 'nn': torch.nn,
          'F': torch.nn.functional, 'np': __import__('numpy')}
    exec(compile(ast.fix_missing_locations(mod), relpath, 'exec'), ns)  # noqa: S102
    return ns


def _assert_state_dicts_compatible(ours, ref):
    ks_o, ks_r = ours.state_dict(), ref.state_dict()
    assert set(ks_o.keys()) == set(ks_r.keys()), (
        sorted(set(ks_o) ^ set(ks_r)))
    for k in ks_o:
        assert tuple(ks_o[k].shape) == tuple(ks_r[k].shape), \
            (k, ks_o[k].shape, ks_r[k].shape)
    # a reference checkpoint must load strict into our model and vice versa
    ours.load_state_dict(ks_r, strict=True)
    ref.load_state_dict(ks_o, strict=True)


def test_geister_net_checkpoint_layout():
    sys.path.insert(0, REFERENCE)
    try:
        from handyrl.envs.geister import GeisterNet as RefNet
    finally:
        sys.path.remove(REFERENCE)
    from handyrl_amd.models.geister_net import GeisterNet
    _assert_state_dicts_compatible(GeisterNet(), RefNet())


def test_tictactoe_net_checkpoint_layout():
    sys.path.insert(0, REFERENCE)
    try:
        from handyrl.envs.tictactoe import SimpleConv2dModel as RefNet
    finally:
        sys.path.remove(REFERENCE)
    from handyrl_amd.envs.tictactoe import SimpleConv2dModel
    _assert_state_dicts_compatible(SimpleConv2dModel(), RefNet())


def test_geese_net_checkpoint_layout():
    """Reference GeeseNet keeps a redundant conv bias under BN; our model
    folds it into running_mean on load (exactly).  Prove key coverage AND
    bit-level output equivalence after the fold."""
    ns = _load_ref_module('handyrl/envs/kaggle/hungry_geese.py',
                          {'TorusConv2d', 'GeeseNet'})
    from handyrl_amd.models.geese_net import GeeseNet, load_reference_state_dict
    ref = ns['GeeseNet']()
    torch.manual_seed(5)
    for p in ref.parameters():
        p.data.uniform_(-0.5, 0.5)
    for m in ref.modules():
        if isinstance(m, torch.nn.BatchNorm2d):
            m.running_mean.uniform_(-0.3, 0.3)
            m.running_var.uniform_(0.6, 1.5)
    ours = GeeseNet()
    load_reference_state_dict(ours, ref.state_dict())
    # key coverage: only the redundant conv biases differ
    extra = set(ref.state_dict()) - set(ours.state_dict())
    assert all(k.endswith('.conv.bias') for k in extra), extra
    ref.eval(); ours.eval()
    x = (torch.rand(37, 17, 7, 11) < 0.2).float()
    with torch.no_grad():
        o_ref = ref(x)
        o_ours = ours(x, None)
    torch.testing.assert_close(o_ours['policy'], o_ref['policy'],
                               rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(o_ours['value'], o_ref['value'],
                               rtol=1e-5, atol=1e-5)


def test_geister_net_forward_equivalence():
    """Same weights -> same outputs: our GeisterNet computes the
    reference's function (policy, value, return and hidden carry)."""
    sys.path.insert(0, REFERENCE)
    try:
        from handyrl.envs.geister import GeisterNet as RefNet
    finally:
        sys.path.remove(REFERENCE)
    from handyrl_amd.models.geister_net import GeisterNet

    torch.manual_seed(11)
    ref = RefNet()
    for p in ref.parameters():
        p.data.uniform_(-0.3, 0.3)
    for m in ref.modules():
        if isinstance(m, torch.nn.BatchNorm2d):
            m.running_mean.uniform_(-0.2, 0.2)
            m.running_var.uniform_(0.6, 1.4)
    ours = GeisterNet()
    ours.load_state_dict(ref.state_dict(), strict=True)
    ref.eval(); ours.eval()

    B = 9
    x = {'scalar': torch.rand(B, 18), 'board': torch.rand(B, 7, 6, 6)}
    h_ref = ref.init_hidden([B])
    h_ours = ours.init_hidden([B])
    with torch.no_grad():
        o_ref = ref({'scalar': x['scalar'].clone(),
                     'board': x['board'].clone()}, h_ref)
        o_ours = ours({'scalar': x['scalar'].clone(),
                       'board': x['board'].clone()}, h_ours)
    for k in ('policy', 'value', 'return'):
        torch.testing.assert_close(o_ours[k], o_ref[k], rtol=1e-5, atol=1e-5,
                                   msg=lambda m, k=k: '%s: %s' % (k, m))
    ho, co = o_ours['hidden']
    hr, cr = o_ref['hidden']
    for a, b in zip(list(ho) + list(co), list(hr) + list(cr)):
        torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-5)


def test_tictactoe_net_forward_equivalence():
    sys.path.insert(0, REFERENCE)
    try:
        from handyrl.envs.tictactoe import SimpleConv2dModel as RefNet
    finally:
        sys.path.remove(REFERENCE)
    from handyrl_amd.envs.tictactoe import SimpleConv2dModel
    torch.manual_seed(13)
    ref = RefNet()
    for p in ref.parameters():
        p.data.uniform_(-0.4, 0.4)
    ours = SimpleConv2dModel()
    ours.load_state_dict(ref.state_dict(), strict=True)
    ref.eval(); ours.eval()
    x = torch.rand(7, 3, 3, 3)
    with torch.no_grad():
        o_ref = ref(x)
        o_ours = ours(x, None)
    torch.testing.assert_close(o_ours['policy'], o_ref['policy'],
                               rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(o_ours['value'], o_ref['value'],
                               rtol=1e-5, atol=1e-5)


def test_geese_net_export_to_reference():
    """The ours -> reference direction: a checkpoint saved by this repo
    (reference_state_dict) loads into the reference GeeseNet with every
    key present (no silently-kept random conv biases) and produces
    identical outputs."""
    ns = _load_ref_module('handyrl/envs/kaggle/hungry_geese.py',
                          {'TorusConv2d', 'GeeseNet'})
    from handyrl_amd.models.geese_net import GeeseNet
    torch.manual_seed(7)
    ours = GeeseNet()
    for p in ours.parameters():
        p.data.uniform_(-0.5, 0.5)
    for m in ours.modules():
        if isinstance(m, torch.nn.BatchNorm2d):
            m.running_mean.uniform_(-0.3, 0.3)
            m.running_var.uniform_(0.6, 1.5)
    sd = ours.reference_state_dict()
    ref = ns['GeeseNet']()
    # full key coverage: strict load must succeed (this is what the
    # reference's strict=False load would silently skip without the export)
    ref.load_state_dict(sd, strict=True)
    ours.eval(); ref.eval()
    x = (torch.rand(23, 17, 7, 11) < 0.2).float()
    with torch.no_grad():
        o_ref = ref(x)
        o_ours = ours(x, None)
    torch.testing.assert_close(o_ours['policy'], o_ref['policy'],
                               rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(o_ours['value'], o_ref['value'],
                               rtol=1e-5, atol=1e-5)
