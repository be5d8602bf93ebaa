"""Export a trained checkpoint to ONNX for the onnxruntime inference
backend (handyrl_amd.evaluation.OnnxModel).

Usage: python scripts/make_onnx_model.py <model.pth> [env_name]
Input names follow the 'input.*' / 'hidden.*' convention with a dynamic
batch axis (parity: reference scripts/make_onnx_model.py).
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from handyrl_amd.environment import make_env, prepare_env
from handyrl_amd.model import ModelWrapper
from handyrl_amd.util import map_r


class OnnxShim(torch.nn.Module):
    """Flattens the dict in/out model interface into positional tensors."""

    def __init__(self, model, obs_template, hidden_template):
        super().__init__()
        self.model = model
        self.obs_template = obs_template
        self.hidden_template = hidden_template

    def forward(self, *flat):
        flat = list(flat)

        def rebuild(template):
            return map_r(template, lambda _: flat.pop(0))

        obs = rebuild(self.obs_template)
        hidden = rebuild(self.hidden_template) if self.hidden_template is not None else None
        out = self.model(obs, hidden)
        outputs, hiddens = [], []
        for k, v in out.items():
            if k == 'hidden' and v is not None:
                map_r(v, lambda t: hiddens.append(t))
            elif v is not None:
                outputs.append((k, v))
        return tuple([v for _, v in outputs] + hiddens)


def main():
    model_path = sys.argv[1] if len(sys.argv) > 1 else 'models/latest.pth'
    env_name = sys.argv[2] if len(sys.argv) > 2 else None

    env_args = {'env': env_name} if env_name else _load_config_env()
    prepare_env(env_args)
    env = make_env(env_args)
    model = env.net()
    model.load_state_dict(torch.load(model_path, map_location='cpu'))
    model.eval()

    env.reset()
    obs = env.observation(env.players()[0])
    wrapped = ModelWrapper(model)
    hidden = model.init_hidden([1]) if hasattr(model, 'init_hidden') else None

    flat_inputs, input_names = [], []

    def collect(prefix, x):
        counter = [0]

        def add(v):
            t = torch.from_numpy(np.array(v, dtype=np.float32)).unsqueeze(0) \
                if not isinstance(v, torch.Tensor) else v
            flat_inputs.append(t)
            input_names.append('%s.%d' % (prefix, counter[0]))
            counter[0] += 1
        map_r(x, add)

    collect('input', obs)
    if hidden is not None:
        collect('hidden', hidden)

    out = wrapped.inference(obs, wrapped.init_hidden())
    output_names = [k for k in out if k != 'hidden' and out[k] is not None]
    if out.get('hidden') is not None:
        n_hidden_out = len([None for _ in _flatten(out['hidden'])])
        output_names += ['hidden.%d' % i for i in range(n_hidden_out)]

    shim = OnnxShim(model, obs, hidden)
    onnx_path = model_path.rsplit('.', 1)[0] + '.onnx'
    torch.onnx.export(
        shim, tuple(flat_inputs), onnx_path,
        input_names=input_names, output_names=output_names,
        dynamic_axes={name: {0: 'batch'} for name in input_names + output_names},
        opset_version=17, dynamo=False)
    print('exported %s' % onnx_path)


def _flatten(x):
    out = []
    map_r(x, out.append)
    return out


def _load_config_env():
    import yaml
    with open('config.yaml') as f:
        return yaml.safe_load(f)['env_args']


if __name__ == '__main__':
    main()
