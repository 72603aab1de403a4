"""Pretrained-weight import.

The reference idiom loads MatConvNet's `imagenet-vgg-f.mat` release of
CNN-F (Chatfield et al. 2014). MatConvNet stores `net.layers` as an array
of structs; conv/fc layers carry `weights` = [filters, biases] with filters
in H x W x Cin x Cout order — which is exactly this framework's HWIO layout,
so the import is a straight copy (fc layers arrive as 1x1xKxN or KxN and are
reshaped to [K, N]).

No network access exists in this environment, so tests build a synthetic
.mat with the same structure (scipy.io.savemat round-trip).
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np
import torch

# canonical MatConvNet layer names for CNN-F -> our VGG-F parameter names
VGGF_LAYER_MAP: Dict[str, str] = {
    "conv1": "features.0", "conv2": "features.3", "conv3": "features.6",
    "conv4": "features.7", "conv5": "features.8",
    "fc6": "fc6", "fc7": "fc7", "fc8": "fc8",
}


def _layers_from_mat(mat: dict) -> List[dict]:
    """Normalize scipy.io.loadmat's nested object arrays into a list of
    {name, filters, biases} dicts for layers that have weights."""
    layers = mat["layers"]
    out = []
    flat = np.ravel(layers)
    for cell in flat:
        entry = np.ravel(cell)[0] if isinstance(cell, np.ndarray) else cell
        try:
            name = str(np.ravel(entry["name"])[0])
        except (KeyError, ValueError, IndexError, TypeError):
            continue
        try:
            weights = entry["weights"]
        except (KeyError, ValueError):
            continue
        w = np.ravel(weights)
        if w.size < 2:
            continue
        out.append({"name": name,
                    "filters": np.asarray(w[0], dtype=np.float32),
                    "biases": np.asarray(w[1], dtype=np.float32).reshape(-1)})
    return out


def load_matconvnet_vggf(path: str, model: torch.nn.Module) -> int:
    """Load MatConvNet CNN-F weights into a VGGF model (fp32, in place).

    Returns the number of parameter tensors loaded. Unknown layer names are
    skipped (MatConvNet files carry relu/norm/pool layers without weights).
    """
    from scipy.io import loadmat
    mat = loadmat(path, struct_as_record=True, squeeze_me=False)
    params = dict(model.named_parameters())
    loaded = 0
    for layer in _layers_from_mat(mat):
        prefix = VGGF_LAYER_MAP.get(layer["name"])
        if prefix is None:
            continue
        wkey, bkey = prefix + ".weight", prefix + ".bias"
        if wkey not in params:
            continue
        f = layer["filters"]
        target = params[wkey]
        if target.dim() == 2:           # fc: MatConvNet 1x1xKxN (or KxN)
            f = f.reshape(-1, f.shape[-1])
        want = tuple(target.shape)
        if tuple(f.shape) != want:
            raise ValueError(f"{layer['name']}: shape {f.shape} != {want}")
        with torch.no_grad():
            target.copy_(torch.from_numpy(f))
            params[bkey].copy_(torch.from_numpy(layer["biases"]))
        loaded += 2
    return loaded
