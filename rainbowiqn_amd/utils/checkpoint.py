"""Checkpoint-format compatibility with the reference.

The reference saves ``torch.save(online_net.state_dict())`` (SURVEY.md section
6.4).  Our module names follow the surveyed reference layout (convs.{0,2,4},
iqn_fc, fc_h_v/fc_h_a/fc_z_v/fc_z_a with NoisyLinear mu/sigma params), so a
reference checkpoint should load directly.  Because the reference mount was
empty at survey time (SURVEY.md header), exact names are [MED]-confidence;
this remapper absorbs the plausible variants so checkpoints load either way.
"""

from __future__ import annotations

from typing import Dict

import torch

# plausible alternate key spellings -> our canonical keys
_ALIASES = {
    "quantile_fc": "iqn_fc",
    "phi": "iqn_fc",
    "cosine_fc": "iqn_fc",
    "iqn_embedding": "iqn_fc",
}


def remap_reference_state_dict(sd: Dict[str, torch.Tensor], model) -> Dict[str, torch.Tensor]:
    ours = model.state_dict()
    out = {}
    for k, v in sd.items():
        key = k
        if key not in ours:
            for alias, canon in _ALIASES.items():
                if key.startswith(alias + "."):
                    key = canon + key[len(alias):]
                    break
        if key in ours:
            out[key] = v
    # keep our values for anything the checkpoint lacks (e.g. noise buffers)
    for k, v in ours.items():
        out.setdefault(k, v)
    return out
