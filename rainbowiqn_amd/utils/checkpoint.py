"""Checkpoint-format compatibility with the reference.

The reference saves ``torch.save(online_net.state_dict())`` (SURVEY.md section
6.4).  Our module names follow the surveyed reference layout (convs.{0,2,4},
iqn_fc, fc_h_v/fc_h_a/fc_z_v/fc_z_a with NoisyLinear mu/sigma params), so a
reference checkpoint should load directly.  Because the reference mount was
empty at survey time (SURVEY.md header), exact names are [MED]-confidence;
this remapper absorbs the plausible variants so checkpoints load either way.

Loading is FAIL-LOUD: a checkpoint key that matches neither our names nor an
alias, or a missing model parameter, raises with the full key diff -- a
reference checkpoint with unexpected names must never silently "load" as a
partially random net.  The only silent backfill is the NoisyLinear epsilon
noise buffers, which are transient draws regenerated every act/learn step.
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

# transient buffers: regenerated from device RNG each step, never required
# in a checkpoint (the reference's NoisyLinear registers them as buffers, so
# they MAY be present; either way is accepted)
_TRANSIENT_SUFFIXES = ("weight_epsilon", "bias_epsilon")


def _is_transient(key: str) -> bool:
    return key.rsplit(".", 1)[-1] in _TRANSIENT_SUFFIXES


class CheckpointKeyError(RuntimeError):
    """Checkpoint/model key-name mismatch, with the full diff attached."""

    def __init__(self, unmatched, missing):
        self.unmatched = sorted(unmatched)
        self.missing = sorted(missing)
        msg = ["checkpoint/state_dict key mismatch (refusing to load "
               "partially -- see utils/checkpoint.py)"]
        if self.unmatched:
            msg.append("  checkpoint keys matching no model key or alias:")
            msg += [f"    {k}" for k in self.unmatched]
        if self.missing:
            msg.append("  model keys absent from the checkpoint:")
            msg += [f"    {k}" for k in self.missing]
        super().__init__("\n".join(msg))


def remap_reference_state_dict(sd: Dict[str, torch.Tensor], model) -> Dict[str, torch.Tensor]:
    ours = model.state_dict()
    out = {}
    unmatched = []
    for k, v in sd.items():
        key = k
        if key not in ours:
            for alias, canon in _ALIASES.items():
                if key.startswith(alias + "."):
                    key = canon + key[len(alias):]
                    break
        if key in ours:
            if tuple(v.shape) != tuple(ours[key].shape):
                raise CheckpointKeyError(
                    [f"{k} (shape {tuple(v.shape)} != model "
                     f"{tuple(ours[key].shape)})"], [])
            out[key] = v
        elif _is_transient(key):
            pass  # stale noise draws from an older layout: safe to drop
        else:
            unmatched.append(k)
    missing = [k for k in ours if k not in out and not _is_transient(k)]
    if unmatched or missing:
        raise CheckpointKeyError(unmatched, missing)
    # transient epsilon buffers may be absent: keep our (regenerated) values
    for k, v in ours.items():
        out.setdefault(k, v)
    return out
