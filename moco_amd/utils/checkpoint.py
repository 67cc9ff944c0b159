"""Checkpoint save/load — the layout is a compatibility contract.

Reference layout (main_moco.py:348-352, resume :210-227): a dict
{epoch, arch, state_dict, optimizer} written per epoch by rank 0 as
checkpoint_%04d.pth.tar; state_dict keys are DDP-prefixed:
module.encoder_q.*, module.encoder_k.*, module.queue [dim,K],
module.queue_ptr [1].

Linear eval loads it by keeping only module.encoder_q.* (minus the fc head),
stripping the prefix (main_lincls.py:172-190); sanity_check (:400-425)
re-loads the pretrain checkpoint after linear training and asserts every
non-fc backbone weight is bitwise unchanged.
"""

import shutil
from typing import Dict

import torch


def save_checkpoint(state: Dict, is_best: bool,
                    filename: str = "checkpoint.pth.tar"):
    torch.save(state, filename)
    if is_best:
        shutil.copyfile(filename, "model_best.pth.tar")


def strip_moco_prefix(state_dict: Dict[str, torch.Tensor]
                      ) -> Dict[str, torch.Tensor]:
    """Extract the query-encoder backbone from a MoCo checkpoint state_dict,
    dropping the fc head — the exact key surgery of main_lincls.py:175-185."""
    out = {}
    for k in list(state_dict.keys()):
        if k.startswith("module.encoder_q.") and \
                not k.startswith("module.encoder_q.fc"):
            out[k[len("module.encoder_q."):]] = state_dict[k]
        elif k.startswith("encoder_q.") and not k.startswith("encoder_q.fc"):
            # non-DDP checkpoints (single-process runs)
            out[k[len("encoder_q."):]] = state_dict[k]
    return out


def load_pretrained_backbone(model: torch.nn.Module, pretrained_path: str):
    """Load a MoCo pretrain checkpoint into a bare classifier model for
    linear evaluation. Asserts that exactly the fc head is left missing."""
    ckpt = torch.load(pretrained_path, map_location="cpu", weights_only=False)
    state_dict = ckpt.get("state_dict", ckpt)
    stripped = strip_moco_prefix(state_dict)
    msg = model.load_state_dict(stripped, strict=False)
    assert set(msg.missing_keys) == {"fc.weight", "fc.bias"}, \
        f"unexpected missing keys: {msg.missing_keys}"
    return ckpt


def sanity_check(state_dict: Dict[str, torch.Tensor], pretrained_path: str):
    """After linear eval training: every non-fc backbone weight must be
    bitwise identical to the pretrain checkpoint (main_lincls.py:400-425)."""
    print("=> loading '{}' for sanity check".format(pretrained_path))
    ckpt = torch.load(pretrained_path, map_location="cpu", weights_only=False)
    pre = ckpt.get("state_dict", ckpt)
    for k in state_dict:
        if "fc.weight" in k or "fc.bias" in k:
            continue
        k_bare = k[len("module."):] if k.startswith("module.") else k
        for pk in ("module.encoder_q." + k_bare, "encoder_q." + k_bare):
            if pk in pre:
                break
        else:
            raise KeyError(f"{k_bare} not found in pretrain checkpoint")
        assert (state_dict[k].cpu() == pre[pk]).all(), \
            f"{k} changed during linear training!"
    print("=> sanity check passed")
