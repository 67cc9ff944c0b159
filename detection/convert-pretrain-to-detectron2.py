#!/usr/bin/env python
"""Convert a MoCo pretrain checkpoint to a detectron2-loadable pickle.

Capability parity with the reference detection/convert-pretrain-to-detectron2.py
(~35 LoC): keep module.encoder_q.* keys, strip the prefix, rename
conv1/bn1 -> stem.* and layerX.Y -> res{X+1}.Y with detectron2 BN naming,
write {"model": ..., "__author__": "MOCO", "matching_heuristics": True}.

Usage: python convert-pretrain-to-detectron2.py input.pth.tar output.pkl
"""

import pickle as pkl
import sys

import torch


def convert(input_path: str, output_path: str):
    obj = torch.load(input_path, map_location="cpu", weights_only=False)
    obj = obj.get("state_dict", obj)

    newmodel = {}
    for k, v in obj.items():
        if not k.startswith("module.encoder_q."):
            continue
        old_k = k
        k = k[len("module.encoder_q."):]
        if "layer" not in k:
            k = "stem." + k
        for t in [1, 2, 3, 4]:
            k = k.replace(f"layer{t}", f"res{t + 1}")
        for t in [1, 2, 3]:
            k = k.replace(f"bn{t}", f"conv{t}.norm")
        k = k.replace("downsample.0", "shortcut")
        k = k.replace("downsample.1", "shortcut.norm")
        print(old_k, "->", k)
        newmodel[k] = v.numpy()

    res = {"model": newmodel, "__author__": "MOCO",
           "matching_heuristics": True}
    with open(output_path, "wb") as f:
        pkl.dump(res, f)


if __name__ == "__main__":
    convert(sys.argv[1], sys.argv[2])
