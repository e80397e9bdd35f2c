#!/usr/bin/env python3
"""Minimal repro: MultiShapeGraphedTrainer over cycling padded shapes."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from wap_amd.models import WAPModel
from wap_amd.parallel import BucketedDDP
from wap_amd.runtime.graph_train import MultiShapeGraphedTrainer

def main():
    torch.manual_seed(0)
    dev = "cuda"
    m = WAPModel(n_tokens=112).to(dev)
    ddp = BucketedDDP(m)
    opt = torch.optim.Adadelta(m.parameters(), lr=1.0, weight_decay=1e-4)
    tr = MultiShapeGraphedTrainer(m, ddp, opt, clip=100.0, autocast_bf16=True)
    # the real padded-shape set of a --synthetic 256 run (T up to 64,
    # feature grids up to 12x48)
    shapes = [(128, 320, 64), (128, 512, 64), (128, 640, 48),
              (192, 320, 48), (192, 448, 32), (192, 448, 64),
              (192, 512, 48), (192, 512, 64), (192, 576, 32),
              (192, 576, 48), (192, 640, 64), (192, 704, 48),
              (192, 704, 64), (192, 768, 32), (192, 768, 48),
              (192, 768, 64)]
    nsteps = int(sys.argv[1]) if len(sys.argv) > 1 else 60
    for step in range(nsteps):
        h, w, t = shapes[step % len(shapes)]
        g = torch.Generator().manual_seed(step)
        batch = {
            "x": torch.rand(16, 1, h, w, generator=g).to(dev),
            "x_mask": torch.ones(16, h, w, device=dev),
            "y": torch.randint(1, 112, (16, t), generator=g).to(dev),
            "y_mask": torch.ones(16, t, device=dev),
        }
        loss = tr.step(batch)
        if step % 10 == 0:
            print(f"step {step} loss {float(loss):.3f}", flush=True)
    print("OK", flush=True)

if __name__ == "__main__":
    main()
