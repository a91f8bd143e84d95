#!/usr/bin/env python3
# -*- coding: utf-8 -*-
"""CIFAR-10 training example driven by YAML configs.

Mirrors the reference example's role (``examples/cifar10/train.py``) as the
de-facto integration test: the same train/predict loops run under every
runtime config in ``config/*.yaml`` (cpu, single-gpu, ddp, ddp+amp/bf16,
ddp+oss+sddp, deepspeed-style zero-2, horovod-compat).

Launch (single process):
    python train.py --config config/cpu.yaml
Distributed (one rank per GPU):
    torchrun --nproc-per-node 8 --master-addr 127.0.0.1 train.py \
        --config config/ddp_bf16.yaml

Uses synthetic CIFAR-shaped data when no dataset is on disk (this container
has no network); pass --data-root to use a real folder of tensors.
"""

import argparse
import os
import sys

import torch
import torch.nn as nn
import yaml

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from benchmarks.models import resnet152  # noqa: E402
from stoke import (  # noqa: E402
    AMPConfig,
    ClipGradNormConfig,
    DDPConfig,
    DeepspeedConfig,
    DeepspeedFP16Config,
    DeepspeedZeROConfig,
    FairscaleOSSConfig,
    FairscaleSDDPConfig,
    HorovodConfig,
    Stoke,
    StokeOptimizer,
)
from stoke.ops.fused_adam import FusedAdamW  # noqa: E402


class SyntheticCIFAR(torch.utils.data.Dataset):
    def __init__(self, n=2048):
        g = torch.Generator().manual_seed(0)
        self.x = torch.randn(n, 3, 32, 32, generator=g)
        self.y = torch.randint(0, 10, (n,), generator=g)

    def __len__(self):
        return len(self.x)

    def __getitem__(self, i):
        return self.x[i], self.y[i]


def build_configs(cfg: dict):
    out = []
    if "ddp" in cfg:
        out.append(DDPConfig(**cfg["ddp"]))
    if "amp" in cfg:
        out.append(AMPConfig(**cfg["amp"]))
    if "oss" in cfg:
        out.append(FairscaleOSSConfig(**cfg["oss"]))
    if "sddp" in cfg:
        out.append(FairscaleSDDPConfig(**cfg["sddp"]))
    if "horovod" in cfg:
        out.append(HorovodConfig(**cfg["horovod"]))
    if "deepspeed" in cfg:
        ds = dict(cfg["deepspeed"])
        zero = ds.pop("zero", None)
        fp16 = ds.pop("fp16", None)
        out.append(
            DeepspeedConfig(
                zero_optimization=DeepspeedZeROConfig(**zero) if zero else DeepspeedZeROConfig(),
                fp16=DeepspeedFP16Config(**fp16) if fp16 else None,
                **ds,
            )
        )
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", required=True)
    ap.add_argument("--epochs", type=int, default=1)
    ap.add_argument("--samples", type=int, default=2048)
    ap.add_argument("--max-steps", type=int, default=0,
                    help="stop each epoch after N steps (0 = full epoch)")
    ap.add_argument("--batch", type=int, default=0,
                    help="override the config batch size")
    args = ap.parse_args()
    with open(args.config) as f:
        cfg = yaml.safe_load(f)

    run = cfg.get("run", {})
    grad_clip = (
        ClipGradNormConfig(**cfg["clip_norm"]) if "clip_norm" in cfg else None
    )
    model = resnet152(num_classes=10, small_input=True)
    stoke_obj = Stoke(
        model=model,
        optimizer=StokeOptimizer(
            optimizer=FusedAdamW if run.get("gpu") else torch.optim.AdamW,
            optimizer_kwargs=cfg.get("optimizer", {"lr": 1e-3}),
        ),
        loss=nn.CrossEntropyLoss(),
        batch_size_per_device=args.batch or run.get("batch_size", 32),
        grad_accum_steps=run.get("grad_accum", 1),
        grad_clip=grad_clip,
        gpu=run.get("gpu", False),
        fp16=run.get("fp16"),
        distributed=run.get("distributed"),
        fairscale_oss=run.get("oss", False),
        fairscale_sddp=run.get("sddp", False),
        fairscale_fsdp=run.get("fsdp", False),
        configs=build_configs(cfg),
        verbose=run.get("verbose", True),
    )
    ds = SyntheticCIFAR(args.samples)
    sampler = None
    if stoke_obj.distributed is not None:
        sampler = torch.utils.data.distributed.DistributedSampler(
            ds, num_replicas=stoke_obj.world_size, rank=stoke_obj.rank
        )
    loader = stoke_obj.DataLoader(ds, sampler=sampler,
                                  shuffle=(sampler is None))
    stoke_obj.print_num_model_parameters()
    for epoch in range(args.epochs):
        if sampler is not None:
            sampler.set_epoch(epoch)
        train(stoke_obj, loader, args.max_steps)
        stoke_obj.print_ema_loss()
    # predict loop
    correct = predict(stoke_obj, loader, args.max_steps)
    stoke_obj.print(f"train accuracy: {correct:.3f}")


def train(stoke_obj, loader, max_steps=0):
    stoke_obj.model_access.train()
    for i, (x, y) in enumerate(loader):
        if max_steps and i >= max_steps:
            break
        out = stoke_obj.model(x)
        loss = stoke_obj.loss(out, y)
        stoke_obj.print_mean_accumulated_synced_loss()
        stoke_obj.backward(loss)
        stoke_obj.step()


def predict(stoke_obj, loader, max_steps=0):
    stoke_obj.model_access.eval()
    hits = n = 0
    with torch.no_grad():
        for i, (x, y) in enumerate(loader):
            if max_steps and i >= max_steps:
                break
            out = stoke_obj.model(x)
            hits += (out.argmax(-1) == y).sum().item()
            n += len(y)
    return hits / max(n, 1)


if __name__ == "__main__":
    main()
