#!/usr/bin/env python3
"""Micro-bench driver for the custom CDNA4 kernels (used under rocprofv3
--pmc for counter evidence): fused LayerNorm fwd/bwd and the multi-source
reduction at the GPT-2 shapes."""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import adapcc_amd._core as core
from adapcc_amd.ops.fused import FusedLayerNorm


def main():
    if not torch.cuda.is_available():
        raise SystemExit("kernel_microbench needs an MI355X GPU")
    rows, cols = 65536, 768
    x = torch.randn(rows, cols, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    g = torch.randn_like(x)
    ln = FusedLayerNorm(cols).to("cuda", torch.bfloat16)
    for _ in range(20):
        y = ln(x)
        y.backward(g)
        x.grad = None
    torch.cuda.synchronize()

    # fused cross-entropy at the GPT-2 logits shape
    logits = torch.randn(8192, 50257, device="cuda", dtype=torch.bfloat16,
                         requires_grad=True)
    tgt = torch.randint(0, 50257, (8192,), device="cuda")
    from adapcc_amd.ops.fused import fused_cross_entropy
    for _ in range(10):
        loss = fused_cross_entropy(logits, tgt)
        loss.backward()
        logits.grad = None
    torch.cuda.synchronize()

    n = 32 << 20
    srcs = [torch.randn(n // 4, device="cuda") for _ in range(8)]
    dst = torch.empty_like(srcs[0])
    stream = torch.cuda.current_stream().cuda_stream
    for _ in range(10):
        core.local_reduce(dst.data_ptr(), [s.data_ptr() for s in srcs],
                          dst.numel(), core.DTYPE_F32, core.OP_SUM, 1.0,
                          stream)
    torch.cuda.synchronize()
    print("microbench done")


if __name__ == "__main__":
    main()
