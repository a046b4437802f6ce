"""Flash-attention kernel timing: hand-written CDNA4 kernels vs torch SDPA.

Run on a GPU box:  python benchmarks/attn_bench.py
"""
import math
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def main():
    from adapcc_amd.ops.attention import _FlashAttnFn

    B, H, S, D = 64, 12, 1024, 64  # GPT-2 small flagship shape
    scale = 1.0 / math.sqrt(D)
    torch.manual_seed(0)
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    g = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)

    # FLOPs (causal): fwd 2 matmuls, bwd 5
    fwd_flops = 4 * B * H * S * S * D / 2
    bwd_flops = 2.5 * fwd_flops

    def fa_fwd():
        return _FlashAttnFn.apply(q, k, v, scale)

    def sdpa_fwd():
        return torch.nn.functional.scaled_dot_product_attention(
            q, k, v, is_causal=True)

    for name, f in (("adapcc-fa", fa_fwd), ("torch-sdpa", sdpa_fwd)):
        ms = bench(lambda: f())
        print(f"{name} fwd:  {ms:7.3f} ms  {fwd_flops/ms/1e9:8.1f} TF/s")

        def fb():
            q.grad = k.grad = v.grad = None
            o = f()
            o.backward(g)

        ms = bench(fb)
        print(f"{name} f+b:  {ms:7.3f} ms  {(fwd_flops+bwd_flops)/ms/1e9:8.1f} TF/s")


if __name__ == "__main__":
    main()
