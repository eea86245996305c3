"""Time conv3x3_wgrad (ky-row-merged taps) vs MIOpen wrw at resnet50 3x3
shapes, batch 64. One line per shape -> gpurun_out/wgrad3.txt friendly."""
import sys
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
from deeplearning_amd.ops.conv1x1 import ext  # noqa: E402


def t_ms(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def main():
    E = ext()
    B = 64
    for H, C in [(56, 64), (28, 128), (14, 256), (7, 512)]:
        x = torch.randn(B, C, H, H, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True).to(memory_format=torch.channels_last)
        w = torch.randn(C, C, 3, 3, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True).to(memory_format=torch.channels_last)
        dy = torch.randn(B, C, H, H, device="cuda", dtype=torch.bfloat16
                         ).to(memory_format=torch.channels_last)
        xf = x.detach().permute(0, 2, 3, 1).reshape(-1, C).contiguous()
        dyf = dy.permute(0, 2, 3, 1).reshape(-1, C).contiguous()

        ours = t_ms(lambda: E.conv3x3_wgrad(dyf, xf, H, H))
        y = torch.nn.functional.conv2d(x, w, padding=1)
        torch.autograd.grad(y, w, dy, retain_graph=True)  # MIOpen find warmup
        miopen = t_ms(lambda: torch.autograd.grad(y, w, dy,
                                                  retain_graph=True))
        # parity vs fp32 eager
        dw9 = E.conv3x3_wgrad(dyf, xf, H, H)
        dw = dw9.view(C, 3, 3, C).permute(0, 3, 1, 2)
        xr = x.detach().float().requires_grad_()
        wr = w.detach().float().requires_grad_()
        yr = torch.nn.functional.conv2d(xr, wr, padding=1)
        yr.backward(dy.float())
        rel = (dw - wr.grad).abs().max() / wr.grad.abs().max()
        print(f"3x3 wgrad {H}x{C}: dla {ours:.3f} ms  miopen-wrw "
              f"{miopen:.3f} ms  ratio {miopen / ours:.2f}  relerr {rel:.2e}")


if __name__ == "__main__":
    main()
