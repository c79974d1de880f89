"""Ad-hoc GPU validation: flagship-adjacent configs that aren't covered by
bench.py's default run.  Each check prints one line; run via gpurun.

Usage: python tools/gpu_validation.py [muon_eva02] [naflex_infer] [swin_train]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import timm_amd
from timm_amd import ops


def _time_steps(fn, steps=8, warmup=4):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps


def muon_eva02():
    """BASELINE config #4: eva02_large_patch14_336 + Muon (1-GPU slice)."""
    from timm_amd.optim import create_optimizer_v2
    torch.manual_seed(0)
    B = 32
    model = timm_amd.create_model('eva02_large_patch14_336', num_classes=1000)
    model = model.to('cuda', torch.bfloat16).train()
    opt = create_optimizer_v2(model, opt='muon', lr=1e-3, weight_decay=0.05)
    x = torch.randn(B, 3, 336, 336, device='cuda', dtype=torch.bfloat16)
    y = torch.randint(0, 1000, (B,), device='cuda')
    losses = []

    def step():
        opt.zero_grad(set_to_none=True)
        loss = torch.nn.functional.cross_entropy(model(x).float(), y)
        loss.backward()
        opt.step()
        losses.append(loss.item())

    ms = _time_steps(step, steps=6, warmup=3) * 1000
    print(f'muon_eva02: {B / (ms / 1000):.1f} samples/s  ms_per_step={ms:.1f} '
          f'loss {losses[0]:.3f}->{losses[-1]:.3f}')
    assert losses[-1] < losses[0], 'Muon loss must decrease on fixed batch'


def naflex_infer():
    """BASELINE config #5: naflexvit mixed-res inference."""
    torch.manual_seed(0)
    model = timm_amd.create_model('naflexvit_base_patch16_gap', num_classes=1000)
    model = model.to('cuda', torch.bfloat16).eval()
    results = []
    for res in (256, 384, 512):
        B = 64
        x = torch.randn(B, 3, res, res, device='cuda', dtype=torch.bfloat16)
        with torch.no_grad():
            ms = _time_steps(lambda: model(x), steps=10, warmup=5) * 1000
        results.append(f'{res}px {B / (ms / 1000):.0f}/s')
    print('naflex_infer:', '  '.join(results))


def swin_train():
    """Swin-T train step — exercises the window-cyclic fused attention path."""
    torch.manual_seed(0)
    B = 256
    model = timm_amd.create_model('swin_tiny_patch4_window7_224', num_classes=1000)
    model = model.to('cuda', torch.bfloat16).train()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    x = torch.randn(B, 3, 224, 224, device='cuda', dtype=torch.bfloat16)
    y = torch.randint(0, 1000, (B,), device='cuda')

    def step():
        opt.zero_grad(set_to_none=True)
        loss = torch.nn.functional.cross_entropy(model(x).float(), y)
        loss.backward()
        opt.step()

    ms = _time_steps(step, steps=8, warmup=4) * 1000
    print(f'swin_tiny_train: {B / (ms / 1000):.1f} samples/s  ms_per_step={ms:.1f}')


if __name__ == '__main__':
    assert torch.cuda.is_available()
    assert ops.has_ext(), 'HIP extension missing'
    checks = sys.argv[1:] or ['muon_eva02', 'naflex_infer', 'swin_train']
    for name in checks:
        globals()[name]()
