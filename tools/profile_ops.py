"""Op-level attribution of a ViT-B train step via torch.profiler.

Identifies which torch-level ops spawn the `elementwise_kernel` copies seen
in rocprof kernel stats (kernel names alone don't say who called them).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import timm_amd


def main():
    torch.manual_seed(0)
    B = 256
    model = timm_amd.create_model('vit_base_patch16_224', num_classes=1000)
    model = model.to('cuda', torch.bfloat16).train()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    x = torch.randn(B, 3, 224, 224, device='cuda', dtype=torch.bfloat16)
    y = torch.randint(0, 1000, (B,), device='cuda')

    def step():
        opt.zero_grad(set_to_none=True)
        loss = torch.nn.functional.cross_entropy(model(x).float(), y)
        loss.backward()
        opt.step()

    for _ in range(3):
        step()
    torch.cuda.synchronize()

    from torch.profiler import profile, ProfilerActivity
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
        for _ in range(2):
            step()
        torch.cuda.synchronize()

    print(prof.key_averages().table(sort_by='self_cuda_time_total', row_limit=40, max_name_column_width=60))


if __name__ == '__main__':
    main()
