#!/usr/bin/env python3
"""Produce a deploy checkpoint: keep only the (EMA-preferring) weights, write
atomically, and tag the filename with a SHA256 prefix.

Behavioral parity: /root/reference/clean_checkpoint.py (same flags, same
hash-suffix naming, same aux-BN stripping for SplitBN-trained weights).
"""
import argparse
import hashlib
import os
import shutil

import torch

from timm_amd.models import load_state_dict

try:
    import safetensors.torch
    _has_safetensors = True
except ImportError:
    _has_safetensors = False


def clean_checkpoint(
        checkpoint,
        output,
        use_ema=True,
        no_hash=False,
        clean_aux_bn=False,
        safe_serialization: bool = False,
):
    """Strip training state and re-save; returns the final filename ('' on error)."""
    if not (checkpoint and os.path.isfile(checkpoint)):
        print(f"Error: Checkpoint ({checkpoint}) doesn't exist")
        return ''

    print(f"=> Loading checkpoint '{checkpoint}'")
    state_dict = load_state_dict(checkpoint, use_ema=use_ema)
    if clean_aux_bn:
        # dropping every aux_bn key reduces SplitBN layers to plain
        # BatchNorm2d so the cleaned weights load into an unmodified model
        state_dict = {k: v for k, v in state_dict.items() if 'aux_bn' not in k}
    print(f"=> Loaded state_dict from '{checkpoint}'")

    if output:
        out_dir, base = os.path.split(output)
        base, ext = os.path.splitext(base)
    else:
        out_dir, ext = '', ''
        base = os.path.splitext(os.path.basename(checkpoint))[0]
    if not ext:
        ext = '.safetensors' if safe_serialization else '.pth'

    tmp_path = '__' + base
    if safe_serialization:
        assert _has_safetensors, '`pip install safetensors` to use .safetensors'
        safetensors.torch.save_file(state_dict, tmp_path)
    else:
        torch.save(state_dict, tmp_path)

    with open(tmp_path, 'rb') as f:
        digest = hashlib.sha256(f.read()).hexdigest()

    final_name = base + ext if no_hash else f'{base}-{digest[:8]}{ext}'
    shutil.move(tmp_path, os.path.join(out_dir, final_name))
    print(f"=> Saved state_dict to '{final_name}, SHA256: {digest}'")
    return final_name


def main():
    parser = argparse.ArgumentParser(description='Checkpoint cleaner')
    parser.add_argument('--checkpoint', default='', type=str, metavar='PATH',
                        help='path to latest checkpoint (default: none)')
    parser.add_argument('--output', default='', type=str, metavar='PATH',
                        help='output path')
    parser.add_argument('--no-use-ema', dest='no_use_ema', action='store_true',
                        help='use ema version of weights if present')
    parser.add_argument('--no-hash', dest='no_hash', action='store_true',
                        help='no hash in output filename')
    parser.add_argument('--clean-aux-bn', dest='clean_aux_bn', action='store_true',
                        help='remove auxiliary batch norm layers (from SplitBN training) from checkpoint')
    parser.add_argument('--safetensors', action='store_true',
                        help='Save weights using safetensors instead of the default torch way (pickle).')
    args = parser.parse_args()

    if os.path.exists(args.output):
        print(f'Error: Output filename ({args.output}) already exists.')
        raise SystemExit(1)

    clean_checkpoint(
        args.checkpoint,
        args.output,
        use_ema=not args.no_use_ema,
        no_hash=args.no_hash,
        clean_aux_bn=args.clean_aux_bn,
        safe_serialization=args.safetensors,
    )


if __name__ == '__main__':
    main()
