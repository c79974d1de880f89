#!/usr/bin/env python3
"""Folder inference -> CSV/JSON/parquet (reference `inference.py`, 389 LoC)."""
import argparse
import json
import logging
import os
import time
from contextlib import suppress
from functools import partial

import numpy as np
import pandas as pd
import torch

from timm_amd.data import create_dataset, create_loader, resolve_data_config
from timm_amd.models import create_model
from timm_amd.utils import AverageMeter, setup_default_logging, set_jit_fuser, ParseKwargs

_logger = logging.getLogger('inference')


parser = argparse.ArgumentParser(description='MI355X-native inference')
parser.add_argument('data', nargs='?', metavar='DIR', const=None, help='path to dataset (positional is *deprecated*)')
parser.add_argument('--data-dir', metavar='DIR', help='path to dataset (root dir)')
parser.add_argument('--dataset', metavar='NAME', default='', help='dataset type + name')
parser.add_argument('--split', metavar='NAME', default='validation', help='dataset split')
parser.add_argument('--model', '-m', metavar='MODEL', default='resnet50', help='model architecture')
parser.add_argument('-j', '--workers', default=2, type=int, metavar='N')
parser.add_argument('-b', '--batch-size', default=256, type=int, metavar='N')
parser.add_argument('--img-size', default=None, type=int, metavar='N')
parser.add_argument('--in-chans', type=int, default=None, metavar='N')
parser.add_argument('--input-size', default=None, nargs=3, type=int, metavar='N N N')
parser.add_argument('--use-train-size', action='store_true', default=False)
parser.add_argument('--crop-pct', default=None, type=float, metavar='N')
parser.add_argument('--crop-mode', default=None, type=str, metavar='N')
parser.add_argument('--mean', type=float, nargs='+', default=None, metavar='MEAN')
parser.add_argument('--std', type=float, nargs='+', default=None, metavar='STD')
parser.add_argument('--interpolation', default='', type=str, metavar='NAME')
parser.add_argument('--num-classes', type=int, default=None)
parser.add_argument('--class-map', default='', type=str, metavar='FILENAME')
parser.add_argument('--log-freq', default=10, type=int, metavar='N')
parser.add_argument('--checkpoint', default='', type=str, metavar='PATH')
parser.add_argument('--pretrained', dest='pretrained', action='store_true')
parser.add_argument('--num-gpu', type=int, default=1)
parser.add_argument('--test-pool', dest='test_pool', action='store_true')
parser.add_argument('--channels-last', action='store_true', default=False)
parser.add_argument('--device', default='cuda', type=str)
parser.add_argument('--amp', action='store_true', default=False)
parser.add_argument('--amp-dtype', default='bfloat16', type=str)
parser.add_argument('--fuser', default='', type=str)
parser.add_argument('--model-kwargs', nargs='*', default={}, action=ParseKwargs)
parser.add_argument('--torchcompile', nargs='?', type=str, default=None, const='inductor')

parser.add_argument('--results-dir', type=str, default=None)
parser.add_argument('--results-file', type=str, default=None)
parser.add_argument('--results-format', type=str, nargs='+', default=['csv'])
parser.add_argument('--results-separate-col', action='store_true', default=False)
parser.add_argument('--topk', default=1, type=int, metavar='N', help='Top-k to output to CSV')
parser.add_argument('--fullname', action='store_true', default=False)
parser.add_argument('--filename-col', type=str, default='filename')
parser.add_argument('--index-col', type=str, default='index')
parser.add_argument('--label-col', type=str, default='label')
parser.add_argument('--output-col', type=str, default=None)
parser.add_argument('--output-type', type=str, default='prob')
parser.add_argument('--label-type', type=str, default='description')
parser.add_argument('--include-index', action='store_true', default=False)
parser.add_argument('--exclude-output', action='store_true', default=False)
parser.add_argument('--no-console-results', action='store_true', default=False)


def main():
    setup_default_logging()
    args = parser.parse_args()
    # use pretrained weights if available for the model and no checkpoint given
    from timm_amd.models import is_model_pretrained
    args.pretrained = args.pretrained or (not args.checkpoint and is_model_pretrained(args.model))

    device = torch.device(args.device)

    # resolve AMP arguments based on PyTorch / Apex availability
    amp_autocast = suppress
    if args.amp:
        assert args.amp_dtype in ('float16', 'bfloat16')
        amp_dtype = torch.bfloat16 if args.amp_dtype == 'bfloat16' else torch.float16
        amp_autocast = partial(torch.autocast, device_type=device.type, dtype=amp_dtype)
        _logger.info('Running inference in mixed precision with native AMP.')
    else:
        _logger.info('Running inference in float32.')

    if args.fuser:
        set_jit_fuser(args.fuser)

    # create model
    in_chans = 3
    if args.in_chans is not None:
        in_chans = args.in_chans
    elif args.input_size is not None:
        in_chans = args.input_size[0]

    model = create_model(
        args.model,
        num_classes=args.num_classes,
        in_chans=in_chans,
        pretrained=args.pretrained,
        checkpoint_path=args.checkpoint,
        **args.model_kwargs,
    )
    if args.num_classes is None:
        assert hasattr(model, 'num_classes'), 'Model must have `num_classes` attr if not set on cmd line/config.'
        args.num_classes = model.num_classes

    _logger.info(
        f'Model {args.model} created, param count: {sum([m.numel() for m in model.parameters()])}')

    data_config = resolve_data_config(vars(args), model=model)
    model = model.to(device)
    model.eval()
    if args.channels_last:
        model = model.to(memory_format=torch.channels_last)

    if args.torchcompile:
        model = torch.compile(model, backend=args.torchcompile)

    if args.num_gpu > 1:
        model = torch.nn.DataParallel(model, device_ids=list(range(args.num_gpu)))

    root_dir = args.data or args.data_dir
    dataset = create_dataset(
        root=root_dir,
        name=args.dataset,
        split=args.split,
        class_map=args.class_map,
    )

    loader = create_loader(
        dataset,
        batch_size=args.batch_size,
        use_prefetcher=device.type == 'cuda',
        num_workers=args.workers,
        device=device,
        **{k: v for k, v in data_config.items() if k in ('input_size', 'interpolation', 'mean', 'std', 'crop_pct', 'crop_mode')},
    )

    to_label = None
    if args.label_type in ('name', 'description', 'detail'):
        # prefer the shipped ImageNet synset metadata when the classifier
        # width matches a known subset; fall back to the dataset's class map
        from timm_amd.data import DatasetInfoLabelMapper, ImageNetInfo, infer_imagenet_subset
        subset = infer_imagenet_subset(model)
        if subset is not None:
            to_label = DatasetInfoLabelMapper(ImageNetInfo(subset), label_type=args.label_type)
        else:
            class_to_idx = getattr(dataset, 'class_to_idx', None) or getattr(dataset.reader, 'class_to_idx', {})
            if class_to_idx:
                idx_to_class = {v: k for k, v in class_to_idx.items()}
                to_label = lambda x: idx_to_class.get(x, str(x))  # noqa: E731
    top_k = min(args.topk, args.num_classes)
    batch_time = AverageMeter()
    end = time.time()
    all_indices = []
    all_labels = []
    all_outputs = []
    use_probs = args.output_type == 'prob'
    with torch.no_grad():
        for batch_idx, (input, _) in enumerate(loader):
            with amp_autocast():
                output = model(input)

            if use_probs:
                output = output.softmax(-1)

            if top_k:
                output, indices = output.topk(top_k)
                np_indices = indices.cpu().numpy()
                if args.include_index:
                    all_indices.append(np_indices)
                if to_label is not None:
                    np_labels = np.vectorize(to_label)(np_indices)
                    all_labels.append(np_labels)

            all_outputs.append(output.float().cpu().numpy())

            # measure elapsed time
            batch_time.update(time.time() - end)
            end = time.time()

            if batch_idx % args.log_freq == 0:
                _logger.info('Predict: [{0}/{1}] Time {batch_time.val:.3f} ({batch_time.avg:.3f})'.format(
                    batch_idx, len(loader), batch_time=batch_time))

    all_indices = np.concatenate(all_indices, axis=0) if all_indices else None
    all_labels = np.concatenate(all_labels, axis=0) if all_labels else None
    all_outputs = np.concatenate(all_outputs, axis=0).astype(np.float32)
    filenames = loader.dataset.filenames(basename=not args.fullname)

    output_col = args.output_col or ('prob' if use_probs else 'logit')
    data_dict = {args.filename_col: filenames}
    if args.results_separate_col and all_outputs.shape[-1] > 1:
        if all_indices is not None:
            for i in range(all_indices.shape[-1]):
                data_dict[f'{args.index_col}_{i}'] = all_indices[:, i]
        if all_labels is not None:
            for i in range(all_labels.shape[-1]):
                data_dict[f'{args.label_col}_{i}'] = all_labels[:, i]
        for i in range(all_outputs.shape[-1]):
            data_dict[f'{output_col}_{i}'] = all_outputs[:, i]
    else:
        if all_indices is not None:
            if all_indices.shape[-1] == 1:
                all_indices = all_indices.squeeze(-1)
            data_dict[args.index_col] = list(all_indices)
        if all_labels is not None:
            if all_labels.shape[-1] == 1:
                all_labels = all_labels.squeeze(-1)
            data_dict[args.label_col] = list(all_labels)
        if all_outputs.shape[-1] == 1:
            all_outputs = all_outputs.squeeze(-1)
        data_dict[output_col] = list(all_outputs)

    df = pd.DataFrame(data=data_dict)

    results_filename = args.results_file
    if results_filename:
        filename_no_ext, ext = os.path.splitext(results_filename)
        if ext and ext in ('.csv', '.json'):
            results_filename = filename_no_ext
    else:
        # base default filename on model name + img-size
        img_size = data_config["input_size"][1]
        results_filename = f'{args.model}-{img_size}'

    if args.results_dir:
        results_filename = os.path.join(args.results_dir, results_filename)

    for fmt in args.results_format:
        save_results(df, results_filename, fmt, args.filename_col)

    print(f'--result')
    if not args.no_console_results:
        print(df.set_index(args.filename_col).to_json(orient='index', indent=4))


def save_results(df, results_filename, results_format='csv', filename_col='filename'):
    np.set_printoptions(threshold=int(1e5))
    results_filename += f'.{results_format}'
    if results_format == 'parquet':
        df.set_index(filename_col).to_parquet(results_filename)
    elif results_format == 'json':
        df.set_index(filename_col).to_json(results_filename, indent=4, orient='index')
    elif results_format == 'json-records':
        df.to_json(results_filename, lines=True, orient='records')
    elif results_format == 'json-split':
        df.to_json(results_filename, indent=4, orient='split', index=False)
    else:
        df.to_csv(results_filename, index=False)


if __name__ == '__main__':
    main()
