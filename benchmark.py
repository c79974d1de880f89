#!/usr/bin/env python3
"""Model benchmark CLI — the reference's north-star metric protocol.

Behavioral parity: /root/reference/benchmark.py (synthetic randn inputs +
random-init weights, warmup then timed steps with device-synchronized
perf_counter, infer/train samples-per-sec rows, OOM batch-decay retry,
CSV/JSON output).  The profiler mode uses torch's built-in FlopCounterMode
instead of the reference's fvcore/deepspeed dependencies.

Redesign notes: one `_MeterLoop` drives warmup/timing/logging for every
runner; runners only define `step()`.
"""
import argparse
import csv
import json
import logging
import time
from collections import OrderedDict
from contextlib import suppress
from functools import partial

import torch
import torch.nn as nn

import timm_amd  # noqa: F401
from timm_amd.data import resolve_data_config
from timm_amd.models import create_model, is_model, list_models
from timm_amd.optim import create_optimizer_v2
from timm_amd.utils import (
    ParseKwargs, check_batch_size_retry, decay_batch_step, set_jit_fuser,
    setup_default_logging,
)

_logger = logging.getLogger('benchmark')

parser = argparse.ArgumentParser(description='MI355X Model Benchmark')
parser.add_argument('--model-list', metavar='NAME', default='',
                    help='txt file based list of model names to benchmark')
parser.add_argument('--bench', default='both', type=str,
                    help="Benchmark mode: 'infer', 'train', 'both', 'profile'")
parser.add_argument('--detail', action='store_true', default=False,
                    help='Provide train fwd/bwd/opt breakdown detail')
parser.add_argument('--no-retry', action='store_true', default=False,
                    help='Do not decay batch size and retry on error.')
parser.add_argument('--results-file', default='', type=str,
                    help='Output csv file for results (summary)')
parser.add_argument('--results-format', default='csv', type=str,
                    help='Format for results file: (csv, json)')
parser.add_argument('--num-warm-iter', default=10, type=int, help='Number of warmup iterations')
parser.add_argument('--num-bench-iter', default=40, type=int, help='Number of benchmark iterations')
parser.add_argument('--device', default='cuda', type=str, help='device to run benchmark on')
parser.add_argument('--model', '-m', metavar='NAME', default='resnet50', help='model architecture')
parser.add_argument('-b', '--batch-size', default=256, type=int, metavar='N')
parser.add_argument('--img-size', default=None, type=int, metavar='N')
parser.add_argument('--input-size', default=None, nargs=3, type=int, metavar='N N N')
parser.add_argument('--use-train-size', action='store_true', default=False)
parser.add_argument('--num-classes', type=int, default=None)
parser.add_argument('--gp', default=None, type=str, metavar='POOL')
parser.add_argument('--channels-last', action='store_true', default=False)
parser.add_argument('--grad-checkpointing', action='store_true', default=False)
parser.add_argument('--amp', action='store_true', default=False)
parser.add_argument('--amp-dtype', default='bfloat16', type=str)
parser.add_argument('--precision', default='float32', type=str,
                    help='Numeric precision when not using AMP: (float32, float16, bfloat16)')
parser.add_argument('--fuser', default='', type=str)
parser.add_argument('--fast-norm', default=False, action='store_true')
parser.add_argument('--model-kwargs', nargs='*', default={}, action=ParseKwargs)
parser.add_argument('--torchcompile', nargs='?', type=str, default=None, const='inductor')
parser.add_argument('--torchscript', dest='torchscript', action='store_true')
# train optimizer parameters
parser.add_argument('--opt', default='sgd', type=str, metavar='OPTIMIZER')
parser.add_argument('--opt-eps', default=None, type=float, metavar='EPSILON')
parser.add_argument('--momentum', type=float, default=0.9, metavar='M')
parser.add_argument('--weight-decay', type=float, default=0.0001)
parser.add_argument('--clip-grad', type=float, default=None, metavar='NORM')
parser.add_argument('--clip-mode', type=str, default='norm')
parser.add_argument('--smoothing', type=float, default=0.1)
parser.add_argument('--drop', type=float, default=0.0, metavar='PCT')
parser.add_argument('--drop-path', type=float, default=None, metavar='PCT')


_PRECISIONS = {
    # name -> (amp_dtype, model/data dtype)
    'amp': (torch.float16, torch.float32),
    'amp_float16': (torch.float16, torch.float32),
    'amp_bfloat16': (torch.bfloat16, torch.float32),
    'float16': (None, torch.float16),
    'bfloat16': (None, torch.bfloat16),
    'float32': (None, torch.float32),
}


def resolve_precision(precision: str):
    assert precision in _PRECISIONS, f'unknown precision {precision}'
    amp_dtype, tensor_dtype = _PRECISIONS[precision]
    return amp_dtype, tensor_dtype, tensor_dtype


def count_params(model: nn.Module):
    return sum(p.numel() for p in model.parameters())


class _Clock:
    """perf_counter with optional device sync before reading."""

    def __init__(self, device):
        self.cuda = 'cuda' in str(device)
        self.device = device

    def __call__(self, sync=False):
        if sync and self.cuda:
            torch.cuda.synchronize(device=self.device)
        return time.perf_counter()


class BenchmarkRunner:
    """Shared setup: model build, precision/layout, synthetic input."""

    def __init__(
            self,
            model_name,
            detail=False,
            device='cuda',
            torchscript=False,
            torchcompile=None,
            precision='float32',
            fuser='',
            num_warm_iter=10,
            num_bench_iter=50,
            use_train_size=False,
            **kwargs,
    ):
        self.model_name = model_name
        self.detail = detail
        self.device = device
        self.amp_dtype, self.model_dtype, self.data_dtype = resolve_precision(precision)
        self.channels_last = kwargs.pop('channels_last', False)
        self.amp_autocast = (
            partial(torch.autocast, device_type=device, dtype=self.amp_dtype)
            if self.amp_dtype is not None else suppress)

        if fuser:
            set_jit_fuser(fuser)
        self.model = create_model(
            model_name,
            num_classes=kwargs.pop('num_classes', None),
            in_chans=3,
            global_pool=kwargs.pop('gp', 'fast'),
            scriptable=torchscript,
            drop_rate=kwargs.pop('drop', 0.),
            drop_path_rate=kwargs.pop('drop_path', None),
            drop_block_rate=kwargs.pop('drop_block', None),
            **kwargs.pop('model_kwargs', {}),
        )
        if kwargs.pop('grad_checkpointing', False):
            self.model.set_grad_checkpointing()
        self.model.to(
            device=self.device,
            dtype=self.model_dtype,
            memory_format=torch.channels_last if self.channels_last else None,
        )
        self.num_classes = self.model.num_classes
        self.param_count = count_params(self.model)
        _logger.info('Model %s created, param count: %d', model_name, self.param_count)

        data_config = resolve_data_config(kwargs, model=self.model, use_test_size=not use_train_size)
        self.input_size = data_config['input_size']
        self.batch_size = kwargs.pop('batch_size', 256)

        self.compiled = False
        if torchscript:
            self.model = torch.jit.script(self.model)
            self.compiled = True
        elif torchcompile:
            torch._dynamo.reset()
            self.model = torch.compile(self.model, backend=torchcompile)
            self.compiled = True

        self.example_inputs = None
        self.num_warm_iter = num_warm_iter
        self.num_bench_iter = num_bench_iter
        self.log_freq = max(num_bench_iter // 5, 1)
        self.time_fn = _Clock(self.device)

    def _init_input(self):
        self.example_inputs = torch.randn(
            (self.batch_size,) + self.input_size, device=self.device, dtype=self.data_dtype)
        if self.channels_last:
            self.example_inputs = self.example_inputs.contiguous(memory_format=torch.channels_last)

    def _base_results(self):
        return dict(
            batch_size=self.batch_size,
            img_size=self.input_size[-1],
            param_count=round(self.param_count / 1e6, 2),
        )

    def _meter_loop(self, step_fn, tag):
        """Warmup, then num_bench_iter timed steps with periodic logging.
        Returns (samples/sec over wall, avg step ms)."""
        for _ in range(self.num_warm_iter):
            step_fn()
        total_step = 0.
        num_samples = 0
        t_run_start = self.time_fn()
        for i in range(self.num_bench_iter):
            total_step += step_fn()
            num_samples += self.batch_size
            steps = i + 1
            if steps % self.log_freq == 0:
                _logger.info(
                    f'{tag} [{steps}/{self.num_bench_iter}].'
                    f' {num_samples / total_step:0.2f} samples/sec.'
                    f' {1000 * total_step / steps:0.3f} ms/step.')
        elapsed = self.time_fn(True) - t_run_start
        return num_samples / elapsed, 1000 * total_step / self.num_bench_iter


class InferenceBenchmarkRunner(BenchmarkRunner):
    """infer_samples_per_sec (reference benchmark.py:293 protocol)."""

    def __init__(self, model_name, device='cuda', torchscript=False, **kwargs):
        super().__init__(model_name=model_name, device=device, torchscript=torchscript, **kwargs)
        self.model.eval()

    def run(self):
        def _step():
            t0 = self.time_fn()
            with self.amp_autocast():
                self.model(self.example_inputs)
            return self.time_fn(True) - t0

        _logger.info(
            f'Running inference benchmark on {self.model_name} for {self.num_bench_iter} steps w/ '
            f'input size {self.input_size} and batch size {self.batch_size}.')
        with torch.no_grad():
            self._init_input()
            sps, step_ms = self._meter_loop(_step, 'Infer')
        results = dict(
            samples_per_sec=round(sps, 2),
            step_time=round(step_ms, 3),
            **self._base_results(),
        )
        _logger.info(
            f"Inference benchmark of {self.model_name} done. "
            f"{results['samples_per_sec']:.2f} samples/sec, {results['step_time']:.2f} ms/step")
        return results


class TrainBenchmarkRunner(BenchmarkRunner):
    """train_samples_per_sec, optional fwd/bwd/opt breakdown
    (reference benchmark.py:368 protocol)."""

    def __init__(self, model_name, device='cuda', torchscript=False, **kwargs):
        super().__init__(model_name=model_name, device=device, torchscript=torchscript, **kwargs)
        self.model.train()
        self.loss = nn.CrossEntropyLoss().to(self.device)
        self.optimizer = create_optimizer_v2(
            self.model,
            opt=kwargs.pop('opt', 'sgd'),
            lr=kwargs.pop('lr', 1e-4))

    def _random_targets(self, n):
        return torch.empty(n, device=self.device, dtype=torch.long).random_(self.num_classes)

    def _step_timed(self):
        self.optimizer.zero_grad()
        t0 = self.time_fn()
        with self.amp_autocast():
            output = self.model(self.example_inputs)
            if isinstance(output, tuple):
                output = output[0]
            self.loss(output, self._random_targets(output.shape[0])).backward()
        self.optimizer.step()
        return self.time_fn(True) - t0

    def _step_detail(self):
        self.optimizer.zero_grad()
        t0 = self.time_fn()
        with self.amp_autocast():
            output = self.model(self.example_inputs)
            if isinstance(output, tuple):
                output = output[0]
            t_fwd = self.time_fn(True)
            self.loss(output, self._random_targets(output.shape[0])).backward()
            t_bwd = self.time_fn(True)
        self.optimizer.step()
        t_end = self.time_fn(True)
        return t_fwd - t0, t_bwd - t_fwd, t_end - t_bwd

    def run(self):
        _logger.info(
            f'Running train benchmark on {self.model_name} for {self.num_bench_iter} steps w/ '
            f'input size {self.input_size} and batch size {self.batch_size}.')
        self._init_input()
        if not self.detail:
            for _ in range(self.num_warm_iter):
                self._step_timed()
            # re-use the meter loop for the plain timed path
            total = 0.
            num_samples = 0
            t_run_start = self.time_fn()
            for i in range(self.num_bench_iter):
                total += self._step_timed()
                num_samples += self.batch_size
                if (i + 1) % self.log_freq == 0:
                    _logger.info(
                        f'Train [{i + 1}/{self.num_bench_iter}].'
                        f' {num_samples / total:0.2f} samples/sec.'
                        f' {1000 * total / (i + 1):0.3f} ms/step.')
            elapsed = self.time_fn() - t_run_start
            results = dict(
                samples_per_sec=round(num_samples / elapsed, 2),
                step_time=round(1000 * total / self.num_bench_iter, 3),
                **self._base_results(),
            )
        else:
            for _ in range(self.num_warm_iter):
                self._step_timed()
            sums = [0., 0., 0.]
            num_samples = 0
            t_run_start = self.time_fn()
            for i in range(self.num_bench_iter):
                deltas = self._step_detail()
                sums = [a + d for a, d in zip(sums, deltas)]
                num_samples += self.batch_size
                if (i + 1) % self.log_freq == 0:
                    steps = i + 1
                    _logger.info(
                        f'Train [{steps}/{self.num_bench_iter}].'
                        f' {num_samples / sum(sums):0.2f} samples/sec.'
                        f' {1000 * sums[0] / steps:0.3f} ms/step fwd,'
                        f' {1000 * sums[1] / steps:0.3f} ms/step bwd,'
                        f' {1000 * sums[2] / steps:0.3f} ms/step opt.')
            elapsed = self.time_fn() - t_run_start
            results = dict(
                samples_per_sec=round(num_samples / elapsed, 2),
                step_time=round(1000 * sum(sums) / self.num_bench_iter, 3),
                fwd_time=round(1000 * sums[0] / self.num_bench_iter, 3),
                bwd_time=round(1000 * sums[1] / self.num_bench_iter, 3),
                opt_time=round(1000 * sums[2] / self.num_bench_iter, 3),
                **self._base_results(),
            )
        _logger.info(
            f"Train benchmark of {self.model_name} done. "
            f"{results['samples_per_sec']:.2f} samples/sec, {results['step_time']:.2f} ms/sample")
        return results


class ProfileRunner(BenchmarkRunner):
    """GMACs/activation profile via torch.utils.flop_counter (no external
    deps; replaces the reference's fvcore/deepspeed integrations)."""

    def __init__(self, model_name, device='cuda', profiler='torch', **kwargs):
        super().__init__(model_name=model_name, device=device, **kwargs)
        self.profiler = profiler
        self.model.eval()

    def run(self):
        from torch.utils.flop_counter import FlopCounterMode
        self._init_input()
        counter = FlopCounterMode(display=False)
        with torch.no_grad(), counter:
            self.model(self.example_inputs)
        total_flops = counter.get_total_flops()
        # FLOPs for the whole batch; report GMACs per sample (flops/2)
        gmacs = total_flops / 2 / self.batch_size / 1e9
        _logger.info(
            f'Profile of {self.model_name}: {gmacs:.2f} GMACs/sample, '
            f'{self.param_count / 1e6:.2f} M params')
        return dict(
            gmacs=round(gmacs, 2),
            macs=round(gmacs, 2),  # reference key alias
            **self._base_results(),
        )


def _try_run(
        model_name,
        bench_fn,
        bench_kwargs,
        initial_batch_size,
        no_batch_size_retry=False,
):
    """Run a benchmark, decaying batch size on OOM-style failures."""
    batch_size = initial_batch_size
    results = dict()
    error_str = 'Unknown'
    while batch_size:
        try:
            if torch.cuda.is_available() and 'cuda' in bench_kwargs.get('device', 'cuda'):
                torch.cuda.empty_cache()
            bench = bench_fn(model_name=model_name, batch_size=batch_size, **bench_kwargs)
            return bench.run()
        except RuntimeError as e:
            error_str = str(e)
            _logger.error(f'"{error_str}" while running benchmark.')
            if no_batch_size_retry or not check_batch_size_retry(error_str):
                break
            batch_size = decay_batch_step(batch_size)
            _logger.warning(f'Reducing batch size to {batch_size} for retry.')
    results['error'] = error_str
    return results


_BENCH_MODES = {
    'infer': (('infer', InferenceBenchmarkRunner),),
    'train': (('train', TrainBenchmarkRunner),),
    'both': (('infer', InferenceBenchmarkRunner), ('train', TrainBenchmarkRunner)),
    'profile': (('', ProfileRunner),),
}


def benchmark(args):
    if args.amp:
        _logger.info('Benchmarking in mixed precision with native AMP.')
        args.precision = 'amp_' + args.amp_dtype
    _logger.info(f'Benchmarking in {args.precision} precision. '
                 f'{"NHWC" if args.channels_last else "NCHW"} layout.')

    bench_kwargs = vars(args).copy()
    for k in ('amp', 'amp_dtype', 'model_list', 'results_file', 'results_format', 'bench'):
        bench_kwargs.pop(k, None)
    model = bench_kwargs.pop('model')
    batch_size = bench_kwargs.pop('batch_size')
    no_retry = bench_kwargs.pop('no_retry')

    model_results = OrderedDict(model=model)
    for prefix, bench_fn in _BENCH_MODES[args.bench]:
        run_results = _try_run(
            model,
            bench_fn,
            bench_kwargs=bench_kwargs,
            initial_batch_size=batch_size,
            no_batch_size_retry=no_retry,
        )
        if prefix and 'error' not in run_results:
            run_results = {f'{prefix}_{k}': v for k, v in run_results.items()}
        model_results.update(run_results)
        if 'error' in run_results:
            break
    if 'error' not in model_results:
        param_count = model_results.pop(
            'infer_param_count', model_results.pop('train_param_count', 0))
        model_results.setdefault('param_count', param_count)
        model_results.pop('train_param_count', 0)
    return model_results


def _expand_models(args):
    """Resolve -m/--model-list into a list of names (empty = single model)."""
    if args.model_list:
        args.model = ''
        with open(args.model_list) as f:
            return [line.rstrip() for line in f]
    if args.model == 'all':
        return list_models(pretrained=True, exclude_filters=['*in21k'])
    if not is_model(args.model):
        return list_models(args.model)  # wildcard
    return []


def main():
    setup_default_logging()
    args = parser.parse_args()

    model_names = _expand_models(args)
    if model_names:
        _logger.info(
            'Running bulk validation on these pretrained models: {}'.format(', '.join(model_names)))
        results = []
        try:
            for name in model_names:
                if not name:
                    continue
                args.model = name
                r = benchmark(args)
                if r:
                    results.append(r)
                time.sleep(10)
        except KeyboardInterrupt:
            pass
        sort_key = 'train_samples_per_sec' if 'train' in args.bench else 'infer_samples_per_sec'
        results = sorted(
            (r for r in results if sort_key in r), key=lambda r: r[sort_key], reverse=True)
    else:
        results = benchmark(args)

    if args.results_file:
        write_results(args.results_file, results, format=args.results_format)

    # JSON to stdout with delimiter for the bulk runner
    print(f'--result\n{json.dumps(results, indent=4)}')


def write_results(results_file, results, format='csv'):
    with open(results_file, mode='w') as f:
        if format == 'json':
            json.dump(results, f, indent=4)
            return
        rows = results if isinstance(results, (list, tuple)) else [results]
        if not rows:
            return
        writer = csv.DictWriter(f, fieldnames=rows[0].keys())
        writer.writeheader()
        writer.writerows(rows)
        f.flush()


if __name__ == '__main__':
    main()
