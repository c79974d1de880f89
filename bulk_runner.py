#!/usr/bin/env python3
"""Subprocess-per-model runner for benchmark/validate sweeps
(reference `bulk_runner.py`, 244 LoC) — isolates each model run against
OOM/crash, merges result CSVs.
"""
import argparse
import csv
import json
import os
import subprocess
import sys
import time
from typing import Callable, List, Tuple, Union

import timm_amd

parser = argparse.ArgumentParser(description='Per-model process launcher')

# model and results args
parser.add_argument('--model-list', metavar='NAME', default='',
                    help='txt file based list of model names to benchmark')
parser.add_argument('--results-file', default='', type=str, metavar='FILENAME',
                    help='Output csv file for validation results (summary)')
parser.add_argument('--sort-key', default='', type=str, metavar='COL', help='Specify sort key for results csv')
parser.add_argument("--pretrained", action='store_true', help="only include models with pretrained weights")
parser.add_argument("--delay", type=float, default=0, help="delay between model invocations")
parser.add_argument('script', type=str, nargs='?', default='benchmark.py',
                    help='script to run for each model (benchmark.py or validate.py)')
parser.add_argument('script_args', nargs=argparse.REMAINDER,
                    help='arguments passed through to the script')


def main():
    args = parser.parse_args()

    if args.model_list == 'all':
        model_names = timm_amd.list_models(pretrained=args.pretrained)
    elif args.model_list and not os.path.exists(args.model_list):
        model_names = timm_amd.list_models(args.model_list, pretrained=args.pretrained)
    elif args.model_list:
        with open(args.model_list) as f:
            model_names = [line.rstrip() for line in f if line.rstrip()]
    else:
        model_names = timm_amd.list_models(pretrained=args.pretrained)

    if not model_names:
        print('No models found to run.')
        return 1

    results = []
    errors = []
    print(f'Running {args.script} for {len(model_names)} models.')
    for model_name in model_names:
        cmd = [sys.executable, args.script, '--model', model_name] + args.script_args
        print(f'Running {model_name}...')
        try:
            proc = subprocess.run(cmd, capture_output=True, text=True, check=False)
            out = proc.stdout
            if proc.returncode != 0:
                errors.append(dict(model=model_name, error=proc.stderr.strip().splitlines()[-1:] or 'unknown'))
                continue
            # scripts emit `--result\n{json}` at the end of stdout
            marker = out.rfind('--result')
            if marker < 0:
                errors.append(dict(model=model_name, error='no result marker'))
                continue
            r = json.loads(out[marker + len('--result'):])
            if isinstance(r, list):
                results.extend(r)
            else:
                results.append(r)
        except Exception as e:
            errors.append(dict(model=model_name, error=str(e)))
        if args.delay:
            time.sleep(args.delay)

    if errors:
        print(f'{len(errors)} model(s) failed:')
        for e in errors:
            print(' ', e)

    if args.sort_key and results and args.sort_key in results[0]:
        results = sorted(results, key=lambda x: x.get(args.sort_key, 0), reverse=True)

    if args.results_file and results:
        with open(args.results_file, mode='w') as cf:
            cw = csv.DictWriter(cf, fieldnames=list(results[0].keys()))
            cw.writeheader()
            for r in results:
                cw.writerow(r)
        print(f'Wrote {len(results)} results to {args.results_file}')

    print(json.dumps(results, indent=4))
    return 0


if __name__ == '__main__':
    sys.exit(main())
