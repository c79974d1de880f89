#!/usr/bin/env python3
"""MI355X-native ImageNet-style training script.

Capability parity with the reference `train.py` (1,533 LoC): YAML config
pre-parser, distributed init (torchrun-compatible, RCCL backend), model/data/
optimizer/scheduler factories, Mixup/CutMix, AMP (bf16 default), grad accum
w/ no_sync, in-house bucketed DDP, EMA V3, CheckpointSaver with the reference
checkpoint layout, distribute_bn, eval + summary.csv.

Launch: torchrun --nproc_per_node=N train.py --data-dir ... --model vit_base_patch16_224
"""
import argparse
import importlib
import json
import logging
import os
import time
from collections import OrderedDict
from contextlib import suppress
from datetime import datetime
from functools import partial

import torch
import torch.nn as nn
import yaml

import timm_amd
from timm_amd import utils
from timm_amd.data import (
    AugMixDataset, FastCollateMixup, Mixup, NaFlexMixup, create_dataset, create_loader,
    resolve_data_config,
)
from timm_amd.data.naflex_loader import create_naflex_loader
from timm_amd.layers import convert_sync_batchnorm
from timm_amd.loss import BinaryCrossEntropy, JsdCrossEntropy, LabelSmoothingCrossEntropy, SoftTargetCrossEntropy
from timm_amd.models import create_model, safe_model_name
from timm_amd.optim import create_optimizer_v2, optimizer_kwargs
from timm_amd.scheduler import create_scheduler_v2, scheduler_kwargs
from timm_amd.task import (
    ClassificationTask, FeatureDistillationTask, LogitDistillationTask, TokenDistillationTask,
    resume_task_checkpoint,
)
from timm_amd.utils import ModelEmaV3, NativeScaler, dispatch_clip_grad

_logger = logging.getLogger('train')

# The first arg parser parses out only the --config argument, this argument is used to
# load a yaml file containing key-values that override the defaults for the main parser below
config_parser = parser = argparse.ArgumentParser(description='Training Config', add_help=False)
parser.add_argument('-c', '--config', default='', type=str, metavar='FILE',
                    help='YAML config file specifying default arguments')

parser = argparse.ArgumentParser(description='MI355X-native training')

# Dataset parameters
group = parser.add_argument_group('Dataset parameters')
group.add_argument('--data-dir', metavar='DIR', default=None, help='path to dataset (root dir)')
group.add_argument('--dataset', metavar='NAME', default='', help='dataset type + name ("<type>/<name>")')
group.add_argument('--train-split', metavar='NAME', default='train', help='dataset train split')
group.add_argument('--val-split', metavar='NAME', default='validation', help='dataset validation split')
group.add_argument('--dataset-download', action='store_true', default=False)
group.add_argument('--class-map', default='', type=str, metavar='FILENAME')
group.add_argument('--input-img-mode', default=None, type=str)
group.add_argument('--synthetic', action='store_true', default=False,
                   help='use synthetic random data (no dataset required)')
group.add_argument('--synthetic-len', type=int, default=1000, help='synthetic dataset length per rank')

# Model parameters
group = parser.add_argument_group('Model parameters')
group.add_argument('--model', default='resnet50', type=str, metavar='MODEL')
group.add_argument('--pretrained', action='store_true', default=False)
group.add_argument('--initial-checkpoint', default='', type=str, metavar='PATH')
group.add_argument('--resume', default='', type=str, metavar='PATH')
group.add_argument('--no-resume-opt', action='store_true', default=False)
group.add_argument('--num-classes', type=int, default=None, metavar='N')
group.add_argument('--gp', default=None, type=str, metavar='POOL')
group.add_argument('--img-size', type=int, default=None, metavar='N')
group.add_argument('--in-chans', type=int, default=None, metavar='N')
group.add_argument('--input-size', default=None, nargs=3, type=int, metavar='N N N')
group.add_argument('--crop-pct', default=None, type=float, metavar='N')
group.add_argument('--mean', type=float, nargs='+', default=None, metavar='MEAN')
group.add_argument('--std', type=float, nargs='+', default=None, metavar='STD')
group.add_argument('--interpolation', default='', type=str, metavar='NAME')
group.add_argument('-b', '--batch-size', type=int, default=128, metavar='N')
group.add_argument('-vb', '--validation-batch-size', type=int, default=None, metavar='N')
group.add_argument('--channels-last', action='store_true', default=False)
group.add_argument('--grad-accum-steps', type=int, default=1, metavar='N')
group.add_argument('--grad-checkpointing', action='store_true', default=False)
group.add_argument('--fuser', default='', type=str)
group.add_argument('--model-kwargs', nargs='*', default={}, action=utils.ParseKwargs)
group.add_argument('--torchcompile', nargs='?', type=str, default=None, const='inductor')

# Device & distributed
group = parser.add_argument_group('Knowledge-distillation parameters')
group.add_argument('--kd-model-name', default=None, type=str,
                   help='Name of teacher model for knowledge distillation')
group.add_argument('--kd-distill-type', default='logit', type=str, choices=['logit', 'feature', 'token'],
                   help='Type of distillation: "logit" (output KL), "feature" (intermediate features), '
                        '"token" (models with distillation heads, e.g. deit distilled) (default: logit)')
group.add_argument('--kd-loss-type', default='kl', type=str,
                   help='Loss function for logit distillation (default: kl)')
group.add_argument('--distill-loss-weight', default=None, type=float,
                   help='Weight for distillation loss. Both weights given -> independent mix; '
                        'only --task-loss-weight -> complementary (distill = 1 - task).')
group.add_argument('--task-loss-weight', default=None, type=float,
                   help='Weight for task (classification) loss; see --distill-loss-weight')
group.add_argument('--kd-temperature', default=4.0, type=float,
                   help='Softmax temperature for distillation (default: 4.0)')
group.add_argument('--kd-teacher-path', default=None, type=str,
                   help='Local checkpoint path for the teacher weights')
group.add_argument('--kd-student-feature-dim', default=None, type=int,
                   help='Student feature dim for feature distillation (auto-detected if unset)')
group.add_argument('--kd-teacher-feature-dim', default=None, type=int,
                   help='Teacher feature dim for feature distillation (auto-detected if unset)')
group.add_argument('--kd-token-distill-type', default='soft', type=str, choices=['soft', 'hard'],
                   help='Token distillation: "soft" KL w/ temperature, "hard" CE vs teacher argmax')

# Device parameters
group = parser.add_argument_group('Device parameters')
group.add_argument('--device', default='cuda', type=str)
group.add_argument('--amp', action='store_true', default=False, help='use mixed precision training')
group.add_argument('--amp-dtype', default='bfloat16', type=str, help='amp dtype (bfloat16 default on MI355X)')
group.add_argument('--no-ddp-bb', action='store_true', default=False)
group.add_argument('--synchronize-step', action='store_true', default=False)
group.add_argument("--local_rank", default=0, type=int)

# Optimizer parameters
group = parser.add_argument_group('Optimizer parameters')
group.add_argument('--opt', default='sgd', type=str, metavar='OPTIMIZER')
group.add_argument('--opt-eps', default=None, type=float, metavar='EPSILON')
group.add_argument('--opt-betas', default=None, type=float, nargs='+', metavar='BETA')
group.add_argument('--momentum', type=float, default=0.9, metavar='M')
group.add_argument('--weight-decay', type=float, default=2e-5)
group.add_argument('--clip-grad', type=float, default=None, metavar='NORM')
group.add_argument('--clip-mode', type=str, default='norm')
group.add_argument('--layer-decay', type=float, default=None)
group.add_argument('--layer-decay-min-scale', type=float, default=0)
group.add_argument('--layer-decay-no-opt-scale', type=float, default=None)
group.add_argument('--opt-kwargs', nargs='*', default={}, action=utils.ParseKwargs)

# Learning rate schedule parameters
group = parser.add_argument_group('Learning rate schedule parameters')
group.add_argument('--sched', type=str, default='cosine', metavar='SCHEDULER')
group.add_argument('--sched-on-updates', action='store_true', default=False)
group.add_argument('--lr', type=float, default=None, metavar='LR')
group.add_argument('--lr-base', type=float, default=0.1, metavar='LR')
group.add_argument('--lr-base-size', type=int, default=256, metavar='DIV')
group.add_argument('--lr-base-scale', type=str, default='', metavar='SCALE')
group.add_argument('--lr-noise', type=float, nargs='+', default=None, metavar='pct, pct')
group.add_argument('--lr-noise-pct', type=float, default=0.67, metavar='PERCENT')
group.add_argument('--lr-noise-std', type=float, default=1.0, metavar='STDDEV')
group.add_argument('--lr-cycle-mul', type=float, default=1.0, metavar='MULT')
group.add_argument('--lr-cycle-decay', type=float, default=0.5, metavar='MULT')
group.add_argument('--lr-cycle-limit', type=int, default=1, metavar='N')
group.add_argument('--lr-k-decay', type=float, default=1.0)
group.add_argument('--warmup-lr', type=float, default=1e-5, metavar='LR')
group.add_argument('--min-lr', type=float, default=0, metavar='LR')
group.add_argument('--epochs', type=int, default=300, metavar='N')
group.add_argument('--epoch-repeats', type=float, default=0., metavar='N')
group.add_argument('--start-epoch', default=None, type=int, metavar='N')
group.add_argument('--decay-milestones', default=[90, 180, 270], type=int, nargs='+', metavar="MILESTONES")
group.add_argument('--decay-epochs', type=float, default=90, metavar='N')
group.add_argument('--warmup-epochs', type=int, default=5, metavar='N')
group.add_argument('--warmup-prefix', action='store_true', default=False)
group.add_argument('--cooldown-epochs', type=int, default=0, metavar='N')
group.add_argument('--patience-epochs', type=int, default=10, metavar='N')
group.add_argument('--decay-rate', '--dr', type=float, default=0.1, metavar='RATE')

# Augmentation & regularization parameters
group = parser.add_argument_group('Augmentation and regularization parameters')
group.add_argument('--no-aug', action='store_true', default=False)
group.add_argument('--train-crop-mode', type=str, default=None)
group.add_argument('--scale', type=float, nargs='+', default=[0.08, 1.0], metavar='PCT')
group.add_argument('--ratio', type=float, nargs='+', default=[3. / 4., 4. / 3.], metavar='RATIO')
group.add_argument('--hflip', type=float, default=0.5)
group.add_argument('--vflip', type=float, default=0.)
group.add_argument('--color-jitter', type=float, default=0.4, metavar='PCT')
group.add_argument('--color-jitter-prob', type=float, default=None, metavar='PCT')
group.add_argument('--grayscale-prob', type=float, default=None, metavar='PCT')
group.add_argument('--gaussian-blur-prob', type=float, default=None, metavar='PCT')
group.add_argument('--aa', type=str, default=None, metavar='NAME')
group.add_argument('--aug-repeats', type=float, default=0)
group.add_argument('--aug-splits', type=int, default=0)
group.add_argument('--jsd-loss', action='store_true', default=False)
group.add_argument('--bce-loss', action='store_true', default=False)
group.add_argument('--bce-sum', action='store_true', default=False)
group.add_argument('--bce-target-thresh', type=float, default=None)
group.add_argument('--bce-pos-weight', type=float, default=None)
group.add_argument('--reprob', type=float, default=0., metavar='PCT')
group.add_argument('--remode', type=str, default='pixel')
group.add_argument('--recount', type=int, default=1)
group.add_argument('--resplit', action='store_true', default=False)
group.add_argument('--mixup', type=float, default=0.0)
group.add_argument('--cutmix', type=float, default=0.0)
group.add_argument('--cutmix-minmax', type=float, nargs='+', default=None)
group.add_argument('--mixup-prob', type=float, default=1.0)
group.add_argument('--mixup-switch-prob', type=float, default=0.5)
group.add_argument('--mixup-mode', type=str, default='batch')
group.add_argument('--mixup-off-epoch', default=0, type=int, metavar='N')
group.add_argument('--smoothing', type=float, default=0.1)
# NaFlex variable-resolution training (reference train.py:423-435)
group.add_argument('--train-img-sizes', type=int, nargs='+', default=None,
                   help='Per-batch square image sizes for scheduled resolution training')
group.add_argument('--train-batch-sizes', type=int, nargs='+', default=None,
                   help='Batch size for each --train-img-sizes choice (default: --batch-size for every choice)')
group.add_argument('--train-size-probs', type=float, nargs='+', default=None,
                   help='Base sampling weights for --train-img-sizes (default: uniform)')
group.add_argument('--train-size-schedule', type=str, default='constant', choices=('constant', 'progressive'),
                   help='Resolution choice schedule; progressive moves from the first size to the last')
group.add_argument('--train-size-schedule-spread', type=float, default=0.65,
                   help='Progressive schedule spread in resolution-choice index units (default: 0.65)')
group.add_argument('--train-size-random-mix', type=float, default=0.1,
                   help='Fraction of uniform random active choices mixed into a progressive schedule')
group.add_argument('--train-batches-per-epoch', '--train-steps-per-epoch', type=int, default=None,
                   help='Fixed loader batches (before grad accumulation) per epoch; inferred if unspecified')
group.add_argument('--variable-batch-loss-scale', default='none', type=str, choices=('none', 'sqrt', 'linear'),
                   help='Scale gradients relative to the policy-average scheduled batch size')
group.add_argument('--naflex-loader', action='store_true', default=False,
                   help='enable the NaFlex variable-seq-len loader')
group.add_argument('--naflex-train-seq-lens', type=int, nargs='+',
                   default=[128, 256, 576, 784, 1024],
                   help='sequence-length buckets for NaFlex training')
group.add_argument('--naflex-max-seq-len', type=int, default=576,
                   help='fixed sequence length for NaFlex validation')
group.add_argument('--naflex-max-tokens-per-batch', type=int, default=4096 * 4,
                   help='token budget per train batch (batch size varies by bucket)')
group.add_argument('--naflex-loss-scale', default='linear', type=str,
                   choices=('none', 'sqrt', 'linear'),
                   help='variable-batch loss rescale mode')
group.add_argument('--train-interpolation', type=str, default='random')
group.add_argument('--drop', type=float, default=0.0, metavar='PCT')
group.add_argument('--drop-path', type=float, default=None, metavar='PCT')
group.add_argument('--drop-block', type=float, default=None, metavar='PCT')

# Batch norm parameters
group = parser.add_argument_group('Batch norm parameters')
group.add_argument('--bn-momentum', type=float, default=None)
group.add_argument('--bn-eps', type=float, default=None)
group.add_argument('--sync-bn', action='store_true')
group.add_argument('--dist-bn', type=str, default='reduce')

# EMA
group = parser.add_argument_group('Model exponential moving average parameters')
group.add_argument('--model-ema', action='store_true', default=False)
group.add_argument('--model-ema-force-cpu', action='store_true', default=False)
group.add_argument('--model-ema-decay', type=float, default=0.9998)
group.add_argument('--model-ema-warmup', action='store_true')

# Misc
group = parser.add_argument_group('Miscellaneous parameters')
group.add_argument('--seed', type=int, default=42, metavar='S')
group.add_argument('--worker-seeding', type=str, default='all')
group.add_argument('--log-interval', type=int, default=50, metavar='N')
group.add_argument('--recovery-interval', type=int, default=0, metavar='N')
group.add_argument('--checkpoint-hist', type=int, default=10, metavar='N')
group.add_argument('-j', '--workers', type=int, default=4, metavar='N')
group.add_argument('--save-images', action='store_true', default=False)
group.add_argument('--pin-mem', action='store_true', default=False)
group.add_argument('--no-prefetcher', action='store_true', default=False)
group.add_argument('--output', default='', type=str, metavar='PATH')
group.add_argument('--experiment', default='', type=str, metavar='NAME')
group.add_argument('--eval-metric', default='top1', type=str, metavar='EVAL_METRIC')
group.add_argument('--log-wandb', action='store_true', default=False)
group.add_argument('--max-steps-per-epoch', type=int, default=None,
                   help='truncate each epoch to this many steps (debug/smoke)')


def _parse_args():
    # Do we have a config file to parse?
    args_config, remaining = config_parser.parse_known_args()
    if args_config.config:
        with open(args_config.config, 'r') as f:
            cfg = yaml.safe_load(f)
            parser.set_defaults(**cfg)

    # The main arg parser parses the rest of the args, the usual
    # defaults will have been overridden if config file specified.
    args = parser.parse_args(remaining)

    # Cache the args as a text string to save them in the output dir later
    args_text = yaml.safe_dump(args.__dict__, default_flow_style=False)
    return args, args_text


class SyntheticDataset(torch.utils.data.Dataset):
    """Random-image dataset for environments with no real data."""

    def __init__(self, length, input_size=(3, 224, 224), num_classes=1000, transform=None):
        self.length = length
        self.input_size = input_size
        self.num_classes = num_classes
        self.transform = transform

    def __len__(self):
        return self.length

    def __getitem__(self, idx):
        import numpy as np
        from PIL import Image
        rng = np.random.RandomState(idx)
        img = Image.fromarray(rng.randint(0, 255, (self.input_size[1], self.input_size[2], 3), dtype=np.uint8))
        if self.transform is not None:
            img = self.transform(img)
        return img, idx % self.num_classes


def main():
    utils.setup_default_logging()
    args, args_text = _parse_args()

    args.prefetcher = not args.no_prefetcher
    device = utils.init_distributed_device(args)
    if args.distributed:
        _logger.info(
            'Training in distributed mode with multiple processes, 1 device per process.'
            f'Process {args.rank}, total {args.world_size}, device {args.device}.')
    else:
        _logger.info(f'Training with a single process on 1 device ({args.device}).')

    use_amp = None
    amp_dtype = torch.float16
    if args.amp:
        use_amp = 'native'
        assert args.amp_dtype in ('float16', 'bfloat16')
        amp_dtype = torch.bfloat16 if args.amp_dtype == 'bfloat16' else torch.float16

    utils.random_seed(args.seed, args.rank)

    in_chans = 3
    if args.in_chans is not None:
        in_chans = args.in_chans
    elif args.input_size is not None:
        in_chans = args.input_size[0]

    model = create_model(
        args.model,
        pretrained=args.pretrained,
        in_chans=in_chans,
        num_classes=args.num_classes,
        drop_rate=args.drop,
        drop_path_rate=args.drop_path,
        drop_block_rate=args.drop_block,
        global_pool=args.gp,
        bn_momentum=args.bn_momentum,
        bn_eps=args.bn_eps,
        checkpoint_path=args.initial_checkpoint,
        **args.model_kwargs,
    )
    if args.num_classes is None:
        assert hasattr(model, 'num_classes'), 'Model must have `num_classes` attr if not set on cmd line/config.'
        args.num_classes = model.num_classes

    if args.grad_checkpointing:
        model.set_grad_checkpointing(enable=True)

    if utils.is_primary(args):
        _logger.info(
            f'Model {safe_model_name(args.model)} created, param count:{sum([m.numel() for m in model.parameters()])}')

    data_config = resolve_data_config(vars(args), model=model, verbose=utils.is_primary(args))

    # setup augmentation batch splits for contrastive loss or split bn
    num_aug_splits = 0
    if args.aug_splits > 0:
        assert args.aug_splits > 1, 'A split of 1 makes no sense'
        num_aug_splits = args.aug_splits

    # move model to GPU
    model.to(device=device)
    if args.channels_last:
        model.to(memory_format=torch.channels_last)

    # setup synchronized BatchNorm for distributed training
    if args.distributed and args.sync_bn:
        args.dist_bn = ''  # disable dist_bn when sync BN active
        model = convert_sync_batchnorm(model)
        if utils.is_primary(args):
            _logger.info(
                'Converted model to use Synchronized BatchNorm. WARNING: You may have issues if using '
                'zero initialized BN layers (enabled by default for ResNets) while sync-bn enabled.')

    # optionally resume from a checkpoint
    loss_scaler = None
    if use_amp == 'native' and amp_dtype == torch.float16:
        loss_scaler = NativeScaler(device=device.type)
    amp_autocast = suppress
    if use_amp == 'native':
        amp_autocast = partial(torch.autocast, device_type=device.type, dtype=amp_dtype)

    # setup learning rate schedule and starting epoch
    resume_epoch = None

    # task wraps model + criterion; distributed wrap happens below
    # setup loss function
    if args.jsd_loss:
        assert num_aug_splits > 1  # JSD only valid with aug splits set
        train_loss_fn = JsdCrossEntropy(num_splits=num_aug_splits, smoothing=args.smoothing)
    elif args.mixup > 0 or args.cutmix > 0. or args.cutmix_minmax is not None:
        # smoothing is handled with mixup target transform which outputs sparse, soft targets
        if args.bce_loss:
            train_loss_fn = BinaryCrossEntropy(
                target_threshold=args.bce_target_thresh,
                sum_classes=args.bce_sum,
                pos_weight=args.bce_pos_weight,
            )
        else:
            train_loss_fn = SoftTargetCrossEntropy()
    elif args.smoothing:
        if args.bce_loss:
            train_loss_fn = BinaryCrossEntropy(
                smoothing=args.smoothing,
                target_threshold=args.bce_target_thresh,
                sum_classes=args.bce_sum,
                pos_weight=args.bce_pos_weight,
            )
        else:
            train_loss_fn = LabelSmoothingCrossEntropy(smoothing=args.smoothing)
    else:
        train_loss_fn = nn.CrossEntropyLoss()
    train_loss_fn = train_loss_fn.to(device=device)
    validate_loss_fn = nn.CrossEntropyLoss().to(device=device)

    if args.kd_model_name is not None:
        if args.kd_distill_type == 'logit':
            task = LogitDistillationTask(
                student_model=model,
                teacher_model=args.kd_model_name,
                criterion=train_loss_fn,
                teacher_pretrained_path=args.kd_teacher_path,
                loss_type=args.kd_loss_type,
                distill_loss_weight=args.distill_loss_weight,
                task_loss_weight=args.task_loss_weight,
                temperature=args.kd_temperature,
                device=device,
                verbose=utils.is_primary(args),
            )
        elif args.kd_distill_type == 'feature':
            task = FeatureDistillationTask(
                student_model=model,
                teacher_model=args.kd_model_name,
                criterion=train_loss_fn,
                teacher_pretrained_path=args.kd_teacher_path,
                distill_loss_weight=args.distill_loss_weight,
                task_loss_weight=args.task_loss_weight,
                student_feature_dim=args.kd_student_feature_dim,
                teacher_feature_dim=args.kd_teacher_feature_dim,
                device=device,
                verbose=utils.is_primary(args),
            )
        elif args.kd_distill_type == 'token':
            task = TokenDistillationTask(
                student_model=model,
                teacher_model=args.kd_model_name,
                criterion=train_loss_fn,
                teacher_pretrained_path=args.kd_teacher_path,
                distill_type=args.kd_token_distill_type,
                distill_loss_weight=args.distill_loss_weight,
                task_loss_weight=args.task_loss_weight,
                temperature=args.kd_temperature,
                device=device,
                verbose=utils.is_primary(args),
            )
        else:
            raise ValueError(f'Unknown distillation type: {args.kd_distill_type}')
        task.teacher.to(device=device)
    else:
        task = ClassificationTask(model, criterion=train_loss_fn)

    # optimizer over the task's trainable module (includes e.g. the feature
    # distillation projection, not just the bare model)
    optimizer = create_optimizer_v2(
        task.model,
        **optimizer_kwargs(cfg=args),
        **args.opt_kwargs,
    )

    if args.resume:
        resume_epoch = resume_task_checkpoint(
            task,
            args.resume,
            optimizer=None if args.no_resume_opt else optimizer,
            loss_scaler=None if args.no_resume_opt else loss_scaler,
            log_info=utils.is_primary(args),
        )

    # setup exponential moving average of model weights, SWA could be used here too
    if args.model_ema:
        task.setup_ema(
            decay=args.model_ema_decay,
            warmup=args.model_ema_warmup,
            device=device,
            force_cpu=args.model_ema_force_cpu,
        )
        if args.resume:
            pass  # EMA restored inside resume_task_checkpoint via load_checkpoint_state

    # torch.compile BEFORE DDP wrap (reference `train.py:1015`)
    if args.torchcompile:
        task.compile(backend=args.torchcompile)

    # setup distributed training
    if args.distributed:
        if utils.is_primary(args):
            _logger.info("Using in-house BucketedDataParallel (RCCL over xGMI).")
        task.prepare_distributed()

    # create the train and eval datasets + loaders
    if args.synthetic or args.data_dir is None:
        if utils.is_primary(args):
            _logger.info('Using synthetic random data.')
        dataset_train = SyntheticDataset(
            args.synthetic_len, input_size=data_config['input_size'], num_classes=args.num_classes)
        dataset_eval = SyntheticDataset(
            max(args.synthetic_len // 10, 2 * args.batch_size),
            input_size=data_config['input_size'], num_classes=args.num_classes)
    else:
        dataset_train = create_dataset(
            args.dataset,
            root=args.data_dir,
            split=args.train_split,
            is_training=True,
            class_map=args.class_map,
            download=args.dataset_download,
            batch_size=args.batch_size,
            seed=args.seed,
            repeats=args.epoch_repeats,
            input_img_mode=args.input_img_mode,
        )
        dataset_eval = create_dataset(
            args.dataset,
            root=args.data_dir,
            split=args.val_split,
            is_training=False,
            class_map=args.class_map,
            download=args.dataset_download,
            batch_size=args.batch_size,
            input_img_mode=args.input_img_mode,
        )

    # setup mixup / cutmix
    collate_fn = None
    mixup_fn = None
    mixup_active = args.mixup > 0 or args.cutmix > 0. or args.cutmix_minmax is not None
    if mixup_active:
        mixup_args = dict(
            mixup_alpha=args.mixup,
            cutmix_alpha=args.cutmix,
            cutmix_minmax=args.cutmix_minmax,
            prob=args.mixup_prob,
            switch_prob=args.mixup_switch_prob,
            mode=args.mixup_mode,
            label_smoothing=args.smoothing,
            num_classes=args.num_classes,
        )
        if args.prefetcher:
            assert not num_aug_splits  # collate conflict (need to support de-interleaving in collate mixup)
            collate_fn = FastCollateMixup(**mixup_args)
        else:
            mixup_fn = Mixup(**mixup_args)
            mixup_fn.mixup_enabled = True

    # wrap dataset in AugMix helper
    if num_aug_splits > 1:
        dataset_train = AugMixDataset(dataset_train, num_splits=num_aug_splits)

    naflex_mode = False
    if args.naflex_loader:
        # NaFlex variable-res pipeline (reference train.py:739-805): token-budget
        # batches over seq-len buckets; mixup runs inside the dataset wrapper
        assert num_aug_splits <= 1, 'Augmentation splits not supported in NaFlex mode'
        naflex_mixup_fn = None
        if mixup_active:
            nf_args = dict(mixup_args)
            nf_args.pop('mode', None)
            nf_args.pop('cutmix_minmax', None)
            naflex_mixup_fn = NaFlexMixup(**nf_args)
            collate_fn = None
            mixup_fn = None
        patch_size = 16
        if hasattr(model, 'patch_embed') and hasattr(model.patch_embed, 'patch_size'):
            patch_size = model.patch_embed.patch_size
            if isinstance(patch_size, (tuple, list)):
                patch_size = patch_size[0]
        elif hasattr(model, 'embeds') and hasattr(model.embeds, 'patch_size'):
            patch_size = model.embeds.patch_size
            if isinstance(patch_size, (tuple, list)):
                patch_size = patch_size[0]
        naflex_mode = True
        loader_train = create_naflex_loader(
            dataset_train,
            patch_size=patch_size,
            train_seq_lens=args.naflex_train_seq_lens,
            max_tokens_per_batch=args.naflex_max_tokens_per_batch,
            mixup_fn=naflex_mixup_fn,
            is_training=True,
            mean=data_config['mean'],
            std=data_config['std'],
            num_workers=args.workers,
            distributed=args.distributed,
            rank=args.rank,
            world_size=args.world_size,
            seed=args.seed,
            pin_memory=args.pin_mem,
            device=device,
            use_prefetcher=args.prefetcher,
        )
        loader_eval = create_naflex_loader(
            dataset_eval,
            patch_size=patch_size,
            max_seq_len=args.naflex_max_seq_len,
            batch_size=args.validation_batch_size or args.batch_size,
            is_training=False,
            mean=data_config['mean'],
            std=data_config['std'],
            num_workers=args.workers,
            distributed=args.distributed,
            rank=args.rank,
            world_size=args.world_size,
            pin_memory=args.pin_mem,
            device=device,
            use_prefetcher=args.prefetcher,
        )

    # create data loaders w/ augmentation pipeline
    train_interpolation = args.train_interpolation
    if args.no_aug or not train_interpolation:
        train_interpolation = data_config['interpolation']
    if naflex_mode:
        pass  # loaders built above
    else:
        loader_train = create_loader(
            dataset_train,
            input_size=data_config['input_size'],
            batch_size=args.batch_size,
            is_training=True,
            no_aug=args.no_aug,
            re_prob=args.reprob,
            re_mode=args.remode,
            re_count=args.recount,
            re_split=args.resplit,
            train_crop_mode=args.train_crop_mode,
            scale=args.scale,
            ratio=args.ratio,
            hflip=args.hflip,
            vflip=args.vflip,
            color_jitter=args.color_jitter,
            color_jitter_prob=args.color_jitter_prob,
            grayscale_prob=args.grayscale_prob,
            gaussian_blur_prob=args.gaussian_blur_prob,
            auto_augment=args.aa,
            num_aug_repeats=args.aug_repeats,
            num_aug_splits=num_aug_splits,
            interpolation=train_interpolation,
            mean=data_config['mean'],
            std=data_config['std'],
            num_workers=args.workers,
            distributed=args.distributed,
            collate_fn=collate_fn,
            pin_memory=args.pin_mem,
            device=device,
            use_prefetcher=args.prefetcher,
            worker_seeding=args.worker_seeding,
            input_size_choices=args.train_img_sizes,
            batch_size_choices=args.train_batch_sizes,
            batch_choice_weights=args.train_size_probs,
            batch_choice_seed=args.seed,
            batch_choice_schedule=args.train_size_schedule,
            batch_schedule_epochs=args.epochs if args.train_size_schedule == 'progressive' else None,
            batch_schedule_spread=args.train_size_schedule_spread,
            batch_schedule_random_mix=args.train_size_random_mix,
            num_batches=args.train_batches_per_epoch,
        )

        loader_eval = create_loader(
            dataset_eval,
            input_size=data_config['input_size'],
            batch_size=args.validation_batch_size or args.batch_size,
            is_training=False,
            interpolation=data_config['interpolation'],
            mean=data_config['mean'],
            std=data_config['std'],
            num_workers=args.workers,
            distributed=args.distributed,
            crop_pct=data_config['crop_pct'],
            pin_memory=args.pin_mem,
            device=device,
            use_prefetcher=args.prefetcher,
        )

    # setup checkpoint saver and eval metric tracking
    eval_metric = args.eval_metric
    decreasing_metric = eval_metric == 'loss'
    best_metric = None
    best_epoch = None
    saver = None
    output_dir = None
    if utils.is_primary(args):
        if args.experiment:
            exp_name = args.experiment
        else:
            exp_name = '-'.join([
                datetime.now().strftime("%Y%m%d-%H%M%S"),
                safe_model_name(args.model),
                str(data_config['input_size'][-1])
            ])
        output_dir = utils.get_outdir(args.output if args.output else './output/train', exp_name)
        saver = utils.CheckpointSaver(
            model=task.model,
            optimizer=optimizer,
            args=args,
            model_ema=task.model_ema,
            amp_scaler=loss_scaler,
            checkpoint_dir=output_dir,
            recovery_dir=output_dir,
            decreasing=decreasing_metric,
            max_history=args.checkpoint_hist,
        )
        with open(os.path.join(output_dir, 'args.yaml'), 'w') as f:
            f.write(args_text)

    # auto-scale LR from global batch size
    if args.lr is None:
        global_batch_size = args.batch_size * args.world_size * args.grad_accum_steps
        batch_ratio = global_batch_size / args.lr_base_size
        if not args.lr_base_scale:
            on = args.opt.lower()
            args.lr_base_scale = 'sqrt' if any([o in on for o in ('ada', 'lamb', 'lars')]) else 'linear'
        if args.lr_base_scale == 'sqrt':
            batch_ratio = batch_ratio ** 0.5
        args.lr = args.lr_base * batch_ratio
        if utils.is_primary(args):
            _logger.info(
                f'Learning rate ({args.lr}) calculated from base learning rate ({args.lr_base}) '
                f'and effective global batch size ({global_batch_size}) with {args.lr_base_scale} scaling.')
        for g in optimizer.param_groups:
            g['lr'] = args.lr
            g['initial_lr'] = args.lr

    # setup learning rate schedule and starting epoch
    updates_per_epoch = (len(loader_train) + args.grad_accum_steps - 1) // args.grad_accum_steps
    lr_scheduler, num_epochs = create_scheduler_v2(
        optimizer,
        **scheduler_kwargs(args, decreasing_metric=decreasing_metric),
        updates_per_epoch=updates_per_epoch,
    )
    start_epoch = 0
    if args.start_epoch is not None:
        # a specified start_epoch will always override the resume epoch
        start_epoch = args.start_epoch
    elif resume_epoch is not None:
        start_epoch = resume_epoch
    if lr_scheduler is not None and start_epoch > 0:
        if args.sched_on_updates:
            lr_scheduler.step_update(start_epoch * updates_per_epoch)
        else:
            lr_scheduler.step(start_epoch)

    if utils.is_primary(args):
        _logger.info(
            f'Scheduled epochs: {num_epochs}. '
            f'LR stepped per {"epoch" if lr_scheduler and lr_scheduler.t_in_epochs else "update"}.')

    results = []
    try:
        scheduled_batch_mode = args.train_img_sizes is not None
        batch_size_reference = float(args.batch_size)
        def _sched_sampler(loader):
            inner = getattr(loader, 'loader', loader)  # unwrap PrefetchLoader
            return inner.batch_sampler
        if scheduled_batch_mode:
            batch_size_reference = _sched_sampler(loader_train).average_batch_size
            if utils.is_primary(args):
                _logger.info(
                    f'Scheduled training resolutions {args.train_img_sizes} '
                    f'batch sizes {_sched_sampler(loader_train).batch_sizes} '
                    f'({len(_sched_sampler(loader_train))} batches/epoch).')

        for epoch in range(start_epoch, num_epochs):
            if hasattr(dataset_train, 'set_epoch'):
                dataset_train.set_epoch(epoch)
            elif args.distributed and hasattr(loader_train.sampler, 'set_epoch'):
                loader_train.sampler.set_epoch(epoch)
            if scheduled_batch_mode:
                _sched_sampler(loader_train).set_epoch(epoch)
            if mixup_fn is not None and args.mixup_off_epoch and epoch >= args.mixup_off_epoch:
                mixup_fn.mixup_enabled = False

            train_metrics = train_one_epoch(
                epoch,
                task,
                loader_train,
                optimizer,
                args,
                lr_scheduler=lr_scheduler,
                saver=saver,
                output_dir=output_dir,
                amp_autocast=amp_autocast,
                loss_scaler=loss_scaler,
                mixup_fn=mixup_fn,
                num_updates_total=num_epochs * updates_per_epoch,
                scheduled_batch_mode=scheduled_batch_mode,
                batch_size_reference=batch_size_reference,
            )

            if args.distributed and args.dist_bn in ('broadcast', 'reduce'):
                if utils.is_primary(args):
                    _logger.info("Distributing BatchNorm running means and vars")
                utils.distribute_bn(task.model, args.world_size, args.dist_bn == 'reduce')

            eval_metrics = validate(
                task.model,
                loader_eval,
                validate_loss_fn,
                args,
                device=device,
                amp_autocast=amp_autocast,
            )

            if task.model_ema is not None and not args.model_ema_force_cpu:
                if args.distributed and args.dist_bn in ('broadcast', 'reduce'):
                    utils.distribute_bn(task.model_ema, args.world_size, args.dist_bn == 'reduce')
                ema_eval_metrics = validate(
                    task.model_ema,
                    loader_eval,
                    validate_loss_fn,
                    args,
                    device=device,
                    amp_autocast=amp_autocast,
                    log_suffix=' (EMA)',
                )
                eval_metrics = ema_eval_metrics

            lrs = [param_group['lr'] for param_group in optimizer.param_groups]

            if output_dir is not None:
                utils.update_summary(
                    epoch,
                    train_metrics,
                    eval_metrics,
                    filename=os.path.join(output_dir, 'summary.csv'),
                    lr=sum(lrs) / len(lrs),
                    write_header=best_metric is None,
                    log_wandb=args.log_wandb,
                )

            if eval_metrics is not None:
                latest_metric = eval_metrics[eval_metric]
            else:
                latest_metric = train_metrics[eval_metric]

            if saver is not None:
                # save proper checkpoint with eval metric
                best_metric, best_epoch = saver.save_checkpoint(epoch, metric=latest_metric)

            if lr_scheduler is not None:
                # step LR for next epoch
                lr_scheduler.step(epoch + 1, latest_metric)

            results.append({
                'epoch': epoch,
                'train': train_metrics,
                'validation': eval_metrics,
            })

    except KeyboardInterrupt:
        pass

    if args.distributed:
        torch.distributed.destroy_process_group()

    results = {'all': results}
    if best_metric is not None:
        results['best'] = results['all'][best_epoch - start_epoch]
        _logger.info('*** Best metric: {0} (epoch {1})'.format(best_metric, best_epoch))
    if utils.is_primary(args):
        print(f'--result\n{json.dumps(results, indent=4)}')


def train_one_epoch(
        epoch,
        task,
        loader,
        optimizer,
        args,
        device=None,
        lr_scheduler=None,
        saver=None,
        output_dir=None,
        amp_autocast=suppress,
        loss_scaler=None,
        mixup_fn=None,
        scheduled_batch_mode=False,
        batch_size_reference=None,
        num_updates_total=None,
):
    device = device or torch.device(args.device)
    second_order = hasattr(optimizer, 'is_second_order') and optimizer.is_second_order
    has_no_sync = hasattr(task, 'no_sync')
    update_time_m = utils.AverageMeter()
    data_time_m = utils.AverageMeter()
    losses_m = utils.AverageMeter()

    task.train()
    model = task.train_model

    accum_steps = args.grad_accum_steps
    last_accum_steps = len(loader) % accum_steps
    updates_per_epoch = (len(loader) + accum_steps - 1) // accum_steps
    num_updates = epoch * updates_per_epoch
    last_batch_idx = len(loader) - 1
    last_batch_idx_to_accum = len(loader) - last_accum_steps

    data_start_time = update_start_time = time.time()
    optimizer.zero_grad()
    if task._distributed_model is not None:
        # zero_grad(set_to_none=True) detaches p.grad from the flat bucket
        # views; re-attach so the first backward accumulates into the buckets
        task._distributed_model.zero_grad_buckets()
    update_sample_count = 0
    for batch_idx, (input, target) in enumerate(loader):
        if args.max_steps_per_epoch and batch_idx >= args.max_steps_per_epoch:
            break
        last_batch = batch_idx == last_batch_idx
        need_update = last_batch or (batch_idx + 1) % accum_steps == 0
        update_idx = batch_idx // accum_steps
        if batch_idx >= last_batch_idx_to_accum:
            accum_steps = last_accum_steps

        naflex_batch = isinstance(input, dict)
        if not args.prefetcher:
            if naflex_batch:
                input = {k: v.to(device) if isinstance(v, torch.Tensor) else v
                         for k, v in input.items()}
                target = target.to(device)
            else:
                input, target = input.to(device), target.to(device)
                if mixup_fn is not None:
                    input, target = mixup_fn(input, target)
        if args.channels_last and not naflex_batch:
            input = input.contiguous(memory_format=torch.channels_last)

        # multiply by accum steps to get equivalent for full update
        data_time_m.update(accum_steps * (time.time() - data_start_time))

        def _forward():
            with amp_autocast():
                out = task(input, target)
                _loss = out['loss']
            if accum_steps > 1:
                _loss /= accum_steps
            return _loss

        def _backward(_loss):
            if loss_scaler is not None:
                loss_scaler(
                    _loss,
                    optimizer,
                    clip_grad=args.clip_grad,
                    clip_mode=args.clip_mode,
                    parameters=task.model.parameters(),
                    create_graph=second_order,
                    need_update=need_update,
                    pre_step_fn=task.finish_gradient_sync if need_update else None,
                )
            else:
                _loss.backward(create_graph=second_order)
                if need_update:
                    task.finish_gradient_sync()
                    if args.clip_grad is not None:
                        dispatch_clip_grad(
                            task.model.parameters(),
                            value=args.clip_grad,
                            mode=args.clip_mode,
                        )
                    optimizer.step()

        batch_size = input['patches'].shape[0] if naflex_batch else input.size(0)

        # variable-batch loss rescale (reference train.py:1334-1370): NaFlex
        # batch sizes differ per bucket AND per rank, so the local loss is
        # scaled to the reference batch size and re-balanced across ranks by
        # an all-reduced global batch size.
        loss_mult = 1.0
        if naflex_batch:
            scale_mode = args.naflex_loss_scale
            if scale_mode and scale_mode != 'none':
                loss_mult = batch_size / float(args.batch_size)
                if scale_mode == 'sqrt':
                    loss_mult = loss_mult ** 0.5
            if args.distributed:
                global_bs = utils.reduce_tensor(
                    torch.tensor(batch_size, device=device, dtype=torch.float32), 1)
                loss_mult = loss_mult * args.world_size * batch_size / global_bs.item()
        elif scheduled_batch_mode:
            # schedule is identical on every rank (seed-derived), so a local
            # rescale to the policy-average batch size is enough
            scale_mode = args.variable_batch_loss_scale
            if scale_mode and scale_mode != 'none':
                ref = batch_size_reference or float(args.batch_size)
                loss_mult = batch_size / ref
                if scale_mode == 'sqrt':
                    loss_mult = loss_mult ** 0.5

        if has_no_sync and not need_update:
            with task.no_sync():
                loss = _forward()
                _backward(loss * loss_mult if loss_mult != 1.0 else loss)
        else:
            loss = _forward()
            _backward(loss * loss_mult if loss_mult != 1.0 else loss)

        losses_m.update(loss.item() * accum_steps, batch_size)
        update_sample_count += batch_size

        if not need_update:
            data_start_time = time.time()
            continue

        num_updates += 1
        optimizer.zero_grad()
        if task._distributed_model is not None:
            task._distributed_model.zero_grad_buckets()
        task.update_ema(step=num_updates)

        if args.synchronize_step and device.type == 'cuda':
            torch.cuda.synchronize()
        time_now = time.time()
        update_time_m.update(time.time() - update_start_time)
        update_start_time = time_now

        if update_idx % args.log_interval == 0:
            lrl = [param_group['lr'] for param_group in optimizer.param_groups]
            lr = sum(lrl) / len(lrl)

            if args.distributed:
                reduced_loss = utils.reduce_tensor(loss.data, args.world_size)
                losses_m.update(reduced_loss.item() * accum_steps, batch_size)
                update_sample_count *= args.world_size

            if utils.is_primary(args):
                _logger.info(
                    f'Train: {epoch} [{update_idx:>4d}/{updates_per_epoch} '
                    f'({100. * (update_idx + 1) / updates_per_epoch:>3.0f}%)]  '
                    f'Loss: {losses_m.val:#.3g} ({losses_m.avg:#.3g})  '
                    f'Time: {update_time_m.val:.3f}s, {update_sample_count / update_time_m.val:>7.2f}/s  '
                    f'({update_time_m.avg:.3f}s, {update_sample_count / update_time_m.avg:>7.2f}/s)  '
                    f'LR: {lr:.3e}  '
                    f'Data: {data_time_m.val:.3f} ({data_time_m.avg:.3f})'
                )

        if saver is not None and args.recovery_interval and (
                (update_idx + 1) % args.recovery_interval == 0):
            saver.save_recovery(epoch, batch_idx=update_idx)

        if lr_scheduler is not None:
            lr_scheduler.step_update(num_updates=num_updates, metric=losses_m.avg)

        update_sample_count = 0
        data_start_time = time.time()

    loss_avg = losses_m.avg

    return OrderedDict([('loss', loss_avg)])


def validate(
        model,
        loader,
        loss_fn,
        args,
        device=None,
        amp_autocast=suppress,
        log_suffix='',
):
    device = device or torch.device(args.device)
    batch_time_m = utils.AverageMeter()
    losses_m = utils.AverageMeter()
    top1_m = utils.AverageMeter()
    top5_m = utils.AverageMeter()

    model.eval()

    end = time.time()
    last_idx = len(loader) - 1
    with torch.no_grad():
        for batch_idx, (input, target) in enumerate(loader):
            last_batch = batch_idx == last_idx
            naflex_batch = isinstance(input, dict)
            if not args.prefetcher:
                if naflex_batch:
                    input = {k: v.to(device) if isinstance(v, torch.Tensor) else v
                             for k, v in input.items()}
                else:
                    input = input.to(device)
                target = target.to(device)
            if args.channels_last and not naflex_batch:
                input = input.contiguous(memory_format=torch.channels_last)

            with amp_autocast():
                output = model(input)
                if isinstance(output, (tuple, list)):
                    output = output[0]

                loss = loss_fn(output, target)
            acc1, acc5 = utils.accuracy(output, target, topk=(1, 5))

            if args.distributed:
                reduced_loss = utils.reduce_tensor(loss.data, args.world_size)
                acc1 = utils.reduce_tensor(acc1, args.world_size)
                acc5 = utils.reduce_tensor(acc5, args.world_size)
            else:
                reduced_loss = loss.data

            if device.type == 'cuda':
                torch.cuda.synchronize()

            losses_m.update(
                reduced_loss.item(),
                input['patches'].shape[0] if naflex_batch else input.size(0))
            top1_m.update(acc1.item(), output.size(0))
            top5_m.update(acc5.item(), output.size(0))

            batch_time_m.update(time.time() - end)
            end = time.time()
            if utils.is_primary(args) and (last_batch or batch_idx % args.log_interval == 0):
                log_name = 'Test' + log_suffix
                _logger.info(
                    f'{log_name}: [{batch_idx:>4d}/{last_idx}]  '
                    f'Time: {batch_time_m.val:.3f} ({batch_time_m.avg:.3f})  '
                    f'Loss: {losses_m.val:>7.3f} ({losses_m.avg:>6.3f})  '
                    f'Acc@1: {top1_m.val:>7.3f} ({top1_m.avg:>7.3f})  '
                    f'Acc@5: {top5_m.val:>7.3f} ({top5_m.avg:>7.3f})'
                )

    metrics = OrderedDict([('loss', losses_m.avg), ('top1', top1_m.avg), ('top5', top5_m.avg)])

    return metrics


if __name__ == '__main__':
    main()
