"""Checkpoint interoperability proof vs the reference implementation.

For one model per flagship family: build the reference model (imported from
/root/reference with torchvision stubs), load its random-init state dict into
our model by exact key match, and assert logits agree.  This is the strongest
possible "timm checkpoint format" guarantee available offline (no network for
real pretrained weights): identical key sets, identical tensor shapes, and
identical math given identical weights.

Reference test style: /root/reference/tests/test_models.py:132-173.
"""
import os
import sys

import pytest
import torch

import timm_amd

REF_PATH = '/root/reference'
TOOLS = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), 'tools')

pytestmark = pytest.mark.skipif(
    not os.path.isdir(os.path.join(REF_PATH, 'timm')),
    reason='reference checkout not available (GPU box)')


@pytest.fixture(scope='module')
def ref_timm():
    sys.path.insert(0, TOOLS)
    from ref_import import load_reference
    return load_reference()


# (model, input size) — small/flagship variant per family
INTEROP_MODELS = [
    ('vit_tiny_patch16_224', 224),
    ('vit_base_patch16_224', 224),
    ('deit_tiny_distilled_patch16_224', 224),
    ('eva02_tiny_patch14_224', 224),
    ('beit_base_patch16_224', 224),
    ('convnext_atto', 128),
    ('convnextv2_atto', 128),
    ('swin_tiny_patch4_window7_224', 224),
    ('swinv2_tiny_window8_256', 256),
    ('aimv2_large_patch14_224', 224),
    ('flexivit_small', 240),
    ('beit3_base_patch16_224', 224),
    ('test_vit3', 160),
    ('mobilenetv3_small_050', 128),
    ('coatnet_nano_rw_224', 224),
    ('coatnet_0_rw_224', 224),
    ('maxvit_rmlp_nano_rw_256', 256),
    ('maxvit_tiny_pm_256', 256),
    ('coatnext_nano_rw_224', 224),
    ('maxxvitv2_nano_rw_256', 256),
    ('mobileone_s0', 128),
    ('regnetz_005', 128),
    ('regnetz_b16', 128),
    ('resnet50_clip_gap', 224),
    ('nf_regnet_b0', 128),
    ('efficientnet_b0', 128),
    ('mobilenetv3_small_100', 128),
    ('mobilenetv4_conv_small', 128),
    ('resnet18', 128),
    ('resnet50', 128),
    ('resnetv2_50', 128),
    ('regnety_002', 128),
    ('hiera_tiny_224', 224),
    ('mixer_s16_224', 224),
    ('vgg11', 128),
    ('densenet121', 128),
    ('nfnet_l0', 224),
    ('ghostnet_100', 128),
    ('repvgg_a0', 128),
    ('xcit_nano_12_p16_224', 224),
    ('pvt_v2_b0', 128),
    ('coat_tiny', 224),
    ('levit_128s', 224),
    ('efficientformer_l1', 224),
    ('poolformer_s12', 128),
    ('gcvit_xxtiny', 224),
    ('focalnet_tiny_srf', 128),
    ('edgenext_xx_small', 128),
    ('cait_xxs24_224', 224),
    ('convmixer_768_32', 224),
    ('dpn68', 128),
    ('hardcorenas_a', 128),
]


@pytest.mark.parametrize('model_name,img_size', INTEROP_MODELS)
def test_state_dict_and_logit_interop(ref_timm, model_name, img_size):
    if not ref_timm.is_model(model_name):
        pytest.skip(f'{model_name} not in reference registry')
    if not timm_amd.is_model(model_name):
        pytest.fail(f'{model_name} missing from timm_amd registry')

    torch.manual_seed(42)
    ref_model = ref_timm.create_model(model_name, num_classes=10)
    our_model = timm_amd.create_model(model_name, num_classes=10)

    ref_sd = ref_model.state_dict()
    our_sd = our_model.state_dict()
    missing = sorted(set(ref_sd) - set(our_sd))
    extra = sorted(set(our_sd) - set(ref_sd))
    assert not missing and not extra, \
        f'state dict keys differ: missing={missing[:5]} extra={extra[:5]}'
    shape_diff = [
        (k, tuple(ref_sd[k].shape), tuple(our_sd[k].shape))
        for k in ref_sd if ref_sd[k].shape != our_sd[k].shape]
    assert not shape_diff, f'shapes differ: {shape_diff[:5]}'

    our_model.load_state_dict(ref_sd)
    ref_model.eval()
    our_model.eval()
    x = torch.randn(2, 3, img_size, img_size)
    with torch.no_grad():
        ref_out = ref_model(x)
        our_out = our_model(x)
    if isinstance(ref_out, (tuple, list)):
        ref_out, our_out = ref_out[0], our_out[0]
    err = (ref_out.float() - our_out.float()).abs().max().item()
    assert err < 1e-4, f'logit mismatch {err}'
