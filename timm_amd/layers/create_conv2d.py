"""Conv2d factory (reference `timm/layers/create_conv2d.py`): one entry point
that picks MixedConv2d (per-group kernel sizes), CondConv2d (expert-routed
weights) or a plain / SAME-padded conv. EfficientNet-family builders call
this for every conv."""
from .cond_conv2d import CondConv2d
from .conv2d_same import create_conv2d_pad
from .mixed_conv2d import MixedConv2d


def create_conv2d(in_channels, out_channels, kernel_size, **kwargs):
    if isinstance(kernel_size, list):
        # a LIST of kernel sizes selects MixedConv2d; ints/tuples stay on the
        # plain path. Mixed + CondConv is not a supported combination.
        assert 'num_experts' not in kwargs
        if 'groups' in kwargs:
            groups = kwargs.pop('groups')
            if groups == in_channels:
                kwargs['depthwise'] = True
            else:
                assert groups == 1
        return MixedConv2d(in_channels, out_channels, kernel_size, **kwargs)

    depthwise = kwargs.pop('depthwise', False)
    # depthwise requires out % in == 0, expressed via groups == in_channels
    groups = in_channels if depthwise else kwargs.pop('groups', 1)
    if kwargs.get('num_experts', 0) > 0:
        return CondConv2d(in_channels, out_channels, kernel_size, groups=groups, **kwargs)
    return create_conv2d_pad(in_channels, out_channels, kernel_size, groups=groups, **kwargs)
