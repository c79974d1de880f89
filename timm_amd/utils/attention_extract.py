"""Extract attention maps (or arbitrary activations) by name
(reference `timm/utils/attention_extract.py:9`).

Two capture methods: 'hook' registers forward/forward-pre hooks on matching
module names; 'fx' traces the graph and taps matching node names (finer
grained — can capture functional ops like the softmax inside fused paths).
"""
import fnmatch
import re
from typing import List, Optional, Union

import torch


class AttentionExtract(torch.nn.Module):
    # cover the common timm attention-module naming
    default_node_names = ['*attn.softmax']
    default_module_names = ['*attn_drop']

    def __init__(
            self,
            model: Union[torch.nn.Module],
            names: Optional[List[str]] = None,
            mode: str = 'eval',
            method: str = 'fx',
            hook_type: str = 'forward',
            use_regex: bool = False,
    ):
        super().__init__()
        assert mode in ('train', 'eval')
        model = model.train() if mode == 'train' else model.eval()
        assert method in ('fx', 'hook')

        def match(candidates, patterns):
            if use_regex:
                regexes = [re.compile(r) for r in patterns]
                return [c for c in candidates if any(r.match(c) for r in regexes)]
            return [c for c in candidates if any(fnmatch.fnmatch(c, p) for p in patterns)]

        if method == 'fx':
            from ..models._features_fx import GraphExtractNet, get_graph_node_names

            node_names = get_graph_node_names(model)[0 if mode == 'train' else 1]
            matched = match(node_names, names or self.default_node_names)
            if not matched:
                raise RuntimeError(f'No node names found matching {names}.')
            self.model = GraphExtractNet(model, matched, return_dict=True)
            self.hooks = None
        else:
            assert hook_type in ('forward', 'forward_pre')
            from ..models._features import FeatureHooks

            module_names = [n for n, _ in model.named_modules()]
            matched = match(module_names, names or self.default_module_names)
            if not matched:
                raise RuntimeError(f'No module names found matching {names}.')
            self.model = model
            self.hooks = FeatureHooks(matched, model.named_modules(), default_hook_type=hook_type)

        self.names = matched
        self.mode = mode
        self.method = method

    def forward(self, x):
        if self.hooks is not None:
            self.model(x)
            return self.hooks.get_output(device=x.device)
        return self.model(x)
