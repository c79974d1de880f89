"""torch.hub entrypoints: every registered model (reference `hubconf.py`)."""
dependencies = ['torch']

import timm_amd

globals().update(timm_amd.models._registry._model_entrypoints)
