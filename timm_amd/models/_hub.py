"""Hugging Face Hub / url weight IO (reference `timm/models/_hub.py`, 601 LoC).

Offline-first: everything network-touching is wrapped so a no-egress
environment degrades to local files / cached dirs.
"""
import json
import logging
import os
from functools import partial
from pathlib import Path
from typing import Any, Dict, Iterable, Optional, Tuple, Union

import torch
from torch.hub import HASH_REGEX, download_url_to_file, urlparse

try:
    from torch.hub import get_dir
except ImportError:
    from torch.hub import _get_torch_home as get_dir

try:
    import safetensors.torch
    _has_safetensors = True
except ImportError:
    _has_safetensors = False

try:
    from huggingface_hub import hf_hub_download, HfApi
    from huggingface_hub.utils import EntryNotFoundError
    hf_hub_download = partial(hf_hub_download, library_name="timm_amd")
    _has_hf_hub = True
except ImportError:
    hf_hub_download = None
    _has_hf_hub = False

_logger = logging.getLogger(__name__)

__all__ = [
    'get_cache_dir', 'download_cached_file', 'has_hf_hub', 'hf_split', 'load_model_config_from_hf',
    'load_model_config_from_path', 'load_state_dict_from_hf', 'save_for_hf', 'push_to_hf_hub',
]

# Default name for a weights file hosted on the Huggingface Hub.
HF_WEIGHTS_NAME = "pytorch_model.bin"  # default pytorch pkl
HF_SAFE_WEIGHTS_NAME = "model.safetensors"  # safetensors version
HF_OPEN_CLIP_WEIGHTS_NAME = "open_clip_pytorch_model.bin"  # default pytorch pkl
HF_OPEN_CLIP_SAFE_WEIGHTS_NAME = "open_clip_model.safetensors"  # safetensors version


def get_cache_dir(child_dir: str = ''):
    """Get the cache dir used for caching downloaded weights."""
    if os.getenv('TORCH_MODEL_ZOO'):
        _logger.warning('TORCH_MODEL_ZOO is deprecated, please use env TORCH_HOME instead')

    hub_dir = get_dir()
    child_dir = () if not child_dir else (child_dir,)
    model_dir = os.path.join(hub_dir, 'checkpoints', *child_dir)
    os.makedirs(model_dir, exist_ok=True)
    return model_dir


def download_cached_file(url, check_hash=True, progress=False, cache_dir=None):
    if isinstance(url, (list, tuple)):
        url, filename = url
    else:
        parts = urlparse(url)
        filename = os.path.basename(parts.path)
    if cache_dir:
        os.makedirs(cache_dir, exist_ok=True)
    else:
        cache_dir = get_cache_dir()
    cached_file = os.path.join(cache_dir, filename)
    if not os.path.exists(cached_file):
        _logger.info('Downloading: "{}" to {}\n'.format(url, cached_file))
        hash_prefix = None
        if check_hash:
            r = HASH_REGEX.search(filename)  # r is Optional[Match[str]]
            hash_prefix = r.group(1) if r else None
        download_url_to_file(url, cached_file, hash_prefix, progress=progress)
    return cached_file


def has_hf_hub(necessary: bool = False):
    if not _has_hf_hub and necessary:
        # if no HF Hub module installed, and it is necessary to continue, raise error
        raise RuntimeError(
            'Hugging Face hub model specified but package not installed. Run `pip install huggingface_hub`.')
    return _has_hf_hub


def hf_split(hf_id: str):
    # FIXME I may change @ -> # and be parsed as fragment in a URI model name scheme
    rev_split = hf_id.split('@')
    assert 0 < len(rev_split) <= 2, 'hf_hub id should only contain one @ character to identify revision.'
    hf_model_id = rev_split[0]
    hf_revision = rev_split[-1] if len(rev_split) > 1 else None
    return hf_model_id, hf_revision


def load_cfg_from_json(json_file: Union[str, Path]):
    with open(json_file, "r", encoding="utf-8") as reader:
        text = reader.read()
    return json.loads(text)


def download_from_hf(model_id: str, filename: str, cache_dir: Optional[str] = None):
    hf_model_id, hf_revision = hf_split(model_id)
    return hf_hub_download(hf_model_id, filename, revision=hf_revision, cache_dir=cache_dir)


def _parse_model_cfg(cfg: Dict[str, Any], extra_fields: Dict[str, Any]):
    """Smooth out nested or flat hf.co config structure into (pretrained_cfg, model_name, model_args)."""
    if 'pretrained_cfg' in cfg:
        # new form, both pretrained_cfg and model_args
        pretrained_cfg = cfg['pretrained_cfg']
        cfg.pop('pretrained_cfg')
    else:
        # old form, pull pretrain_cfg out of the base dict
        pretrained_cfg = cfg
        cfg = {}
    model_name = cfg.pop('architecture', pretrained_cfg.pop('architecture', None))
    model_args = cfg.pop('model_args', None)
    num_classes = cfg.pop('num_classes', pretrained_cfg.get('num_classes', None))
    if num_classes is not None:
        pretrained_cfg['num_classes'] = num_classes
    label_names = cfg.pop('label_names', None)
    if label_names:
        pretrained_cfg['label_names'] = label_names
    label_descriptions = cfg.pop('label_descriptions', None)
    if label_descriptions:
        pretrained_cfg['label_descriptions'] = label_descriptions
    pretrained_cfg.update(extra_fields)
    return pretrained_cfg, model_name, model_args


def load_model_config_from_hf(model_id: str, cache_dir: Optional[str] = None):
    has_hf_hub(True)
    cached_file = download_from_hf(model_id, 'config.json', cache_dir=cache_dir)
    cfg = load_cfg_from_json(cached_file)
    return _parse_model_cfg(cfg, {'hf_hub_id': model_id, 'source': 'hf-hub'})


def load_model_config_from_path(model_path: Union[str, Path]):
    model_path = Path(model_path)
    cfg_file = model_path / 'config.json'
    if not cfg_file.exists():
        raise FileNotFoundError(f'Config file not found at {cfg_file}')
    cfg = load_cfg_from_json(cfg_file)
    extra_fields = {'file': str(model_path), 'source': 'local-dir'}
    return _parse_model_cfg(cfg, extra_fields)


def load_state_dict_from_hf(
        model_id: str,
        filename: str = HF_WEIGHTS_NAME,
        weights_only: bool = False,
        cache_dir: Optional[str] = None,
):
    """Load weights from HF hub, preferring safetensors (reference `_hub.py:214`)."""
    has_hf_hub(True)
    hf_model_id, hf_revision = hf_split(model_id)

    # Look for .safetensors alternatives and load from it if it exists
    if _has_safetensors:
        for safe_filename in _get_safe_alternatives(filename):
            try:
                cached_safe_file = hf_hub_download(
                    repo_id=hf_model_id, filename=safe_filename, revision=hf_revision, cache_dir=cache_dir)
                _logger.info(
                    f"[{model_id}] Safe alternative available for '{filename}' "
                    f"(as '{safe_filename}'). Loading weights using safetensors.")
                return safetensors.torch.load_file(cached_safe_file, device="cpu")
            except EntryNotFoundError:
                pass

    # Otherwise, load using pytorch.load
    cached_file = hf_hub_download(hf_model_id, filename=filename, revision=hf_revision, cache_dir=cache_dir)
    _logger.debug(f"[{model_id}] Safe alternative not found for '{filename}'. Loading weights using default pytorch.")
    try:
        state_dict = torch.load(cached_file, map_location='cpu', weights_only=weights_only)
    except TypeError:
        state_dict = torch.load(cached_file, map_location='cpu')
    return state_dict


def _get_safe_alternatives(filename: str) -> Iterable[str]:
    """Return potential safetensors alternatives for a given filename."""
    if filename == HF_WEIGHTS_NAME:
        yield HF_SAFE_WEIGHTS_NAME
    if filename == HF_OPEN_CLIP_WEIGHTS_NAME:
        yield HF_OPEN_CLIP_SAFE_WEIGHTS_NAME
    if filename not in (HF_WEIGHTS_NAME, HF_OPEN_CLIP_WEIGHTS_NAME) and filename.endswith(".bin"):
        yield filename[:-4] + ".safetensors"


def save_config_for_hf(
        model: torch.nn.Module,
        config_path: str,
        model_config: Optional[Dict] = None,
        model_args: Optional[Dict] = None,
):
    model_config = model_config or {}
    hf_config: Dict[str, Any] = {}
    pretrained_cfg = getattr(model, 'pretrained_cfg', {})
    from ._pretrained import filter_pretrained_cfg
    pretrained_cfg = filter_pretrained_cfg(dict(pretrained_cfg), remove_source=True, remove_null=True)
    # set some values at root config level
    hf_config['architecture'] = pretrained_cfg.pop('architecture')
    hf_config['num_classes'] = model_config.pop('num_classes', model.num_classes)

    global_pool_type = model_config.pop('global_pool', getattr(model, 'global_pool', None))
    if isinstance(global_pool_type, str) and global_pool_type:
        hf_config['global_pool'] = global_pool_type

    hf_config['pretrained_cfg'] = pretrained_cfg
    hf_config.update(model_config)
    if model_args:
        hf_config['model_args'] = model_args

    with open(config_path, 'w') as f:
        json.dump(hf_config, f, indent=2)


def save_for_hf(
        model: torch.nn.Module,
        save_directory: str,
        model_config: Optional[Dict] = None,
        model_args: Optional[Dict] = None,
        safe_serialization: Union[bool, str] = 'both',
):
    """Save weights + config for HF hub layout (reference `_hub.py:378`)."""
    save_directory = Path(save_directory)
    save_directory.mkdir(exist_ok=True, parents=True)

    # Save model weights, either safely (using safetensors), or using legacy pytorch approach or both.
    tensors = model.state_dict()
    if safe_serialization is True or safe_serialization == 'both':
        assert _has_safetensors, "`pip install safetensors` to use .safetensors"
        safetensors.torch.save_file(tensors, str(save_directory / HF_SAFE_WEIGHTS_NAME))
    if safe_serialization is False or safe_serialization == 'both':
        torch.save(tensors, str(save_directory / HF_WEIGHTS_NAME))

    config_path = save_directory / 'config.json'
    save_config_for_hf(model, str(config_path), model_config=model_config, model_args=model_args)


def push_to_hf_hub(
        model: torch.nn.Module,
        repo_id: str,
        commit_message: str = 'Add model',
        token: Optional[str] = None,
        revision: Optional[str] = None,
        private: bool = False,
        create_pr: bool = False,
        model_config: Optional[Dict] = None,
        model_card: Optional[Dict] = None,
        model_args: Optional[Dict] = None,
        safe_serialization: Union[bool, str] = 'both',
):
    """Push model + config to HF hub (reference `_hub.py:406`)."""
    has_hf_hub(True)
    from huggingface_hub import create_repo, upload_folder
    import tempfile
    repo_url = create_repo(repo_id, token=token, private=private, exist_ok=True)
    repo_id = repo_url.repo_id

    with tempfile.TemporaryDirectory() as tmpdir:
        save_for_hf(
            model, tmpdir,
            model_config=model_config,
            model_args=model_args,
            safe_serialization=safe_serialization,
        )
        readme_path = Path(tmpdir) / "README.md"
        if not readme_path.exists():
            model_card = model_card or {}
            model_name = repo_id.split('/')[-1]
            readme_text = generate_readme(model_card, model_name)
            readme_path.write_text(readme_text)

        return upload_folder(
            repo_id=repo_id,
            folder_path=tmpdir,
            revision=revision,
            create_pr=create_pr,
            commit_message=commit_message,
        )


def generate_readme(model_card: dict, model_name: str):
    tags = model_card.get('tags', None) or ['image-classification', 'timm']
    readme_text = "---\n"
    if tags:
        readme_text += "tags:\n"
        for t in tags:
            readme_text += f"- {t}\n"
    readme_text += f"library_name: {model_card.get('library_name', 'timm')}\n"
    readme_text += f"license: {model_card.get('license', 'apache-2.0')}\n"
    readme_text += "---\n"
    readme_text += f"# Model card for {model_name}\n"
    if 'description' in model_card:
        readme_text += f"\n{model_card['description']}\n"
    return readme_text
