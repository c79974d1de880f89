"""Class-name -> index map loading (.txt one-per-line or pickled dict).

Behavioral parity: /root/reference/timm/data/readers/class_map.py.
"""
import os
import pickle


def load_class_map(map_or_filename, root=''):
    if isinstance(map_or_filename, dict):
        assert map_or_filename, 'class_map dict must be non-empty'
        return map_or_filename
    path = map_or_filename
    if not os.path.exists(path):
        path = os.path.join(root, path)
        assert os.path.exists(path), \
            f'Cannot locate specified class map file ({map_or_filename})'
    ext = os.path.splitext(map_or_filename)[-1].lower()
    if ext == '.txt':
        with open(path) as f:
            return {line.strip(): idx for idx, line in enumerate(f)}
    if ext == '.pkl':
        with open(path, 'rb') as f:
            return pickle.load(f)
    raise AssertionError(f'Unsupported class map file extension ({ext}).')
