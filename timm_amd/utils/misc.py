"""Small shared utilities (reference `timm/utils/misc.py`)."""
import argparse
import ast
import re

_NUM_SPLIT = re.compile(r'(\d+)')


def natural_key(string_):
    """Sort key treating digit runs as numbers ('b2' < 'b10')."""
    return [int(part) if part.isdigit() else part for part in _NUM_SPLIT.split(string_.lower())]


def add_bool_arg(parser, name, default=False, help=''):
    """Register a --name / --no-name flag pair."""
    dest = name.replace('-', '_')
    group = parser.add_mutually_exclusive_group(required=False)
    group.add_argument('--' + name, dest=dest, action='store_true', help=help)
    group.add_argument('--no-' + name, dest=dest, action='store_false', help='')
    parser.set_defaults(**{dest: default})


class ParseKwargs(argparse.Action):
    """argparse action for free-form `key=value` lists (`--model-kwargs a=1 b=c`)."""

    def __call__(self, parser, namespace, values, option_string=None):
        parsed = {}
        for item in values:
            key, _, raw = item.partition('=')
            try:
                parsed[key] = ast.literal_eval(raw)
            except (ValueError, SyntaxError):
                parsed[key] = raw  # plain string; avoids shell-escaping quotes
        setattr(namespace, self.dest, parsed)
