"""Logging setup.  Behavioral parity: /root/reference/timm/utils/log.py:19
(bare INFO lines on console, timestamped rotating file log when requested).
"""
import logging
import logging.handlers

__all__ = ['FormatterNoInfo', 'setup_default_logging']


class FormatterNoInfo(logging.Formatter):
    """INFO records print bare; other levels keep the 'LEVEL: msg' prefix."""

    def __init__(self, fmt='%(levelname)s: %(message)s'):
        super().__init__(fmt)

    def format(self, record):
        if record.levelno == logging.INFO:
            return str(record.getMessage())
        return super().format(record)


def setup_default_logging(default_level=logging.INFO, log_path=''):
    console = logging.StreamHandler()
    console.setFormatter(FormatterNoInfo())
    root = logging.root
    root.addHandler(console)
    root.setLevel(default_level)
    if log_path:
        rotating = logging.handlers.RotatingFileHandler(
            log_path, maxBytes=10 * (1 << 20), backupCount=3)
        rotating.setFormatter(logging.Formatter(
            '%(asctime)s - %(name)20s: [%(levelname)8s] - %(message)s'))
        root.addHandler(rotating)
