"""Logging setup (reference dfd/timm/utils.py:343-357): root logger with a
bare-message INFO formatter."""

import logging


class FormatterNoInfo(logging.Formatter):
    def __init__(self, fmt="%(levelname)s: %(message)s"):
        super().__init__(fmt)

    def format(self, record):
        if record.levelno == logging.INFO:
            return str(record.getMessage())
        return super().format(record)


def setup_default_logging(default_level=logging.INFO):
    console_handler = logging.StreamHandler()
    console_handler.setFormatter(FormatterNoInfo())
    root = logging.getLogger()
    if not root.handlers:
        root.addHandler(console_handler)
    root.setLevel(default_level)
