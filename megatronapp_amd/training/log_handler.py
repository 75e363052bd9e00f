"""Logging handler filtering third-party noise (reference
training/log_handler.py): framework logs go to stdout, chatty
torch.distributed records are dropped."""

import sys
from logging import LogRecord, StreamHandler

BLACKLISTED_MODULES = ["torch.distributed"]


class CustomHandler(StreamHandler):
    def __init__(self):
        super().__init__(stream=sys.stdout)

    def filter(self, record: LogRecord) -> bool:
        return not any(record.name.startswith(m)
                       for m in BLACKLISTED_MODULES)
