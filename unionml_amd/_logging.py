"""Framework logger (reference: unionml/_logging.py:3-6)."""

import logging

logger = logging.getLogger("unionml_amd")
logger.setLevel(logging.INFO)

_handler = logging.StreamHandler()
_handler.setFormatter(logging.Formatter("[unionml_amd] %(levelname)s: %(message)s"))
logger.addHandler(_handler)
logger.propagate = False
