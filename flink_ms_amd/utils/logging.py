"""Logging setup (slf4j+log4j stand-in; reference configures log4j in every
pom, e.g. flink-als/pom.xml:99-108)."""
import logging
import os
import sys


def get_logger(name: str) -> logging.Logger:
    logger = logging.getLogger(name)
    if not logging.getLogger("flink_ms_amd").handlers:
        root = logging.getLogger("flink_ms_amd")
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(logging.Formatter(
            "%(asctime)s %(levelname)s %(name)s - %(message)s"))
        root.addHandler(h)
        root.setLevel(os.environ.get("FLINK_MS_AMD_LOGLEVEL", "INFO"))
    return logger
