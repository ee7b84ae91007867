"""Structured fit-stage logging + timers.

Replaces Spark's ``Instrumentation`` (``GaussianProcessCommons.scala:69,89,108``)
with a logging-module based recorder that also keeps per-stage wall-clock
timings (queriable after fit, printed by the bench harness and useful next to
rocprofv3 kernel traces)."""

from __future__ import annotations

import logging
from typing import Dict

logger = logging.getLogger("spark_gp_amd")


class Instrumentation:
    def __init__(self, name: str = "fit"):
        self.name = name
        self.timings: Dict[str, float] = {}

    def log(self, msg: str):
        logger.info("[%s] %s", self.name, msg)

    def log_timing(self, stage: str, seconds: float):
        self.timings[stage] = self.timings.get(stage, 0.0) + seconds
        logger.info("[%s] stage %s: %.3f s", self.name, stage, seconds)

    def log_success(self):
        logger.info("[%s] training finished successfully", self.name)
