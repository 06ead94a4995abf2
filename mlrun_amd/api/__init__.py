# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""API service: FastAPI app + scheduler + alerts over the run DB."""

from .main import create_app, serve  # noqa: F401
from .scheduler import CronTrigger, Scheduler  # noqa: F401
from .events import AlertConfig, process_event  # noqa: F401
