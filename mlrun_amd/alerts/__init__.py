# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Client-side alerts package (reference mlrun/alerts)."""

from .alert import AlertConfig  # noqa: F401
