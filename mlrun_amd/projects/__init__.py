# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
from .project import (  # noqa: F401
    MlrunProject,
    ProjectMetadata,
    ProjectSpec,
    get_or_create_project,
    load_project,
    new_project,
    pipeline_context,
)
from .operations import (  # noqa: F401
    build_function,
    deploy_function,
    run_function,
)
from .pipelines import load_and_run  # noqa: F401,E402
from .project import ProjectStatus  # noqa: F401,E402
