"""Config layer: strict YAML schemas + typed settings accessors.

The spine of the framework: mirrors the
reference's 8 config families (credentials/config/pool/jobs/fs/monitor/
federation/slurm) with an MI355X-local surface."""

from .loader import (ConfigBundle, ConfigType, SchemaViolation,  # noqa
                     load_config_file, resolve_config_files, validate_config)
from . import settings  # noqa: F401
