from .config import (
    DotDict,
    apply_scaling_rules_to_cfg,
    get_default_config,
    load_yaml,
    setup_config,
    setup_job,
    exit_job,
    write_config,
)

__all__ = [
    "DotDict",
    "get_default_config",
    "load_yaml",
    "setup_config",
    "setup_job",
    "exit_job",
    "apply_scaling_rules_to_cfg",
    "write_config",
]
