from .catalog import (  # noqa: F401
    COMPUTE_MODES,
    MEMORY_MODES,
    PartitionProfile,
    gfx950_catalog,
    profiles_for,
    validate_mode_combo,
)
