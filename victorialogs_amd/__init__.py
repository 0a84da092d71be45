"""victorialogs_amd — MI355X-native engine for VictoriaLogs' block-scan hot path.

The compute path is the in-tree HIP library (libvlogsql.so, built for gfx950).
There is NO CPU fallback: loading the scan API on a machine without the
library, or scanning without a GPU, raises immediately.
"""

from .api import (  # noqa: F401
    Filter,
    Part,
    Stage,
    generate_part,
    lib_path,
    load_product,
    oracle_lib_path,
    oracle_helpers,
    write_custom_part,
    OracleScanner,
)
