"""Application configuration.

Single source of defaults + SAIL_-prefixed environment overrides — the
analogue of the reference's figment-loaded application.yaml
(ref: crates/sail-common/src/config/application.yaml, loader.rs).
"""
from __future__ import annotations

import os
from typing import Dict

DEFAULTS: Dict[str, str] = {
    # execution
    "sail.mode": "local",                       # local | spmd
    "sail.execution.device": "auto",            # auto | cpu | cuda[:N]
    "sail.execution.result_batch_rows": "65536",
    # optimizer (ref: application.yaml optimizer.*)
    "sail.optimizer.enable_join_reorder": "true",
    "sail.optimizer.join_reorder_max_relations": "12",
    "sail.optimizer.agg_mask_min_selectivity": "0.2",
    "sail.execution.checkpoint_path": "",       # df.checkpoint() root (temp dir when empty)
    "sail.execution.max_recursion": "100",      # WITH RECURSIVE iteration cap
    # distributed exchanges (SPMD over RCCL)
    "sail.exec.broadcast_threshold_bytes": str(2 << 30),  # gathered build side above this -> hash shuffle
    "sail.exec.agg_shuffle_threshold_groups": "4000000",  # est. groups above this -> shuffled aggregation
    # scan path: GPU parquet page decode (datasource/gpu_parquet.py)
    # auto = GPU decode when supported, host pyarrow otherwise;
    # force = raise instead of falling back (tests); off = host only
    "sail.io.gpu_parquet": "auto",
    # kernels
    "sail.kernels.require_on_gpu": "true",
    "sail.kernels.grouped_agg_max_lds_groups": "4096",
    # debugging (the compute-sanitizer analogue)
    "sail.debug.sync_kernels": "false",         # device sync + error check after every operator
    # tracing (ref: sail-telemetry)
    "sail.trace": "false",
    # spark-compatible session confs
    "spark.sql.session.timeZone": "UTC",
    "spark.sql.ansi.enabled": "false",
    "spark.sql.caseSensitive": "false",
    "spark.sql.shuffle.partitions": "1",
}


def load_config() -> Dict[str, str]:
    """Defaults overridden by SAIL_FOO_BAR env vars (SAIL_ prefix, _ -> .)."""
    conf = dict(DEFAULTS)
    for k, v in os.environ.items():
        if k.startswith("SAIL_") and k not in ("SAIL_TRACE",):
            key = "sail." + k[len("SAIL_"):].lower().replace("_", ".")
            conf[key] = v
    if os.environ.get("SAIL_TRACE") == "1":
        conf["sail.trace"] = "true"
    return conf
