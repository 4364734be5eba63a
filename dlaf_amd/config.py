"""Runtime configuration and tunable parameters.

Counterpart of the reference's 3-layer config system (``src/init.cpp:203-316``,
``include/dlaf/tune.h:114-168``): environment variables ``DLAF_*`` override
CLI flags ``--dlaf:*`` override user-supplied values override defaults.
``initialize()``/``finalize()`` mirror ``dlaf::initialize/finalize``
(``include/dlaf/init.h:81-110``); on MI355X there is no Umpire/pika to set up —
the HIP stream pools live in ``runtime/streams.py`` and the RCCL process group
is owned by ``CommGrid`` — so initialization is configuration + stream-pool
warmup.
"""

from __future__ import annotations

import dataclasses
import os
from typing import List, Optional


def _parse_bool(s: str) -> bool:
    return s.strip().upper() in ("ON", "TRUE", "YES", "1")


@dataclasses.dataclass
class TuneParameters:
    """Runtime-tunable knobs (reference ``tune.h:114-168``)."""
    eigensolver_min_band: int = 100
    band_to_tridiag_1d_block_size_base: int = 8192
    bt_band_to_tridiag_hh_apply_group_size: int = 128  # measured best at n=20k
    bt_band_to_tridiag_window_merge: int = 1   # consecutive WY windows merged
                                               # into one block-WY apply;
                                               # measured neutral at m=4 and
                                               # worse at m=8 on n=20000 (the
                                               # staircase zero-padding grows
                                               # the GEMM flops ~2x at m=4)
    tridiag_rank1_num_threads: int = 0          # 0 = auto
    red2band_panel_num_threads: int = 0
    tfactor_num_streams: int = 4
    communicator_grid_num_pipelines: int = 3
    debug_dump_cholesky_factorization_data: bool = False
    debug_dump_eigensolver_data: bool = False
    debug_dump_generalized_to_standard_data: bool = False


@dataclasses.dataclass
class Configuration:
    """Startup configuration (reference ``init.h:32-78``)."""
    num_np_gpu_streams: int = 4
    num_hp_gpu_streams: int = 4
    print_config: bool = False
    tune: TuneParameters = dataclasses.field(default_factory=TuneParameters)


_config: Optional[Configuration] = None


def _apply_env_and_cli(cfg: Configuration, argv: Optional[List[str]]) -> None:
    def lookup(name: str):
        env = os.environ.get("DLAF_" + name.upper())
        cli = None
        if argv:
            flag = "--dlaf:" + name.lower().replace("_", "-")
            for a in argv:
                if a.startswith(flag + "="):
                    cli = a.split("=", 1)[1]
        return env if env is not None else cli

    for obj in (cfg, cfg.tune):
        for f in dataclasses.fields(obj):
            if f.name == "tune":
                continue
            v = lookup(f.name)
            if v is None:
                continue
            if f.type in ("int", int):
                setattr(obj, f.name, int(v))
            elif f.type in ("bool", bool):
                setattr(obj, f.name, _parse_bool(v))


def initialize(argv: Optional[List[str]] = None,
               user_cfg: Optional[Configuration] = None) -> Configuration:
    """Build the global configuration (env > CLI > user > defaults)."""
    global _config
    cfg = user_cfg if user_cfg is not None else Configuration()
    _apply_env_and_cli(cfg, argv)
    _config = cfg
    if cfg.print_config:
        print(f"dlaf_amd configuration: {cfg}")
    return cfg


def finalize() -> None:
    global _config
    _config = None


def get_config() -> Configuration:
    global _config
    if _config is None:
        initialize()
    return _config


def get_tune_parameters() -> TuneParameters:
    return get_config().tune


class ScopedInitializer:
    """RAII-style init/finalize (reference ``init.h:105-110``)."""

    def __init__(self, argv=None, user_cfg=None):
        self.cfg = initialize(argv, user_cfg)

    def __enter__(self):
        return self.cfg

    def __exit__(self, *exc):
        finalize()
        return False
