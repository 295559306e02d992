from murmura_amd.config.schema import (
    AggregationConfig,
    AttackConfig,
    ComputeConfig,
    Config,
    DataConfig,
    DistributedConfig,
    DMTTConfig,
    ExperimentConfig,
    MobilityConfig,
    ModelConfig,
    TopologyConfig,
    TrainingConfig,
)
from murmura_amd.config.loader import load_config, save_config

__all__ = [
    "Config",
    "ExperimentConfig",
    "TopologyConfig",
    "AggregationConfig",
    "AttackConfig",
    "TrainingConfig",
    "DataConfig",
    "ModelConfig",
    "DistributedConfig",
    "MobilityConfig",
    "DMTTConfig",
    "ComputeConfig",
    "load_config",
    "save_config",
]
