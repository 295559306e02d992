"""Pydantic config schema — reference-compatible YAML surface.

The YAML schema is an explicit compatibility surface (BASELINE.json); keys and
defaults follow the reference schema (reference: murmura/config/schema.py:7-202)
with additive MI355X extensions (``backend: rccl``, per-node device pinning,
compute dtype). ``extra="forbid"`` everywhere, like the reference.
"""

from __future__ import annotations

from typing import Any, Dict, Literal, Optional

from pydantic import BaseModel, ConfigDict, Field, model_validator


class _Strict(BaseModel):
    model_config = ConfigDict(extra="forbid")


class ExperimentConfig(_Strict):
    name: str = "experiment"
    seed: int = 42
    rounds: int = 20
    verbose: bool = True


class TopologyConfig(_Strict):
    type: Literal[
        "ring", "fully", "full", "erdos", "er", "erdos-renyi", "k-regular", "kregular"
    ] = "ring"
    num_nodes: int = 10
    p: float = 0.3  # erdos edge probability
    k: int = 4  # k-regular degree
    seed: int = 12345


class AggregationConfig(_Strict):
    algorithm: Literal[
        "fedavg", "krum", "balance", "sketchguard", "ubar", "evidential_trust"
    ] = "fedavg"
    params: Dict[str, Any] = Field(default_factory=dict)


class AttackConfig(_Strict):
    enabled: bool = False
    type: Literal["gaussian", "directed_deviation", "topology_liar"] = "gaussian"
    percentage: float = 0.0
    params: Dict[str, Any] = Field(default_factory=dict)


class TrainingConfig(_Strict):
    local_epochs: int = 1
    batch_size: int = 64
    lr: float = 0.01
    max_samples: Optional[int] = None


class DataConfig(_Strict):
    adapter: str = "synthetic"
    params: Dict[str, Any] = Field(default_factory=dict)


class ModelConfig(_Strict):
    factory: str = "models.mlp"
    params: Dict[str, Any] = Field(default_factory=dict)


class MobilityConfig(_Strict):
    """Bounded random walk on a 2-D torus; deterministic from seed so every rank
    computes identical G^t without communication (reference: topology/dynamic.py:1-8)."""

    area_size: float = 100.0
    comm_range: float = 30.0
    max_speed: float = 5.0
    seed: int = 42
    ensure_connected: bool = True


class DMTTConfig(_Strict):
    """Dynamic trust-protocol parameters (reference: config/schema.py:114-139)."""

    budget_B: int = 5
    rho: float = 0.1  # link-reliability EMA rate
    lambda_forget: float = 0.9  # Beta-evidence forgetting
    w_d: float = 1.0  # direct-evidence weight
    w_c: float = 0.5  # corroboration weight
    w_x: float = 1.0  # contradiction weight
    tau_U: float = 0.3  # uncertainty threshold for topology trust
    eta: float = 5.0  # uncertainty penalty sharpness
    w_a: float = 0.7  # accuracy weight in model score
    tau_u: float = 0.5  # vacuity threshold in model score
    lambda1: float = 0.4  # model-score weight in collaborator score
    lambda2: float = 0.3  # topology-trust weight
    lambda3: float = 0.2  # link-reliability weight
    lambda4: float = 0.1  # comm-cost weight


class DistributedConfig(_Strict):
    """Multi-process backend settings.

    The reference used ZMQ endpoints + wall-clock rounds
    (reference: config/schema.py:7-51); the RCCL backend replaces the transport
    but keeps the knobs that still make sense, plus rendezvous settings.

    Reference YAMLs carrying ZMQ-era keys (``transport``, ``ipc_dir``,
    ``host``, ``coordinator_*_port``, ``base_port``, ``node_hosts``,
    ``startup_grace_s``) are ACCEPTED and ignored with a warning — the YAML
    schema is an explicit compatibility surface (extend
    additively without breaking existing keys). ``round_duration_s`` maps
    through to the straggler budget.
    """

    # rendezvous for torch.distributed (one process per node/GPU)
    master_addr: str = "127.0.0.1"
    master_port: int = 29511
    # communication backend: "nccl" is RCCL on ROCm; "gloo" for CPU tests
    comm_backend: Literal["auto", "nccl", "gloo"] = "auto"
    # straggler semantics: intra-box RCCL is deterministic-synchronous; a
    # wall-clock round budget is kept for parity experiments only (0 = off).
    # Maps 1:1 from the reference's wall-clock round window
    # (reference: config/schema.py:46-51).
    round_duration_s: float = 0.0
    # overlap neighbor exchange with eval/scoring compute on a side stream
    overlap_exchange: bool = True
    # exchange sketches first, full states only with accepted neighbors
    # (sketchguard wire-compression mode; reference kept it latent,
    # sketchguard.py:114-132)
    sketch_wire_mode: bool = False
    # out-of-band metrics: each rank appends rows to
    # <metrics_dir>/metrics_rank<k>.jsonl (local write, no collective) and
    # the in-band gather is skipped — the reference Monitor's passive,
    # crash-tolerant property (monitor.py:1-15). None = in-band gather.
    metrics_dir: Optional[str] = None

    # ---- ZMQ-era keys (reference: config/schema.py:10-51): accepted for
    # compatibility, IGNORED by the RCCL backend (warned at load time).
    transport: Optional[Literal["ipc", "tcp"]] = None
    ipc_dir: Optional[str] = None
    host: Optional[str] = None
    coordinator_pub_port: Optional[int] = None
    coordinator_pull_port: Optional[int] = None
    base_port: Optional[int] = None
    node_hosts: Optional[Dict[int, str]] = None
    startup_grace_s: Optional[float] = None

    _ZMQ_LEGACY_KEYS = (
        "transport", "ipc_dir", "host", "coordinator_pub_port",
        "coordinator_pull_port", "base_port", "node_hosts", "startup_grace_s",
    )

    @model_validator(mode="after")
    def _warn_legacy(self) -> "DistributedConfig":
        legacy = [k for k in self._ZMQ_LEGACY_KEYS if getattr(self, k) is not None]
        if legacy:
            import warnings

            warnings.warn(
                f"distributed: ZMQ-era keys {legacy} are accepted for "
                "reference-YAML compatibility but ignored by the RCCL backend",
                UserWarning,
                stacklevel=2,
            )
        return self


class ComputeConfig(_Strict):
    """MI355X-native additions (not in the reference schema)."""

    dtype: Literal["fp32", "bf16"] = "fp32"
    # pin node i to cuda:(i % num_devices); "auto" = cuda if available else cpu
    device: str = "auto"
    # use fused HIP kernels when on GPU (fail loudly if extension missing)
    native_kernels: bool = True
    # force MIOpen deterministic conv algorithms (8-17x slower on MI355X;
    # seeded RNGs alone already reproduce training curves)
    deterministic_kernels: bool = False
    # NHWC weights/activations for conv models (removes MIOpen's internal
    # batched_transpose kernels on CDNA4)
    channels_last: bool = True


class Config(_Strict):
    experiment: ExperimentConfig = Field(default_factory=ExperimentConfig)
    topology: TopologyConfig = Field(default_factory=TopologyConfig)
    aggregation: AggregationConfig = Field(default_factory=AggregationConfig)
    attack: AttackConfig = Field(default_factory=AttackConfig)
    training: TrainingConfig = Field(default_factory=TrainingConfig)
    data: DataConfig = Field(default_factory=DataConfig)
    model: ModelConfig = Field(default_factory=ModelConfig)
    backend: Literal["simulation", "distributed", "rccl"] = "simulation"
    distributed: DistributedConfig = Field(default_factory=DistributedConfig)
    mobility: Optional[MobilityConfig] = None
    dmtt: Optional[DMTTConfig] = None
    compute: ComputeConfig = Field(default_factory=ComputeConfig)

    @model_validator(mode="after")
    def _check(self) -> "Config":
        if self.attack.enabled and not (0.0 <= self.attack.percentage <= 1.0):
            raise ValueError("attack.percentage must be in [0, 1]")
        if self.dmtt is not None and self.mobility is None:
            # DMTT scores claims against the deterministic G^t; it requires mobility
            raise ValueError("dmtt requires a mobility config (dynamic topology G^t)")
        return self
