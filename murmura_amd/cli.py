"""CLI: ``murmura run CONFIG``, ``murmura run-node``, ``murmura list-components``
(reference: murmura/cli.py:34-304).

The ``distributed``/``rccl`` backend replaces the reference's ZMQ runner with
the RCCL process-per-GPU runner (murmura_amd/parallel/).
"""

from __future__ import annotations

import json
from pathlib import Path
from typing import Optional

import typer
from rich.console import Console
from rich.table import Table

app = typer.Typer(name="murmura", help="MI355X-native decentralized federated learning")
console = Console()


@app.command()
def run(
    config_path: Path = typer.Argument(..., help="YAML/JSON config"),
    device: Optional[str] = typer.Option(None, "--device", help="device override"),
    verbose: bool = typer.Option(True, "--verbose/--quiet"),
    output: Optional[Path] = typer.Option(None, "--output", help="write history JSON here"),
    checkpoint: Optional[Path] = typer.Option(
        None, "--checkpoint", help="checkpoint file (written every --checkpoint-every rounds)"
    ),
    checkpoint_every: int = typer.Option(0, "--checkpoint-every"),
    resume: bool = typer.Option(False, "--resume", help="resume from --checkpoint"),
) -> None:
    """Run an experiment from a config (simulation or distributed backend)."""
    from murmura_amd.config.loader import load_config

    config = load_config(config_path)
    if device is not None:
        config.compute.device = device
    if config.backend == "simulation":
        history = _run_simulation(config, verbose, checkpoint=checkpoint,
                                  checkpoint_every=checkpoint_every, resume=resume)
    else:
        from murmura_amd.parallel.runner import DistributedRunner

        history = DistributedRunner(config).run()
    if output is not None:
        output.write_text(json.dumps(history, indent=2))
    _display_results(history, config)


def _run_simulation(config, verbose: bool, checkpoint=None, checkpoint_every=0,
                    resume=False):
    import torch

    from murmura_amd.core.network import Network
    from murmura_amd.utils import factories, seed

    seed.set_seed(config.experiment.seed,
                  deterministic_kernels=config.compute.deterministic_kernels)
    model_factory = factories.build_model_factory(config)
    adapter = factories.build_dataset_adapter(config)
    agg_factory = factories.build_aggregator_factory(config, model_factory)
    criterion_factory = factories.build_criterion_factory(config)
    evidential = factories.is_evidential(config)
    device = None
    if config.compute.device not in (None, "auto"):
        device = torch.device(config.compute.device)
    network = Network.from_config(
        config,
        model_factory,
        adapter,
        agg_factory,
        device=device,
        criterion_factory=criterion_factory,
        evidential=evidential,
    )
    if verbose:
        console.print(
            f"[bold green]murmura-amd[/bold green] experiment "
            f"'{config.experiment.name}': {config.topology.num_nodes} nodes, "
            f"{config.topology.type} topology, {config.aggregation.algorithm}, "
            f"backend=simulation"
        )
    start_round = 0
    if resume and checkpoint is not None:
        start_round = network.resume_from(str(checkpoint))
        if verbose:
            console.print(f"resumed from {checkpoint} at round {start_round}")
    return network.train(
        rounds=config.experiment.rounds,
        local_epochs=config.training.local_epochs,
        lr=config.training.lr,
        verbose=verbose and config.experiment.verbose,
        checkpoint_path=str(checkpoint) if checkpoint else None,
        checkpoint_every=checkpoint_every,
        start_round=start_round,
    )


@app.command("run-node")
def run_node(
    config_path: Path = typer.Argument(...),
    node_id: Optional[int] = typer.Option(
        None, "--node-id", "-n",
        help="rank of this node; defaults to the RANK env var (torchrun)",
    ),
    world_size: Optional[int] = typer.Option(None, "--world-size"),
    master_addr: Optional[str] = typer.Option(None, "--master-addr"),
    master_port: Optional[int] = typer.Option(None, "--master-port"),
) -> None:
    """Launch a single FL node process (multi-machine / manual / torchrun)."""
    import os

    from murmura_amd.config.loader import load_config
    from murmura_amd.parallel.node_process import run_node_process

    config = load_config(config_path)
    if master_addr:
        config.distributed.master_addr = master_addr
    if master_port:
        config.distributed.master_port = master_port
    if node_id is None:
        env_rank = os.environ.get("RANK")
        if env_rank is None:
            raise typer.BadParameter("--node-id required (or set RANK)")
        node_id = int(env_rank)
    ws = world_size or int(os.environ.get("WORLD_SIZE", config.topology.num_nodes))
    history = run_node_process(config, rank=node_id, world_size=ws)
    if history is not None:
        _display_results(history, config)


@app.command("list-components")
def list_components(
    kind: str = typer.Argument("all", help="topologies|aggregators|attacks|backends|models|all")
) -> None:
    listings = {
        "topologies": ["ring", "fully", "erdos", "k-regular", "(+ mobility G^t via mobility: block)"],
        "aggregators": ["fedavg", "krum", "balance", "sketchguard", "ubar", "evidential_trust"],
        "attacks": ["gaussian", "directed_deviation", "topology_liar"],
        "backends": ["simulation (in-process oracle)", "rccl (one process per MI355X GPU)"],
        "models": [
            "models.mlp", "models.femnist", "models.celeba", "models.resnet18",
            "examples.leaf.*", "examples.wearables.*",
        ],
    }
    kinds = list(listings) if kind == "all" else [kind]
    for k in kinds:
        if k not in listings:
            console.print(f"[red]unknown component kind: {k}[/red]")
            raise typer.Exit(1)
        console.print(f"[bold]{k}[/bold]")
        for item in listings[k]:
            console.print(f"  - {item}")


def _display_results(history, config) -> None:
    table = Table(title=f"Results: {config.experiment.name}")
    cols = ["round", "mean_accuracy", "std_accuracy", "mean_loss", "honest_accuracy"]
    evidential = any(v != 0.0 for v in history.get("mean_vacuity", []))
    if evidential:
        cols += ["mean_vacuity", "mean_entropy", "mean_strength"]
    for c in cols:
        table.add_column(c)
    for i in range(len(history["round"])):
        table.add_row(*[f"{history[c][i]:.4f}" if c != "round" else str(history[c][i]) for c in cols])
    console.print(table)


def main() -> None:
    app()


if __name__ == "__main__":
    main()
