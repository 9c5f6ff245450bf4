"""Console observability: rich panels for experiment/cycle info
(reference: utils/harness_utils.py:248-351 display_training_info) and a
working reset_optimizer (dead code in the reference,
harness_utils.py:24-46 — optimizer freshness normally comes from
rebuilding per level; this utility supports explicit rewind flows)."""

from __future__ import annotations

import os
from typing import Any, Dict, Optional

import torch


def display_training_info(config_info: Dict[str, Any],
                          optimizer_info: Dict[str, Any],
                          cycle_info: Optional[Dict[str, Any]] = None,
                          training_info: Optional[Dict[str, Any]] = None
                          ) -> None:
    """Print experiment configuration panels (rich when available)."""
    try:
        from rich.console import Console
        from rich.panel import Panel
        from rich.tree import Tree

        console = Console()

        def tree_of(title, info):
            t = Tree(f"[bold]{title}[/bold]")
            for k, v in info.items():
                if "expt_dir" in str(k):
                    v = os.path.basename(str(v))
                t.add(f"[cyan]{k}[/cyan]: [yellow]{v}[/yellow]")
            return t

        console.print(Panel(tree_of("Training Harness Configuration",
                                    config_info),
                            title="Hardware Configuration",
                            border_style="cyan"))
        if training_info:
            console.print(Panel(tree_of("Experiment Configuration",
                                        training_info),
                                title="Training Configuration",
                                border_style="cyan"))
        if cycle_info:
            console.print(Panel(tree_of("Cycle", cycle_info),
                                title="Current Cycle Information",
                                border_style="cyan"))
        console.print(Panel(tree_of("Optimizer Configuration",
                                    optimizer_info),
                            title="Optimizer Details", border_style="cyan"))
    except ImportError:
        for name, info in (("config", config_info),
                           ("training", training_info),
                           ("cycle", cycle_info),
                           ("optimizer", optimizer_info)):
            if info:
                print(f"--- {name} ---")
                for k, v in info.items():
                    print(f"  {k}: {v}")


def reset_optimizer(expt_dir: str, optimizer: torch.optim.Optimizer,
                    training_type: str) -> torch.optim.Optimizer:
    """Load the optimizer state matching the training type: imp/lrr ->
    artifacts/optimizer_init.pt, wr -> artifacts/optimizer_rewind.pt."""
    if training_type in ("imp", "lrr"):
        path = os.path.join(expt_dir, "artifacts", "optimizer_init.pt")
    elif training_type == "wr":
        path = os.path.join(expt_dir, "artifacts", "optimizer_rewind.pt")
    else:
        return optimizer
    optimizer.load_state_dict(
        torch.load(path, map_location="cpu", weights_only=False))
    return optimizer
