"""Run configuration (reference internals/config.py:65-140 behavior).

Environment variables:
  PATHWAY_THREADS / PATHWAY_PROCESSES — worker counts (reference semantics)
  PW_DEVICE — engine device override ("cpu", "cuda", "cuda:0", ...)
On GPU hosts the engine defaults to cuda (one worker per GPU, RCCL
exchange); in CPU-only containers it runs on torch-cpu tensors.
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field


@dataclass
class PathwayConfig:
    device: str | None = None
    ignore_asserts: bool = bool(os.environ.get("PATHWAY_IGNORE_ASSERTS"))
    terminate_on_error: bool = True
    runtime_typechecking: bool = False
    license_key: str | None = os.environ.get("PATHWAY_LICENSE_KEY")
    monitoring_server: str | None = None
    detailed_metrics_dir: str | None = None
    process_id: int = int(os.environ.get("PATHWAY_PROCESS_ID", "0"))
    processes: int = int(os.environ.get("PATHWAY_PROCESSES", "1"))
    threads: int = int(os.environ.get("PATHWAY_THREADS", "1"))

    def resolve_device(self) -> str:
        if self.device is not None:
            return self.device
        env = os.environ.get("PW_DEVICE")
        if env:
            return env
        try:
            import torch

            if torch.cuda.is_available():
                local_rank = int(os.environ.get("LOCAL_RANK", "0"))
                return f"cuda:{local_rank % max(torch.cuda.device_count(), 1)}"
        except Exception:
            pass
        return "cpu"


pathway_config = PathwayConfig()


def set_device(device: str | None) -> None:
    pathway_config.device = device


def get_device() -> str:
    return pathway_config.resolve_device()


def set_license_key(key: str | None) -> None:
    pathway_config.license_key = key


def set_monitoring_config(*, server_endpoint: str | None = None,
                          detailed_metrics_dir: str | None = None,
                          **kwargs) -> None:
    pathway_config.monitoring_server = server_endpoint
    pathway_config.detailed_metrics_dir = detailed_metrics_dir
