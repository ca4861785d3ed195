"""Model plugin surface.

Byte-compatible mirror of
/root/reference/cosmos_curate/core/interfaces/model_interface.py:20-55:
same class name, same abstract surface (conda_env_name, model_id_names,
setup), same meaning.  On the MI355X rebuild there is a single ROCm
environment, so implementations return a constant env name (SURVEY.md §2
"Env system": keep the API, collapse the envs).
"""

from __future__ import annotations

import abc


class ModelInterface(abc.ABC):
    """Interface for models used inside pipeline stages.

    Mirrors model_interface.py:20-55: weight handling + environment setup;
    no restriction on how inference runs.
    """

    @property
    @abc.abstractmethod
    def conda_env_name(self) -> str:
        """Environment the model must run in (constant on the rebuild)."""

    @property
    @abc.abstractmethod
    def model_id_names(self) -> list[str]:
        """Model IDs (HF-style names) used by this model."""

    @abc.abstractmethod
    def setup(self) -> None:
        """Load weights / build the model.  Runs in the worker process."""
