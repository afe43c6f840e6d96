"""Model families built on the metis_amd MI355X ops."""

from metis_amd.models.gpt import GPTModel, GPTModelSpec, MODEL_SPECS

__all__ = ["GPTModel", "GPTModelSpec", "MODEL_SPECS"]
