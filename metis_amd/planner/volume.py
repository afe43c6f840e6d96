"""GPT activation/parameter volume model.

Parity: reference model/activation_parameter.py:5-51. Layer ids follow the
profile convention: layer 0 = input embedding, layers 1..n-2 = transformer
blocks, layer n-1 = output head.

``activation_dtype_bytes`` defaults to 1 (element counts — reference
quirk Q8: its "bytes" are really element counts and the p2p cost divides
them by bandwidth directly). Set 2 for true bf16 bytes when the cluster
bandwidths are real measured GB/s.
"""

from __future__ import annotations

from typing import List, Sequence

from metis_amd.config import ModelConfig


class GPTVolume:
    def __init__(
        self,
        model_config: ModelConfig,
        parameters_per_layer_bytes: Sequence[float],
        activation_dtype_bytes: int = 1,
    ) -> None:
        self.config = model_config
        self.input_params = float(parameters_per_layer_bytes[0])
        self.transformer_params = float(parameters_per_layer_bytes[1])
        self.output_params = float(parameters_per_layer_bytes[-1])
        self.act_bytes = activation_dtype_bytes

    @property
    def num_layers(self) -> int:
        return self.config.num_layers

    def activation_size(self, layer_id: int, batch_size: int, tp_deg: int) -> float:
        """Activation volume crossing the stage boundary after ``layer_id``."""
        c = self.config
        if layer_id == c.num_layers - 1:
            return batch_size * c.sequence_length * c.vocab_size / tp_deg * self.act_bytes
        return batch_size * c.sequence_length * c.hidden_size * self.act_bytes

    def parameter_sizes(self, tp_deg: int) -> List[float]:
        """Per-layer parameter bytes under TP sharding."""
        n = self.config.num_layers
        return (
            [self.input_params / tp_deg]
            + [self.transformer_params / tp_deg] * (n - 2)
            + [self.output_params / tp_deg]
        )

    def stage_parameter_size(self, tp_deg: int, start_layer: int, end_layer: int) -> float:
        """Total parameter bytes of layers [start_layer, end_layer) under TP."""
        n_transformer = end_layer - start_layer
        total = 0.0
        if start_layer == 0:
            total += self.input_params / tp_deg
            n_transformer -= 1
        if end_layer == self.config.num_layers:
            total += self.output_params / tp_deg
            n_transformer -= 1
        total += self.transformer_params / tp_deg * n_transformer
        return total


class MoEVolume(GPTVolume):
    """MoE (EP axis) volume model — an MI355X extension with no
    reference counterpart (the reference models dense GPT only,
    model/activation_parameter.py).

    EP degree == TP degree in the runtime (models/moe.py): experts
    shard across the group while the router (fp32) and norms stay
    replicated on every rank, so per-rank parameter bytes are
    ``replicated + (total - replicated) / tp`` rather than the dense
    ``total / tp``. Activations crossing stage boundaries are the dense
    ``bs*seq*hidden`` (the top-k expert combine happens inside the
    block), so activation_size is inherited.
    """

    def __init__(
        self,
        model_config: ModelConfig,
        parameters_per_layer_bytes: Sequence[float],
        activation_dtype_bytes: int = 1,
    ) -> None:
        super().__init__(model_config, parameters_per_layer_bytes,
                         activation_dtype_bytes)
        h = model_config.hidden_size
        # replicated per block: fp32 router [E, h] + bias, 2 LayerNorms
        # (w+b), row-parallel proj bias — bytes
        self.block_replicated = float(
            (h * model_config.num_experts + model_config.num_experts) * 4
            + (4 * h + h) * 2)
        self.block_replicated = min(self.block_replicated,
                                    self.transformer_params)

    def parameter_sizes(self, tp_deg: int) -> List[float]:
        n = self.config.num_layers
        block = (self.block_replicated
                 + (self.transformer_params - self.block_replicated) / tp_deg)
        # embedding replicated across TP in the runtime; head sharded
        return ([self.input_params]
                + [block] * (n - 2)
                + [self.output_params / tp_deg])

    def stage_parameter_size(self, tp_deg: int, start_layer: int,
                             end_layer: int) -> float:
        sizes = self.parameter_sizes(tp_deg)
        return float(sum(sizes[start_layer:end_layer]))


def make_volume(
    model_config: ModelConfig,
    parameters_per_layer_bytes: Sequence[float],
    activation_dtype_bytes: int = 1,
):
    """GPTVolume for dense configs, MoEVolume when num_experts > 0."""
    cls = MoEVolume if model_config.num_experts > 0 else GPTVolume
    return cls(model_config, parameters_per_layer_bytes,
               activation_dtype_bytes)


def uniform_layer_split(total_layers: int, num_stages: int) -> List[int]:
    """Even split of (total-2) transformer layers over stages, remainder to
    stages 1..r, +1 (embed/head) on first and last stage.
    Parity: model/utils.py:5-31."""
    base = (total_layers - 2) // num_stages
    rem = (total_layers - 2) % num_stages
    counts = [base] * num_stages
    for i in range(1, rem + 1):
        counts[i] += 1
    counts[0] += 1
    counts[-1] += 1
    return counts
