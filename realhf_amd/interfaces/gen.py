"""Generation-only interface (reference:
realhf/impl/model/interface/gen_interface.py, registered "generation")."""
import dataclasses

import torch

from realhf_amd.api.data import SequenceSample
from realhf_amd.api.model import (
    GenerationHyperparameters,
    Model,
    ModelInterface,
    register_interface,
)
from realhf_amd.models.generation import concat_prompt_to_generation_output


@dataclasses.dataclass
class GenerationInterface(ModelInterface):
    gconfig: GenerationHyperparameters = dataclasses.field(
        default_factory=GenerationHyperparameters
    )

    def __post_init__(self):
        if isinstance(self.gconfig, dict):
            self.gconfig = GenerationHyperparameters(**self.gconfig)

    def generate(self, model: Model, data: SequenceSample, n_mbs=None):
        outs = model.module.generate(
            data, tokenizer=model.tokenizer, gconfig=self.gconfig, n_mbs=n_mbs
        )
        if outs is None:  # pp mid stage
            return None
        all_ids, all_pm, seqlens = [], [], []
        for gen_out, prompts, cu in outs:
            packed, cu_full, pmask = concat_prompt_to_generation_output(
                prompts, cu, gen_out
            )
            all_ids.append(packed)
            all_pm.append(pmask)
            seqlens += [int(cu_full[i + 1] - cu_full[i]) for i in range(cu.shape[0] - 1)]
        return SequenceSample(
            keys=("packed_input_ids", "prompt_mask"),
            ids=list(data.ids),
            seqlens={
                "packed_input_ids": [[l] for l in seqlens],
                "prompt_mask": [[l] for l in seqlens],
            },
            data={
                "packed_input_ids": torch.cat(all_ids),
                "prompt_mask": torch.cat(all_pm),
            },
        )


register_interface("generation", GenerationInterface)
