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
    # Dump generations as jsonl under LOG_ROOT (reference:
    # gen_interface.py:86-152 output_file).  One file per dp rank
    # (suffix .rankN) instead of the reference's lock-file serialization
    # — same records, race-free.
    output_file: str = None

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
        records = []
        for gen_out, prompts, cu in outs:
            packed, cu_full, pmask = concat_prompt_to_generation_output(
                prompts, cu, gen_out
            )
            all_ids.append(packed)
            all_pm.append(pmask)
            seqlens += [int(cu_full[i + 1] - cu_full[i]) for i in range(cu.shape[0] - 1)]
            if self.output_file is not None:
                for i in range(cu.shape[0] - 1):
                    p = prompts[int(cu[i]):int(cu[i + 1])]
                    gl = int(gen_out.gen_lengths[i])
                    a = gen_out.gen_tokens[i, :gl]
                    records.append((p.cpu(), a.cpu()))
        if records:
            self._dump(model, data.ids, records)
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

    def _dump(self, model: Model, ids, records):
        import json
        import os

        from realhf_amd.base import constants

        if constants.has_current() and (
            constants.tp_rank() != 0 or not constants.is_last_pipe_stage()
        ):
            return  # one writer per dp shard
        root = constants.LOG_ROOT(constants.experiment_name(),
                                  constants.trial_name())
        os.makedirs(root, exist_ok=True)
        dp = constants.dp_rank() if constants.has_current() else 0
        path = os.path.join(root, f"{self.output_file}.rank{dp}")
        tok = model.tokenizer
        with open(path, "a") as f:
            for _id, (p, a) in zip(ids, records):
                if tok is not None:
                    rec = dict(
                        id=str(_id),
                        prompt=tok.decode(p.tolist(), skip_special_tokens=True),
                        answer=tok.decode(a.tolist(), skip_special_tokens=True),
                        seq=tok.decode(torch.cat([p, a]).tolist(),
                                       skip_special_tokens=True),
                    )
                else:  # no tokenizer (synthetic runs): dump token ids
                    rec = dict(id=str(_id), prompt_ids=p.tolist(),
                               answer_ids=a.tolist())
                f.write(json.dumps(rec) + "\n")


register_interface("generation", GenerationInterface)
