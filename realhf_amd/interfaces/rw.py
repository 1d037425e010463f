"""Paired reward-model interface (reference:
realhf/impl/model/interface/rw_interface.py, registered "paired_rw").

Training data: each sample packs [pos_seq, neg_seq] (2 seqlens per sample
under key "packed_input_ids").  Loss: Bradley-Terry -logsigmoid(r+ - r-)
on end-of-sequence scores.  Inference: per-sequence score at the last
token -> "rewards".
"""
import dataclasses
from typing import Dict

import torch

from realhf_amd.api.data import SequenceSample
from realhf_amd.api.model import Model, ModelInterface, register_interface
from realhf_amd.runtime.engine import sample_to_packed


def _end_scores(values: torch.Tensor, cu: torch.Tensor) -> torch.Tensor:
    ends = cu[1:].long() - 1
    return values.squeeze(-1).float()[ends]


@dataclasses.dataclass
class PairedRewardInterface(ModelInterface):
    output_scaling: float = 1.0
    output_bias: float = 0.0

    def _loss_fn(self, values, mb: SequenceSample):
        ids, cu, _ = sample_to_packed(mb)
        scores = _end_scores(values, cu)  # [2*bs]: interleaved pos/neg per sample
        pos, neg = scores[0::2], scores[1::2]
        loss = -torch.nn.functional.logsigmoid(pos - neg).mean()
        acc = float((pos > neg).float().mean())
        return loss, {
            "loss": float(loss.detach()),
            "acc": acc,
            "pos_score": float(pos.detach().mean()),
            "neg_score": float(neg.detach().mean()),
        }

    def train_step(self, model: Model, data: SequenceSample, n_mbs=None) -> Dict:
        stats = model.module.train_batch(
            data, self._loss_fn, version_steps=model.version.global_step, n_mbs=n_mbs
        )
        model.inc_version()
        return stats

    @torch.no_grad()
    def inference(self, model: Model, data: SequenceSample, n_mbs=None):
        def post_hook(values, mb):
            _, cu, _ = sample_to_packed(mb)
            return _end_scores(values, cu)

        scores = model.module.forward(data, n_mbs=n_mbs, post_hook=post_hook)
        if scores is None:  # pp mid stage
            return None
        scores = (scores - self.output_bias) * self.output_scaling
        return SequenceSample(
            keys=("rewards",),
            ids=list(data.ids),
            seqlens={"rewards": [[1]] * data.bs},
            data={"rewards": scores.float()},
        )

    def save(self, model: Model, save_dir: str):
        from realhf_amd.models import hf as hf_reg

        m = model.module.module if hasattr(model.module, "module") else model.module
        hf_reg.save_to_hf(m, m.config.family or "llama", save_dir, model.tokenizer)


register_interface("paired_rw", PairedRewardInterface)
