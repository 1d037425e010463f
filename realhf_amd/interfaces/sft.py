"""SFT interface (reference: realhf/impl/model/interface/sft_interface.py:87).

Loss: next-token CE over the answer tokens (prompt tokens masked via
prompt_mask).  Works on vocab-parallel logits.
"""
import dataclasses
from typing import Dict

import torch

from realhf_amd.api.data import SequenceSample
from realhf_amd.api.model import Model, ModelInterface, register_interface
from realhf_amd.models import moe as moe_mod
from realhf_amd.parallel.tp import packed_shifted_logprobs
from realhf_amd.runtime.engine import sample_to_packed
from realhf_amd.utils.functional import build_shift_one_indices


def sft_loss_fn(logits: torch.Tensor, mb: SequenceSample):
    ids, cu, _ = sample_to_packed(mb)
    logp = packed_shifted_logprobs(logits, cu, ids)
    if "prompt_mask" in mb.keys:
        pm = mb.data["prompt_mask"]
        shift = build_shift_one_indices(ids.shape[0], cu)
        loss_mask = (~pm[shift].bool()).float()
    else:
        loss_mask = torch.ones_like(logp)
    n = loss_mask.sum().clamp(min=1)
    loss = -(logp * loss_mask).sum() / n
    for aux in moe_mod.pop_aux_losses():
        loss = loss + aux
    return loss, {"loss": float(loss.detach()), "n_tokens": float(n)}


@dataclasses.dataclass
class SFTInterface(ModelInterface):
    def train_step(self, model: Model, data: SequenceSample, n_mbs=None) -> Dict:
        stats = model.module.train_batch(
            data, sft_loss_fn, version_steps=model.version.global_step, n_mbs=n_mbs
        )
        model.inc_version()
        return stats

    def evaluate(self, model: Model, eval_dataloader) -> Dict:
        losses, n = 0.0, 0
        for batch in eval_dataloader:
            st = model.module.eval_batch(batch.to_device(model.device), sft_loss_fn)
            losses += st.get("loss", 0.0)
            n += 1
        return {"eval_loss": losses / max(1, n)}

    def save(self, model: Model, save_dir: str):
        from realhf_amd.models import hf as hf_reg

        m = model.module.module if hasattr(model.module, "module") else model.module
        hf_reg.save_to_hf(m, m.config.family or "llama", save_dir, model.tokenizer)


register_interface("sft", SFTInterface)
