"""DPO interface (reference: realhf/impl/model/interface/dpo_interface.py:99
+ dpo_functional.py:30).

Two MFCs: ref_inf computes per-sequence log-probs of pos/neg under the
frozen ref model ("seqlogp"); dpo_train computes the DPO loss against them.
Samples pack [pos, neg] pairs (2 seqlens per sample).
"""
import dataclasses
from typing import Dict

import torch

from realhf_amd.api.data import SequenceSample
from realhf_amd.api.model import Model, ModelInterface, register_interface
from realhf_amd.parallel.tp import packed_shifted_logprobs
from realhf_amd.runtime.engine import sample_to_packed
from realhf_amd.utils.functional import build_shift_one_indices


def _seq_logps(logits, mb: SequenceSample) -> torch.Tensor:
    """Sum of answer-token logprobs per sequence -> [n_seqs_total]."""
    ids, cu, _ = sample_to_packed(mb)
    logp = packed_shifted_logprobs(logits, cu, ids)
    shift = build_shift_one_indices(ids.shape[0], cu)
    if "prompt_mask" in mb.keys:
        mask = (~mb.data["prompt_mask"].bool()[shift]).float()
    else:
        mask = torch.ones_like(logp)
    logp = logp * mask
    bs = cu.shape[0] - 1
    scu = cu - torch.arange(bs + 1, device=cu.device, dtype=cu.dtype)
    seg = torch.bucketize(
        torch.arange(logp.shape[0], device=logp.device), scu[1:].long(), right=True
    )
    out = torch.zeros(bs, dtype=logp.dtype, device=logp.device)
    out.scatter_add_(0, seg, logp)
    return out


@dataclasses.dataclass
class DPOInterface(ModelInterface):
    beta: float = 0.1

    @torch.no_grad()
    def inference(self, model: Model, data: SequenceSample, n_mbs=None):
        def post_hook(logits, mb):
            return _seq_logps(logits, mb)

        seqlogp = model.module.forward(data, n_mbs=n_mbs, post_hook=post_hook)
        if seqlogp is None:  # pp mid stage
            return None
        # 2 sequences (pos, neg) per sample
        return SequenceSample(
            keys=("seqlogp",),
            ids=list(data.ids),
            seqlens={"seqlogp": [[1, 1]] * data.bs},
            data={"seqlogp": seqlogp.float()},
        )

    def _loss_fn(self, logits, mb: SequenceSample):
        seqlogp = _seq_logps(logits, mb)
        pi_pos, pi_neg = seqlogp[0::2], seqlogp[1::2]
        ref = mb.data["seqlogp"]
        ref_pos, ref_neg = ref[0::2], ref[1::2]
        logits_diff = self.beta * ((pi_pos - ref_pos) - (pi_neg - ref_neg))
        loss = -torch.nn.functional.logsigmoid(logits_diff).mean()
        return loss, {
            "loss": float(loss.detach()),
            "acc": float((logits_diff > 0).float().mean()),
            "margin": float(logits_diff.detach().mean()),
        }

    def train_step(self, model: Model, data: SequenceSample, n_mbs=None) -> Dict:
        stats = model.module.train_batch(
            data, self._loss_fn, version_steps=model.version.global_step, n_mbs=n_mbs
        )
        model.inc_version()
        return stats

    def save(self, model: Model, save_dir: str):
        from realhf_amd.models import hf as hf_reg

        m = model.module.module if hasattr(model.module, "module") else model.module
        hf_reg.save_to_hf(m, m.config.family or "llama", save_dir, model.tokenizer)


register_interface("dpo", DPOInterface)
